from .env_maker import EnvBase  # noqa: F401
from .worker import Worker  # noqa: F401
from .manager import Manager, storage_shard_ports  # noqa: F401
from .learner_storage import LearnerStorage  # noqa: F401
from .learner import Learner, BatchStager, find_latest_checkpoint  # noqa: F401
