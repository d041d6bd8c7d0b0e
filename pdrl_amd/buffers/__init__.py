from .trajectory import Trajectory  # noqa: F401
from .rollout_assembler import RolloutAssembler, stack_trajectory  # noqa: F401
from .shared_ring import SharedRolloutRing, rollout_fields  # noqa: F401
