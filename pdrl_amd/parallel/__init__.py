from .ddp import GradReducer, init_distributed, distributed_env  # noqa: F401
