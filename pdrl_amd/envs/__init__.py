from .base import Space, Discrete, Box, make  # noqa: F401
from .cartpole import CartPoleEnv  # noqa: F401
from .mountain_car import MountainCarContinuousEnv  # noqa: F401
from .fake import FakeEnv  # noqa: F401
