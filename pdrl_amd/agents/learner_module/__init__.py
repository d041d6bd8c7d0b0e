"""Algorithm updaters + the (algo → updater, model) switchboard
(capability parity with the reference's module_switcher, main.py:99-116,
and the agents/__init__.py wrapper decorators)."""
from .ppo import PPOUpdater  # noqa: F401
from .impala import ImpalaUpdater  # noqa: F401
from .v_mpo import VMPOUpdater  # noqa: F401
from .sac import SACUpdater  # noqa: F401
from .sac_continuous import SACContinuousUpdater  # noqa: F401
from . import compute_loss  # noqa: F401

from pdrl_amd.networks import (
    MlpLSTMSingle,
    MlpLSTMSingleContinuous,
    MlpLSTMSeperate,
    MlpLSTMSeperateContinuous,
)

# algo name (as found in parameters.json "algo") → (updater cls, model cls)
MODULE_SWITCHER = {
    "PPO": (PPOUpdater, MlpLSTMSingle),
    "PPO-Continuous": (PPOUpdater, MlpLSTMSingleContinuous),
    "IMPALA": (ImpalaUpdater, MlpLSTMSingle),
    "V-MPO": (VMPOUpdater, MlpLSTMSingle),
    "SAC": (SACUpdater, MlpLSTMSeperate),
    "SAC-Continuous": (SACContinuousUpdater, MlpLSTMSeperateContinuous),
}

ON_POLICY_ALGOS = {"PPO", "PPO-Continuous", "IMPALA", "V-MPO"}


def switch_module(algo: str):
    if algo not in MODULE_SWITCHER:
        raise ValueError(f"unknown algo '{algo}'; available: {sorted(MODULE_SWITCHER)}")
    return MODULE_SWITCHER[algo]


def is_on_policy(algo: str) -> bool:
    return algo in ON_POLICY_ALGOS
