"""Neuroevolution problem domains (L3). Reference parity:
/root/reference/src/evotorch/neuroevolution/__init__.py."""

from .deploy import load_policy, save_policy
from .neproblem import BaseNEProblem, NEProblem
from .runningnorm import ObsNormLayer, RunningNorm, RunningStat
from .supervisedne import SupervisedNE
from .synthetic import SyntheticRolloutProblem
from .synthetic_env import SyntheticEnvSpec, rollout_eager
from .vecenv import GymVectorEnvAdapter, SyntheticTorchEnv, VecEnvNE, VecGymNE

__all__ = [
    "load_policy",
    "save_policy",
    "GymVectorEnvAdapter",
    "BaseNEProblem",
    "NEProblem",
    "ObsNormLayer",
    "RunningNorm",
    "RunningStat",
    "SupervisedNE",
    "SyntheticEnvSpec",
    "SyntheticRolloutProblem",
    "SyntheticTorchEnv",
    "VecEnvNE",
    "VecGymNE",
    "rollout_eager",
]


def __getattr__(name):
    # GymNE needs gymnasium; import lazily so the package works offline
    if name == "GymNE":
        from .gymne import GymNE

        return GymNE
    if name == "ActClipLayer":
        from .gymne import ActClipLayer

        return ActClipLayer
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
