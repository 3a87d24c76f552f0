"""Neuroevolution problem domains (L3). Reference parity:
/root/reference/src/evotorch/neuroevolution/__init__.py."""

from .runningnorm import ObsNormLayer, RunningNorm
from .synthetic import SyntheticRolloutProblem
from .synthetic_env import SyntheticEnvSpec, rollout_eager

__all__ = [
    "ObsNormLayer",
    "RunningNorm",
    "SyntheticEnvSpec",
    "SyntheticRolloutProblem",
    "rollout_eager",
]

try:  # full NEProblem family lands with the net/ infrastructure
    from .neproblem import NEProblem  # noqa: F401
    from .supervisedne import SupervisedNE  # noqa: F401

    __all__ += ["NEProblem", "SupervisedNE"]
except ImportError:
    pass
