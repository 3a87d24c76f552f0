"""Variation operators (L4). Object API in base/real/sequence, stateless
functional API in functional.py. Reference parity:
/root/reference/src/evotorch/operators/__init__.py."""

from .base import CopyingOperator, CrossOver, Operator, SingleObjOperator

__all__ = ["CopyingOperator", "CrossOver", "Operator", "SingleObjOperator"]

try:  # concrete operators land with the GA stack
    from .real import (  # noqa: F401
        CosynePermutation,
        GaussianMutation,
        MultiPointCrossOver,
        OnePointCrossOver,
        PolynomialMutation,
        SimulatedBinaryCrossOver,
        TwoPointCrossOver,
    )
    from .sequence import CutAndSplice  # noqa: F401

    __all__ += [
        "CosynePermutation",
        "GaussianMutation",
        "MultiPointCrossOver",
        "OnePointCrossOver",
        "PolynomialMutation",
        "SimulatedBinaryCrossOver",
        "TwoPointCrossOver",
        "CutAndSplice",
    ]
except ImportError:
    pass
