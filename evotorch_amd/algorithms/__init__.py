"""Search algorithms (L5). Reference parity:
/root/reference/src/evotorch/algorithms/__init__.py."""

from .gaussian import CEM, PGPE, SNES, XNES, GaussianSearchAlgorithm
from .searchalgorithm import LazyReporter, LazyStatusDict, SearchAlgorithm, SinglePopulationAlgorithmMixin

__all__ = [
    "CEM",
    "PGPE",
    "SNES",
    "XNES",
    "GaussianSearchAlgorithm",
    "LazyReporter",
    "LazyStatusDict",
    "SearchAlgorithm",
    "SinglePopulationAlgorithmMixin",
]
