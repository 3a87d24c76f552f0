"""Search algorithms (L5). Reference parity:
/root/reference/src/evotorch/algorithms/__init__.py."""

from .cmaes import CMAES
from .ga import ExtendedPopulationMixin, Cosyne, GeneticAlgorithm, SteadyStateGA
from .gaussian import CEM, PGPE, SNES, XNES, GaussianSearchAlgorithm
from .graphed import GraphedSearch
from .mapelites import make_feature_grid, MAPElites
from .restarter import BIPOP, IPOP, ModifyingRestart, Restart
from .searchalgorithm import LazyReporter, LazyStatusDict, SearchAlgorithm, SinglePopulationAlgorithmMixin

try:
    from .pycmaes import PyCMAES  # requires the optional `cma` package at use time
except ImportError:  # pragma: no cover
    PyCMAES = None

__all__ = [
    "CEM",
    "CMAES",
    "Cosyne",
    "GaussianSearchAlgorithm",
    "GraphedSearch",
    "GeneticAlgorithm",
    "IPOP",
    "LazyReporter",
    "LazyStatusDict",
    "MAPElites",
    "ModifyingRestart",
    "PGPE",
    "PyCMAES",
    "Restart",
    "SNES",
    "SearchAlgorithm",
    "SinglePopulationAlgorithmMixin",
    "SteadyStateGA",
    "XNES",
]
