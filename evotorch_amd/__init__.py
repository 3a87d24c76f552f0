"""evotorch_amd — an MI355X-native evolutionary-computation framework.

A from-scratch re-design of the capabilities of nnaisense/evotorch
(reference mounted at /root/reference) for AMD Instinct MI355X (gfx950,
CDNA4): PyTorch-ROCm tensor substrate, hand-written HIP kernels for the
hot ops (population sampling, ES gradient reductions, fused updates,
batched policy rollout, non-dominated sorting), and RCCL over xGMI for
population-parallel scaling — one process per GPU, no Ray.

Layer map:
  utils/          L1 tensor substrate
  core            L2 Problem / SolutionBatch / Solution
  distributions   L2 search distributions
  operators/      L4 variation operators
  algorithms/     L5 searchers (PGPE, SNES, CMA-ES, GA/NSGA-II, ...)
  logging         L6 logger sinks
  neuroevolution/ L3 problem domains (NEProblem, SupervisedNE, VecEnvNE)
  models/         policy infrastructure (parser DSL, functional modules)
  ops/            HIP kernel dispatch (gfx950)
  parallel/       RCCL/xGMI comm layer
"""

__version__ = "0.1.0"

import logging as _py_logging
import os as _os

_verbose = {"-1": -1, "0": _py_logging.WARNING, "1": _py_logging.INFO, "2": _py_logging.DEBUG}.get(
    str(_os.environ.get("EVOTORCH_AMD_VERBOSE_LEVEL", "1"))
)
if _verbose is not None and _verbose >= 0:
    _logger = _py_logging.getLogger("evotorch_amd")
    if not _logger.handlers:
        _handler = _py_logging.StreamHandler()
        _handler.setFormatter(_py_logging.Formatter("[%(asctime)s] %(name)s %(levelname)s: %(message)s"))
        _logger.addHandler(_handler)
    _logger.setLevel(_verbose)

from . import utils
from .core import Problem, ObjectTypedProblemBoundEvaluator, ProblemBoundEvaluator, Solution, SolutionBatch

from . import algorithms, decorators, distributions, logging, models, neuroevolution, operators, ops, optimizers, parallel, testing, tools  # noqa: E402

__all__ = [
    "__version__",
    "Problem",
    "ObjectTypedProblemBoundEvaluator",
    "ProblemBoundEvaluator",
    "Solution",
    "SolutionBatch",
    "algorithms",
    "decorators",
    "distributions",
    "logging",
    "models",
    "neuroevolution",
    "operators",
    "ops",
    "optimizers",
    "parallel",
    "testing",
    "tools",
    "utils",
]
