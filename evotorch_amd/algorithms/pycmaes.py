"""PyCMAES: wrapper around the external `cma` package (ask/tell across the
numpy boundary). Reference parity:
/root/reference/src/evotorch/algorithms/pycmaes.py:39. Import-guarded: the
`cma` pip package is optional."""

from typing import Optional

import numpy as np
import torch

from ..core import Problem, SolutionBatch
from ..utils import RealOrVector, to_stdev_init
from ..utils.misc import ensure_tensor_length_and_dtype, numpy_copy
from .searchalgorithm import SearchAlgorithm, SinglePopulationAlgorithmMixin

__all__ = ["PyCMAES"]


class PyCMAES(SearchAlgorithm, SinglePopulationAlgorithmMixin):
    def __init__(
        self,
        problem: Problem,
        *,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        popsize: Optional[int] = None,
        center_init: Optional[RealOrVector] = None,
        cma_options: Optional[dict] = None,
        obj_index: Optional[int] = None,
    ):
        try:
            import cma
        except ImportError as e:
            raise ImportError("PyCMAES requires the external `cma` package (pip install cma)") from e
        problem.ensure_numeric()
        SearchAlgorithm.__init__(self, problem, center=lambda: self._center())
        self._obj_index = 0 if obj_index is None else int(obj_index)
        n = problem.solution_length
        stdev_spec = to_stdev_init(solution_length=n, stdev_init=stdev_init, radius_init=radius_init)
        sigma_vec = numpy_copy(ensure_tensor_length_and_dtype(stdev_spec, n, torch.float64, about="stdev_init"))
        if center_init is None:
            x0 = numpy_copy(problem.generate_values(1).reshape(-1), dtype=np.float64)
        else:
            x0 = numpy_copy(ensure_tensor_length_and_dtype(center_init, n, torch.float64, about="center_init"))
        opts = dict(cma_options or {})
        if popsize is not None:
            opts["popsize"] = int(popsize)
        sigma0 = float(sigma_vec.mean())
        if not np.allclose(sigma_vec, sigma0):
            opts["CMA_stds"] = (sigma_vec / sigma0).tolist()
        self._es = cma.CMAEvolutionStrategy(x0, sigma0, opts)
        self._population: Optional[SolutionBatch] = None
        SinglePopulationAlgorithmMixin.__init__(self)

    @property
    def population(self) -> Optional[SolutionBatch]:
        return self._population

    def _center(self):
        return torch.as_tensor(self._es.mean, dtype=self.problem.dtype, device=self.problem.device)

    def _step(self):
        problem = self.problem
        asked = self._es.ask()
        lam = len(asked)
        batch = SolutionBatch(problem, popsize=lam, empty=True)
        batch.access_values().copy_(torch.as_tensor(np.asarray(asked), dtype=problem.dtype, device=problem.device))
        problem.evaluate(batch)
        self._population = batch
        fitnesses = numpy_copy(batch.access_evals()[:, self._obj_index], dtype=np.float64)
        if problem.senses[self._obj_index] == "max":
            fitnesses = -fitnesses  # cma minimizes
        self._es.tell(asked, fitnesses.tolist())
