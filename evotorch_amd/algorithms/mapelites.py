"""MAPElites: quality-diversity / illumination search over a feature grid.

Reference parity: /root/reference/src/evotorch/algorithms/mapelites.py
(MAPElites :70, `_step` :380, `make_feature_grid` :404, cell-assignment
kernel :24-68). The population holds one slot per hypergrid cell; each
generation the operator pipeline proposes children, and every cell keeps
the best solution whose features fall inside its box — computed as one
batched (cells × solutions) masked argmax on device (K9 in SURVEY.md §2.9).
"""

from typing import Iterable, Optional

import torch

from ..core import Problem, SolutionBatch
from .ga import ExtendedPopulationMixin
from .searchalgorithm import SearchAlgorithm, SinglePopulationAlgorithmMixin

__all__ = ["MAPElites", "make_feature_grid"]


class MAPElites(SearchAlgorithm, SinglePopulationAlgorithmMixin, ExtendedPopulationMixin):
    def __init__(
        self,
        problem: Problem,
        *,
        operators: Iterable,
        feature_grid: torch.Tensor,
        re_evaluate: bool = True,
        re_evaluate_parents_first: Optional[bool] = None,
    ):
        problem.ensure_single_objective()
        if problem.eval_data_length is None or problem.eval_data_length < 1:
            raise ValueError("MAPElites requires a problem with eval_data_length >= 1 (the feature descriptors)")
        SearchAlgorithm.__init__(self, problem)
        self._init_extended(operators=operators, re_evaluate=re_evaluate, re_evaluate_parents_first=re_evaluate_parents_first)
        feature_grid = torch.as_tensor(feature_grid, dtype=problem.eval_dtype, device=problem.device)
        if feature_grid.ndim != 3 or feature_grid.shape[-1] != 2:
            raise ValueError("feature_grid must have shape (num_cells, num_features, 2) [lower, upper]")
        self._feature_grid = feature_grid
        self._population: Optional[SolutionBatch] = None
        self._filled: Optional[torch.Tensor] = None
        SinglePopulationAlgorithmMixin.__init__(self)
        self.add_status_getters({"filled": lambda: self._filled})

    @property
    def population(self) -> Optional[SolutionBatch]:
        return self._population

    @property
    def filled(self) -> Optional[torch.Tensor]:
        """Boolean vector: which cells currently hold a real solution."""
        return self._filled

    @staticmethod
    def make_feature_grid(lower_bounds, upper_bounds, num_bins, *, device=None, dtype=torch.float32) -> torch.Tensor:
        """Uniform hypergrid: returns (prod(num_bins), num_features, 2)
        cell boxes; outermost cells extend to ±inf (reference
        mapelites.py:404)."""
        lower = torch.as_tensor(lower_bounds, dtype=dtype, device=device).reshape(-1)
        upper = torch.as_tensor(upper_bounds, dtype=dtype, device=device).reshape(-1)
        nf = lower.shape[0]
        if isinstance(num_bins, int):
            bins = [num_bins] * nf
        else:
            bins = [int(b) for b in num_bins]
        per_feature = []
        for f in range(nf):
            edges = torch.linspace(lower[f], upper[f], bins[f] + 1, dtype=dtype, device=device)
            lo = edges[:-1].clone()
            hi = edges[1:].clone()
            lo[0] = float("-inf")
            hi[-1] = float("inf")
            per_feature.append(torch.stack([lo, hi], dim=-1))  # (bins, 2)
        # cartesian product of per-feature intervals
        grids = torch.meshgrid(*[torch.arange(b, device=device) for b in bins], indexing="ij")
        idx = torch.stack([g.reshape(-1) for g in grids], dim=-1)  # (cells, nf)
        cells = torch.stack([per_feature[f][idx[:, f]] for f in range(nf)], dim=1)  # (cells, nf, 2)
        return cells

    def _assign_cells(self, batch: SolutionBatch):
        """For every cell, pick the best batch row whose features fall in
        the cell box. Returns (chosen_index_per_cell, any_valid_per_cell)."""
        problem = self.problem
        nf = self._feature_grid.shape[1]
        evals = batch.access_evals()
        fitness = evals[:, 0]
        features = evals[:, 1 : 1 + nf]  # (N, F)
        lo = self._feature_grid[:, :, 0].unsqueeze(1)  # (C, 1, F)
        hi = self._feature_grid[:, :, 1].unsqueeze(1)
        sense = problem.senses[0]
        utils = fitness if sense == "max" else -fitness
        utils = torch.nan_to_num(utils.to(torch.float32), nan=float("-inf"))
        if features.is_cuda:
            # K9 HIP kernel: streamed best-in-box argmax, O(C + N) memory
            # (the eager broadcast below materializes a (C, N) matrix)
            from ..ops.dispatch import _allow_eager_on_gpu, hip_required

            if not _allow_eager_on_gpu():
                best_idx, any_valid = hip_required().mapelites_assign(self._feature_grid, features, utils)
                return best_idx, any_valid
        f = features.unsqueeze(0)  # (1, N, F)
        inside = ((f >= lo) & (f <= hi)).all(dim=-1)  # (C, N)
        masked = torch.where(inside, utils.unsqueeze(0), torch.full_like(utils, float("-inf")).unsqueeze(0).expand_as(inside))
        best_idx = masked.argmax(dim=1)  # (C,)
        any_valid = inside.any(dim=1)
        return best_idx, any_valid

    def _step(self):
        problem = self.problem
        num_cells = self._feature_grid.shape[0]
        if self._population is None:
            self._population = problem.generate_batch(num_cells)
            problem.evaluate(self._population)
            idx, valid = self._assign_cells(self._population)
            new_pop = self._population.take(idx)
            self._population = new_pop
            self._filled = valid
            # invalidate unfilled cells' evals so they lose take_best ties
            self._mask_unfilled()
            return
        extended = self._make_extended_population(self._population)
        idx, valid = self._assign_cells(extended)
        self._population = extended.take(idx)
        self._filled = valid
        self._mask_unfilled()

    def _mask_unfilled(self):
        if self._filled is None:
            return
        sense = self.problem.senses[0]
        bad = float("-inf") if sense == "max" else float("inf")
        evals = self._population.access_evals()
        evals[~self._filled, 0] = bad


# module-level alias of the staticmethod, mirroring the reference export
make_feature_grid = MAPElites.make_feature_grid
