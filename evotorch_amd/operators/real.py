"""Concrete real-vector variation operators (K8 in SURVEY.md §2.9).

Reference parity: /root/reference/src/evotorch/operators/real.py:30-706.
All operators are whole-population tensor transforms (the population
dimension maps onto the GPU grid); per-row RNG comes from the problem's
generator so runs are reproducible.
"""

from typing import Optional, Union

import torch

from ..core import Problem, SolutionBatch
from ..utils import RealOrVector
from ..utils.misc import make_gaussian, make_uniform
from .base import CopyingOperator, CrossOver

__all__ = [
    "GaussianMutation",
    "MultiPointCrossOver",
    "OnePointCrossOver",
    "TwoPointCrossOver",
    "SimulatedBinaryCrossOver",
    "PolynomialMutation",
    "CosynePermutation",
]


class GaussianMutation(CopyingOperator):
    """Adds Gaussian noise; optional `mutation_probability` masks which
    genes mutate (reference real.py:30)."""

    def __init__(self, problem: Problem, *, stdev: Union[float, RealOrVector], mutation_probability: Optional[float] = None):
        super().__init__(problem)
        self._stdev = stdev
        self._p = None if mutation_probability is None else float(mutation_probability)

    def _do(self, batch: SolutionBatch) -> SolutionBatch:
        result = batch.take(torch.arange(len(batch)))
        data = result.access_values()
        noise = make_gaussian(data.shape, stdev=self._stdev, dtype=data.dtype, device=data.device, generator=self._problem.generator)
        if self._p is not None:
            mask = make_uniform(data.shape, dtype=data.dtype, device=data.device, generator=self._problem.generator) < self._p
            noise = noise * mask
        data += noise
        data.copy_(self._respect_bounds(data))
        return result


class MultiPointCrossOver(CrossOver):
    """k-point crossover via per-pair random cut masks (reference
    real.py:69: the cut-point count matrix formulation)."""

    def __init__(
        self,
        problem: Problem,
        *,
        tournament_size: int,
        num_points: Optional[int] = None,
        obj_index: Optional[int] = None,
        num_children: Optional[int] = None,
        cross_over_rate: Optional[float] = None,
    ):
        super().__init__(problem, tournament_size=tournament_size, obj_index=obj_index, num_children=num_children, cross_over_rate=cross_over_rate)
        self._num_points = int(num_points if num_points is not None else 1)
        if self._num_points < 1:
            raise ValueError("num_points must be >= 1")

    def _crossover_mask(self, num_pairs: int, length: int, device) -> torch.Tensor:
        """Boolean (num_pairs, length): True = take gene from parent2.
        Built by summing step functions at k sorted random cut points."""
        g = self._problem.generator
        gen = g if (g is not None and g.device == torch.device(device)) else None
        cuts = torch.randint(1, length, (num_pairs, self._num_points), device=device, generator=gen)
        positions = torch.arange(length, device=device).unsqueeze(0).unsqueeze(0)  # (1,1,L)
        crossed = (positions >= cuts.unsqueeze(-1)).sum(dim=1)  # (num_pairs, L): #cuts passed
        return (crossed % 2) == 1

    def _do_cross_over(self, parents1: torch.Tensor, parents2: torch.Tensor) -> SolutionBatch:
        num_pairs, length = parents1.shape
        mask = self._crossover_mask(num_pairs, length, parents1.device)
        child1 = torch.where(mask, parents2, parents1)
        child2 = torch.where(mask, parents1, parents2)
        return self._make_children_batch(torch.cat([child1, child2], dim=0))


class OnePointCrossOver(MultiPointCrossOver):
    """Single-point crossover (reference real.py:210)."""

    def __init__(self, problem: Problem, *, tournament_size: int, obj_index=None, num_children=None, cross_over_rate=None):
        super().__init__(problem, tournament_size=tournament_size, num_points=1, obj_index=obj_index, num_children=num_children, cross_over_rate=cross_over_rate)


class TwoPointCrossOver(MultiPointCrossOver):
    """Two-point crossover (reference real.py:299)."""

    def __init__(self, problem: Problem, *, tournament_size: int, obj_index=None, num_children=None, cross_over_rate=None):
        super().__init__(problem, tournament_size=tournament_size, num_points=2, obj_index=obj_index, num_children=num_children, cross_over_rate=cross_over_rate)


class SimulatedBinaryCrossOver(CrossOver):
    """SBX (Deb & Agrawal 1995); `eta` is the distribution index
    (reference real.py:391)."""

    def __init__(
        self,
        problem: Problem,
        *,
        tournament_size: int,
        eta: float,
        obj_index: Optional[int] = None,
        num_children: Optional[int] = None,
        cross_over_rate: Optional[float] = None,
    ):
        super().__init__(problem, tournament_size=tournament_size, obj_index=obj_index, num_children=num_children, cross_over_rate=cross_over_rate)
        self._eta = float(eta)

    def _do_cross_over(self, parents1: torch.Tensor, parents2: torch.Tensor) -> SolutionBatch:
        g = self._problem.generator
        gen = g if (g is not None and g.device == parents1.device) else None
        u = torch.rand(parents1.shape, dtype=parents1.dtype, device=parents1.device, generator=gen)
        betas = torch.where(
            u <= 0.5,
            (2.0 * u) ** (1.0 / (self._eta + 1.0)),
            (0.5 / (1.0 - u)) ** (1.0 / (self._eta + 1.0)),
        )
        child1 = 0.5 * ((1.0 + betas) * parents1 + (1.0 - betas) * parents2)
        child2 = 0.5 * ((1.0 - betas) * parents1 + (1.0 + betas) * parents2)
        return self._make_children_batch(torch.cat([child1, child2], dim=0))


class PolynomialMutation(CopyingOperator):
    """Polynomial mutation (Deb & Deb 2014) over bounded problems
    (reference real.py:484)."""

    def __init__(self, problem: Problem, *, eta: Optional[float] = None, mutation_probability: Optional[float] = None):
        super().__init__(problem)
        problem.ensure_numeric()
        if problem.lower_bounds is None or problem.upper_bounds is None:
            raise ValueError("PolynomialMutation requires a bounded problem")
        self._eta = float(eta) if eta is not None else 20.0
        self._p = float(mutation_probability) if mutation_probability is not None else (1.0 / problem.solution_length)

    def _do(self, batch: SolutionBatch) -> SolutionBatch:
        result = batch.take(torch.arange(len(batch)))
        x = result.access_values()
        lb = self._problem.lower_bounds.to(x.device, x.dtype)
        ub = self._problem.upper_bounds.to(x.device, x.dtype)
        g = self._problem.generator
        gen = g if (g is not None and g.device == x.device) else None
        mutate = torch.rand(x.shape, dtype=x.dtype, device=x.device, generator=gen) < self._p
        u = torch.rand(x.shape, dtype=x.dtype, device=x.device, generator=gen)
        span = ub - lb
        delta1 = (x - lb) / span
        delta2 = (ub - x) / span
        mut_pow = 1.0 / (self._eta + 1.0)
        # u <= 0.5 branch
        xy1 = 1.0 - delta1
        val1 = 2.0 * u + (1.0 - 2.0 * u) * xy1 ** (self._eta + 1.0)
        dq1 = val1**mut_pow - 1.0
        # u > 0.5 branch
        xy2 = 1.0 - delta2
        val2 = 2.0 * (1.0 - u) + 2.0 * (u - 0.5) * xy2 ** (self._eta + 1.0)
        dq2 = 1.0 - val2**mut_pow
        dq = torch.where(u <= 0.5, dq1, dq2)
        mutated = x + dq * span
        x.copy_(torch.where(mutate, mutated, x))
        x.copy_(self._respect_bounds(x))
        return result


class CosynePermutation(CopyingOperator):
    """Column-wise permutation of the population (CoSyNE's decorrelation
    operator, reference real.py:606). With `permute_all=False`, each
    value's permutation probability depends on its solution's rank."""

    def __init__(self, problem: Problem, obj_index: Optional[int] = None, *, permute_all: bool = False):
        super().__init__(problem)
        self._obj_index = obj_index
        self._permute_all = bool(permute_all)

    def _do(self, batch: SolutionBatch) -> SolutionBatch:
        n, length = len(batch), batch.solution_length
        device = batch.device
        g = self._problem.generator
        gen = g if (g is not None and g.device == torch.device(device)) else None
        if self._permute_all:
            to_permute = torch.ones(n, length, dtype=torch.bool, device=device)
        else:
            # probability of staying = (rank/n)^(1/4), following the
            # reference's rank-dependent scheme
            utils = batch.utility(self._obj_index, ranking_method="linear")  # [0,1], 1=best
            prob_permute = (1.0 - (utils.to(torch.float32)) ** 0.25).unsqueeze(-1).expand(n, length)
            to_permute = torch.rand(n, length, device=device, generator=gen) < prob_permute
        values = batch.unsafe_values
        new_values = values.clone()
        # per-column permutation among the selected rows: rank selected rows
        # by random keys (masked rows keep their position via +inf keys)
        keys = torch.rand(n, length, device=device, generator=gen)
        keys = torch.where(to_permute, keys, torch.full_like(keys, float("inf")))
        order = keys.argsort(dim=0)  # for each column, permuted rows first
        num_sel = to_permute.sum(dim=0)  # per column
        # positions of selected entries per column, in original row order:
        sel_sorted_rows = torch.where(to_permute, torch.arange(n, device=device).unsqueeze(-1).expand(n, length).to(torch.float32), torch.full((n, length), float("inf"), device=device)).argsort(dim=0)
        # scatter: the i-th selected row (by random order) takes the value of
        # the i-th selected row (by original order) — a random permutation
        col = torch.arange(length, device=device).unsqueeze(0).expand(n, length)
        src_rows = order
        dst_rows = sel_sorted_rows
        mask_i = torch.arange(n, device=device).unsqueeze(-1).expand(n, length) < num_sel.unsqueeze(0)
        flat_dst = dst_rows[mask_i] * length + col[mask_i]
        flat_src = src_rows[mask_i] * length + col[mask_i]
        new_values.view(-1)[flat_dst] = values.reshape(-1)[flat_src]
        result = SolutionBatch(like=batch, popsize=n)
        result.access_values().copy_(new_values)
        return result
