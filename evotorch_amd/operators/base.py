"""Operator base classes: Operator, CopyingOperator, CrossOver.

Reference parity: /root/reference/src/evotorch/operators/base.py:27-438.
"""

from typing import Optional

import torch

from ..core import Problem, SolutionBatch
from ..utils.misc import clip_tensor

__all__ = ["Operator", "CopyingOperator", "SingleObjOperator", "CrossOver"]


class Operator:
    """An in-place population transform: call with a SolutionBatch."""

    def __init__(self, problem: Problem):
        self._problem = problem

    @property
    def problem(self) -> Problem:
        return self._problem

    @property
    def dtype(self):
        return self._problem.dtype

    @property
    def eval_dtype(self):
        return self._problem.eval_dtype

    @property
    def device(self):
        return self._problem.device

    def _respect_bounds(self, x: torch.Tensor) -> torch.Tensor:
        """Clamp decision variables into the problem's bounds
        (reference base.py:75)."""
        lb, ub = self._problem.lower_bounds, self._problem.upper_bounds
        if lb is None and ub is None:
            return x
        return clip_tensor(x, lb=lb, ub=ub, ensure_copy=False)

    def _do(self, batch: SolutionBatch):
        raise NotImplementedError

    def __call__(self, batch: SolutionBatch):
        self._do(batch)


class CopyingOperator(Operator):
    """An out-of-place transform: call returns a new SolutionBatch."""

    def _do(self, batch: SolutionBatch) -> SolutionBatch:
        raise NotImplementedError

    def __call__(self, batch: SolutionBatch) -> SolutionBatch:
        return self._do(batch)


class SingleObjOperator(Operator):
    def __init__(self, problem: Problem):
        problem.ensure_single_objective()
        super().__init__(problem)


class CrossOver(CopyingOperator):
    """Base for crossover operators: pairs parents via two parallel
    tournaments, children fill a batch of the same size as the parents
    (reference base.py:157-366)."""

    def __init__(
        self,
        problem: Problem,
        *,
        tournament_size: int,
        obj_index: Optional[int] = None,
        num_children: Optional[int] = None,
        cross_over_rate: Optional[float] = None,
    ):
        super().__init__(problem)
        self._tournament_size = int(tournament_size)
        self._obj_index = None if obj_index is None else int(obj_index)
        if num_children is not None and cross_over_rate is not None:
            raise ValueError("Provide at most one of num_children, cross_over_rate")
        self._num_children = None if num_children is None else int(num_children)
        self._cross_over_rate = None if cross_over_rate is None else float(cross_over_rate)

    @property
    def obj_index(self) -> Optional[int]:
        return self._obj_index

    def _tournament(self, batch: SolutionBatch) -> tuple:
        """Run two parallel tournaments returning (parents1, parents2)
        value tensors, each num_children/2 rows."""
        popsize = len(batch)
        if self._num_children is not None:
            num_children = self._num_children
        elif self._cross_over_rate is not None:
            num_children = int(popsize * self._cross_over_rate)
        else:
            num_children = popsize
        num_pairings = max(1, num_children // 2)

        if self._problem.is_multi_objective and self._obj_index is None:
            # crowded-comparison operator (Deb 2002): primary key = pareto
            # rank, tie-break = crowding distance within the front
            ranks, crowd = batch.compute_pareto_ranks(crowdsort=True)
            n = len(batch)
            crowd_order = torch.nan_to_num(crowd.to(torch.float64), posinf=1e300).argsort(descending=True)
            crowd_pos = torch.empty_like(crowd_order)
            crowd_pos.scatter_(0, crowd_order, torch.arange(n, device=ranks.device))
            utils = -(ranks.to(torch.float64) * (n + 1) + crowd_pos).to(torch.float32)
        else:
            utils = batch.utility(self._obj_index, ranking_method="centered")

        device = batch.device
        g = self._problem.generator
        n_tournaments = num_pairings * 2
        contenders = torch.randint(0, popsize, (n_tournaments, self._tournament_size), device=device, generator=g if (g is not None and g.device == device) else None)
        scores = utils[contenders]
        winners = contenders.gather(1, scores.argmax(dim=1, keepdim=True)).reshape(-1)
        values = batch.unsafe_values
        parents1 = values[winners[:num_pairings]]
        parents2 = values[winners[num_pairings:]]
        return parents1, parents2

    def _do_cross_over(self, parents1: torch.Tensor, parents2: torch.Tensor) -> SolutionBatch:
        raise NotImplementedError

    def _make_children_batch(self, child_values: torch.Tensor) -> SolutionBatch:
        result = SolutionBatch(self._problem, popsize=child_values.shape[0], empty=True, device=child_values.device)
        result.access_values().copy_(child_values)
        return result

    def _do(self, batch: SolutionBatch) -> SolutionBatch:
        parents1, parents2 = self._tournament(batch)
        children = self._do_cross_over(parents1, parents2)
        children.access_values(keep_evals=True).copy_(self._respect_bounds(children.unsafe_values))
        return children
