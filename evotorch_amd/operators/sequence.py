"""Variable-length (object-dtype) operators.

Reference parity: /root/reference/src/evotorch/operators/sequence.py:25.
"""

import torch

from ..core import SolutionBatch
from ..utils import ObjectArray
from .base import CrossOver

__all__ = ["CutAndSplice"]


class CutAndSplice(CrossOver):
    """Cut-and-splice crossover for variable-length sequence solutions
    (object dtype): each parent is cut at an independent random point and
    the tails are swapped."""

    def _cut_and_splice(self, a, b, rng):
        a = list(a)
        b = list(b)
        cut_a = int(rng.integers(0, len(a) + 1)) if len(a) else 0
        cut_b = int(rng.integers(0, len(b) + 1)) if len(b) else 0
        child1 = a[:cut_a] + b[cut_b:]
        child2 = b[:cut_b] + a[cut_a:]
        return child1, child2

    def _do_cross_over(self, parents1, parents2) -> SolutionBatch:
        import numpy as np

        num_pairs = len(parents1)
        rng = np.random.default_rng(
            int(torch.randint(0, 2**31, (1,), generator=self._problem.generator).item())
            if self._problem.generator is not None
            else None
        )
        children = []
        for i in range(num_pairs):
            c1, c2 = self._cut_and_splice(parents1[i], parents2[i], rng)
            children.append(c1)
            children.append(c2)
        result = SolutionBatch(self._problem, popsize=len(children), empty=True)
        for i, c in enumerate(children):
            result._values[i] = c
        return result

    def _tournament(self, batch: SolutionBatch):
        # object-dtype values cannot be fancy-indexed as tensors; gather rows
        popsize = len(batch)
        num_children = self._num_children if self._num_children is not None else popsize
        if self._cross_over_rate is not None:
            num_children = int(popsize * self._cross_over_rate)
        num_pairings = max(1, num_children // 2)
        utils = batch.utility(self._obj_index, ranking_method="centered")
        device = utils.device
        g = self._problem.generator
        contenders = torch.randint(0, popsize, (num_pairings * 2, self._tournament_size), device=device, generator=g if (g is not None and g.device == device) else None)
        scores = utils[contenders]
        winners = contenders.gather(1, scores.argmax(dim=1, keepdim=True)).reshape(-1)
        values: ObjectArray = batch.unsafe_values
        parents1 = [values[int(i)] for i in winners[:num_pairings]]
        parents2 = [values[int(i)] for i in winners[num_pairings:]]
        return parents1, parents2

    def _do(self, batch: SolutionBatch) -> SolutionBatch:
        parents1, parents2 = self._tournament(batch)
        return self._do_cross_over(parents1, parents2)
