"""Core runtime: Problem, SolutionBatch, Solution, ProblemBoundEvaluator.

MI355X-native re-design of the reference's `core.py`
(/root/reference/src/evotorch/core.py:365-5256). The major architectural
departure: the reference parallelizes evaluation with Ray actors and an
object store (core.py:115-348, 1977-2131); here the parallel substrate is
SPMD — one process per GPU joined by RCCL over xGMI through
`evotorch_amd.parallel.Comm`. A Problem may be attached to a Comm, after
which `evaluate()` shards the population rows across ranks and all-gathers
the fitnesses (P1 in SURVEY.md §2.8), and `sample_and_compute_gradients()`
computes rank-local ES gradients merged by a single all-reduce (P2).

Storage layout matches the reference's contract (verified by the aliasing
tests): a SolutionBatch owns a contiguous 2-D `values` tensor of shape
(popsize, solution_length) and a 2-D `evals` tensor of shape
(popsize, num_objectives + eval_data_length) where NaN marks "not yet
evaluated"; basic slicing returns shared-memory views.
"""

import logging
import math
import os
from typing import Any, Callable, Iterable, List, Optional, Union

import numpy as np
import torch

from .utils import (
    Device,
    DType,
    Hook,
    ObjectArray,
    ReadOnlyTensor,
    Serializable,
    TensorMakerMixin,
    as_read_only_tensor,
    deep_clone,
    is_dtype_object,
    to_torch_dtype,
)
from .utils.misc import ensure_tensor_length_and_dtype, split_workload
from .utils.profiling import record_range

_logger = logging.getLogger(__name__)

__all__ = ["Problem", "SolutionBatch", "SolutionBatchPieces", "Solution", "ProblemBoundEvaluator"]

ObjectiveSense = Union[str, Iterable[str]]
BoundsPair = Any


def _normalize_sense(objective_sense: ObjectiveSense) -> List[str]:
    if isinstance(objective_sense, str):
        senses = [objective_sense]
    else:
        senses = list(objective_sense)
    for s in senses:
        if s not in ("min", "max"):
            raise ValueError(f"Objective sense must be 'min' or 'max', got {s!r}")
    return senses


# ============================================================================
# Pareto machinery (K7 in SURVEY.md §2.9)
# ============================================================================


def _domination_matrix(utils: torch.Tensor) -> torch.Tensor:
    """Boolean (N, N) matrix: entry [i, j] is True iff solution i dominates
    solution j. `utils` is (N, M) with *higher is better* for every column
    (senses already folded in)."""
    a = utils.unsqueeze(1)  # (N, 1, M)
    b = utils.unsqueeze(0)  # (1, N, M)
    ge = (a >= b).all(dim=-1)
    gt = (a > b).any(dim=-1)
    return ge & gt


def _compute_pareto_ranks(utils: torch.Tensor, crowdsort: bool = True, min_assigned: Optional[int] = None):
    """Device-dispatching wrapper: HIP front-peel kernels on ROCm (K7),
    eager masked reductions on CPU. `min_assigned` allows the peel to
    stop early once that many solutions hold final ranks (exact for
    take_best(n): the boundary front is always fully peeled, so the
    first n of the (rank, crowding) order are unchanged)."""
    if utils.device.type == "cuda":
        from . import ops

        ranks = ops.pareto_ranks(utils, min_assigned)
        crowd = _crowding_distances(utils, ranks) if crowdsort else None
        return ranks, crowd
    return _compute_pareto_ranks_eager(utils, crowdsort, min_assigned=min_assigned)


def _compute_pareto_ranks_eager(utils: torch.Tensor, crowdsort: bool = True, min_assigned: Optional[int] = None):
    """Non-dominated ranking by iterative front peeling over domination
    counts — formulated as masked reductions so the only host sync is the
    loop-termination check (SURVEY.md §7 hard-parts note on K7).

    Returns (ranks, crowd) where ranks[i] is the index of i's pareto front
    (0 = best front) and crowd[i] is the crowding distance (or None)."""
    n = utils.shape[0]
    stop_at = n if (min_assigned is None or min_assigned <= 0) else min(int(min_assigned), n)
    dom = _domination_matrix(utils)
    dom_count = dom.sum(dim=0).to(torch.int64)  # how many dominate me
    ranks = torch.full((n,), -1, dtype=torch.int64, device=utils.device)
    assigned = torch.zeros(n, dtype=torch.bool, device=utils.device)
    front_index = 0
    domf = dom.to(torch.int64)
    while not bool(assigned.all()):
        if int(assigned.sum()) >= stop_at:
            ranks[~assigned] = front_index  # beyond-last lump (early stop)
            break
        current = (dom_count == 0) & (~assigned)
        if not bool(current.any()):
            # numerical corner: break ties by assigning the rest to one front
            ranks[~assigned] = front_index
            break
        ranks[current] = front_index
        assigned |= current
        # remove the front's domination contributions
        dom_count = dom_count - domf[current].sum(dim=0)
        dom_count[assigned] = -1
        front_index += 1
    crowd = _crowding_distances(utils, ranks) if crowdsort else None
    return ranks, crowd


def _crowding_distances(utils: torch.Tensor, ranks: torch.Tensor) -> torch.Tensor:
    """NSGA-II crowding distance (reference core.py:3432): within each
    pareto front, per objective, interior solutions get
    (next − prev) / front-span and boundary solutions +inf; objectives
    with zero span within a front contribute nothing.

    Fully vectorized across fronts: one lexicographic (front, value) sort
    per objective + segmented neighbor diffs — a python loop over fronts
    would launch thousands of tiny kernels at large popsizes (measured
    ~300 ms/generation at popsize 8192 before this formulation).
    """
    n, m = utils.shape
    device = utils.device
    ranks_f = ranks.to(torch.float64)
    crowd = torch.zeros(n, dtype=utils.dtype, device=device)
    false1 = torch.zeros(1, dtype=torch.bool, device=device)
    inf = float("inf")
    for j in range(m):
        vals = utils[:, j].to(torch.float64)
        vmin = vals.min()
        span_all = (vals.max() - vmin).clamp_min(1e-300)
        nv = (vals - vmin) / span_all  # in [0, 1]
        order = (ranks_f * 4.0 + nv).argsort()  # sort by (front, value)
        s_rank = ranks[order]
        s_vals = vals[order]
        same_prev = torch.cat([false1, s_rank[1:] == s_rank[:-1]])
        same_next = torch.cat([s_rank[:-1] == s_rank[1:], false1])
        prev_vals = torch.cat([s_vals[:1], s_vals[:-1]])
        next_vals = torch.cat([s_vals[1:], s_vals[-1:]])
        # per-front value span via scatter-reduce keyed by rank
        front_max = torch.full((n,), -inf, dtype=torch.float64, device=device)
        front_max.scatter_reduce_(0, ranks, vals, "amax", include_self=False)
        front_min = torch.full((n,), inf, dtype=torch.float64, device=device)
        front_min.scatter_reduce_(0, ranks, vals, "amin", include_self=False)
        span_f = (front_max - front_min).gather(0, s_rank.clamp(min=0))
        interior = same_prev & same_next
        contrib = torch.where(interior, (next_vals - prev_vals) / span_f.clamp_min(1e-300), torch.full_like(s_vals, inf))
        contrib = torch.where(span_f == 0, torch.zeros_like(contrib), contrib)
        crowd.scatter_add_(0, order, contrib.to(crowd.dtype))
    return crowd


# ============================================================================
# SolutionBatch
# ============================================================================


class SolutionBatch(Serializable):
    """A population: a 2-D values tensor plus a 2-D evals tensor.

    Mirrors the reference's container semantics
    (/root/reference/src/evotorch/core.py:3590-4601): NaN rows of `evals`
    mean "unevaluated"; slicing with a contiguous slice shares memory with
    the parent batch; `split`/`concat`/`take`/`take_best` reshape
    populations for GA-style algorithms.
    """

    def __init__(
        self,
        problem: Optional["Problem"] = None,
        popsize: Optional[int] = None,
        *,
        device: Optional[Device] = None,
        slice_of: Optional[tuple] = None,
        like: Optional["SolutionBatch"] = None,
        merging_of: Optional[Iterable["SolutionBatch"]] = None,
        empty: bool = False,
    ):
        self._num_objs: int
        self._eval_data_length: int
        self._senses: List[str]

        if slice_of is not None:
            source, sl = slice_of
            if isinstance(sl, slice):
                start, stop, step = sl.indices(len(source))
                if step != 1:
                    raise ValueError("SolutionBatch slices must be contiguous (step 1); use take() for fancy indexing")
            else:
                start, stop = sl
            self._values = source._values[start:stop]
            self._evals = source._evals[start:stop]
            self._num_objs = source._num_objs
            self._eval_data_length = source._eval_data_length
            self._senses = source._senses
            return

        if merging_of is not None:
            batches = list(merging_of)
            if len(batches) == 0:
                raise ValueError("Cannot merge zero batches")
            first = batches[0]
            self._num_objs = first._num_objs
            self._eval_data_length = first._eval_data_length
            self._senses = first._senses
            if isinstance(first._values, ObjectArray):
                total = sum(len(b) for b in batches)
                self._values = ObjectArray(total)
                i = 0
                for b in batches:
                    for j in range(len(b)):
                        self._values[i] = b._values[j]
                        i += 1
            else:
                self._values = torch.cat([b._values for b in batches], dim=0)
            self._evals = torch.cat([b._evals for b in batches], dim=0)
            return

        if like is not None:
            n = popsize if popsize is not None else len(like)
            self._num_objs = like._num_objs
            self._eval_data_length = like._eval_data_length
            self._senses = like._senses
            if isinstance(like._values, ObjectArray):
                self._values = ObjectArray(n)
            else:
                self._values = torch.empty((n, like._values.shape[1]), dtype=like._values.dtype, device=device if device is not None else like._values.device)
            self._evals = torch.full(
                (n, self._num_objs + self._eval_data_length), float("nan"), dtype=like._evals.dtype, device=device if device is not None else like._evals.device
            )
            return

        if problem is None:
            raise ValueError("SolutionBatch requires one of: problem, slice_of, like, merging_of")
        if popsize is None:
            popsize = 1
        popsize = int(popsize)
        self._num_objs = len(problem.senses)
        self._eval_data_length = problem.eval_data_length
        self._senses = list(problem.senses)
        dev = device if device is not None else problem.device
        if problem.dtype_is_object:
            self._values = ObjectArray(popsize)
        else:
            self._values = torch.empty((popsize, problem.solution_length), dtype=problem.dtype, device=dev)
        self._evals = torch.full((popsize, self._num_objs + self._eval_data_length), float("nan"), dtype=problem.eval_dtype, device=dev)
        if not empty:
            problem._fill(self._values)

    # -- basic properties ----------------------------------------------------

    def __len__(self) -> int:
        return len(self._values) if isinstance(self._values, ObjectArray) else self._values.shape[0]

    @property
    def values_shape(self) -> tuple:
        return tuple(self._values.shape)

    @property
    def eval_shape(self) -> tuple:
        return tuple(self._evals.shape)

    @property
    def solution_length(self) -> Optional[int]:
        if isinstance(self._values, ObjectArray):
            return None
        return self._values.shape[1]

    @property
    def objective_sense(self) -> Union[str, List[str]]:
        return self._senses[0] if len(self._senses) == 1 else list(self._senses)

    @property
    def senses(self) -> List[str]:
        return list(self._senses)

    @property
    def num_objectives(self) -> int:
        return self._num_objs

    @property
    def eval_data_length(self) -> int:
        return self._eval_data_length

    @property
    def device(self):
        return self._evals.device

    @property
    def dtype(self):
        return object if isinstance(self._values, ObjectArray) else self._values.dtype

    @property
    def values_dtype(self):
        return self.dtype

    @property
    def eval_dtype(self):
        return self._evals.dtype

    @property
    def is_multi_objective(self) -> bool:
        return self._num_objs > 1

    # -- value/eval access ---------------------------------------------------

    def access_values(self, *, keep_evals: bool = False):
        """Mutable access to the values tensor; evals are reset to NaN
        unless `keep_evals=True` (reference core.py:3733)."""
        if not keep_evals:
            self.forget_evals()
        return self._values

    def access_evals(self, obj_index: Optional[int] = None) -> torch.Tensor:
        if obj_index is None:
            return self._evals
        return self._evals[:, self._normalize_obj_index(obj_index)]

    @property
    def values(self):
        if isinstance(self._values, ObjectArray):
            return self._values.get_read_only_view()
        return as_read_only_tensor(self._values)

    @property
    def evals(self) -> ReadOnlyTensor:
        return as_read_only_tensor(self._evals)

    @property
    def unsafe_values(self):
        """The raw (mutable) values tensor, WITHOUT forgetting the evals."""
        return self._values

    @property
    def unsafe_evals(self) -> torch.Tensor:
        return self._evals

    def forget_evals(self, *, solutions: Optional[Union[slice, torch.Tensor]] = None):
        if solutions is None:
            self._evals.fill_(float("nan"))
        else:
            self._evals[solutions] = float("nan")

    def set_values(self, values, *, solutions: Optional[Union[slice, torch.Tensor]] = None):
        if solutions is None:
            solutions = slice(None)
        if isinstance(self._values, ObjectArray):
            self._values[solutions] = values
        else:
            self._values[solutions] = torch.as_tensor(values, dtype=self._values.dtype, device=self._values.device)
        self.forget_evals(solutions=solutions)

    def set_evals(self, evals: torch.Tensor, eval_data: Optional[torch.Tensor] = None, *, solutions: Optional[Union[slice, torch.Tensor]] = None):
        if solutions is None:
            solutions = slice(None)
        evals = torch.as_tensor(evals, dtype=self._evals.dtype, device=self._evals.device)
        if evals.ndim == 1:
            if self._num_objs != 1:
                evals = evals.reshape(-1, self._num_objs)
            else:
                evals = evals.reshape(-1, 1)
        if evals.shape[1] == self._num_objs + self._eval_data_length and eval_data is None:
            self._evals[solutions] = evals
            return
        self._evals[solutions, : self._num_objs] = evals
        if eval_data is not None:
            eval_data = torch.as_tensor(eval_data, dtype=self._evals.dtype, device=self._evals.device)
            self._evals[solutions, self._num_objs :] = eval_data

    @property
    def evals_are_ready(self) -> bool:
        return not bool(torch.isnan(self._evals[:, : self._num_objs]).any())

    # -- indexing ------------------------------------------------------------

    def __getitem__(self, i):
        if isinstance(i, slice):
            return SolutionBatch(slice_of=(self, i))
        if isinstance(i, (torch.Tensor, np.ndarray, list)):
            return self.take(i)
        return Solution(parent=self, index=int(i))

    def __iter__(self):
        for i in range(len(self)):
            yield self[i]

    def take(self, indices) -> "SolutionBatch":
        """New batch whose rows are copies of the given rows."""
        if isinstance(indices, torch.Tensor):
            idx = indices.to(dtype=torch.int64)
        else:
            idx = torch.as_tensor(np.asarray(indices), dtype=torch.int64)
        result = SolutionBatch(like=self, popsize=len(idx))
        if isinstance(self._values, ObjectArray):
            for out_i, src_i in enumerate(idx.tolist()):
                result._values[out_i] = self._values[src_i]
        else:
            idx_dev = idx.to(self._values.device)
            result._values.copy_(self._values[idx_dev])
        result._evals.copy_(self._evals[idx.to(self._evals.device)])
        return result

    def take_best(self, n: Optional[int] = None, *, obj_index: Optional[int] = None) -> Union["SolutionBatch", "Solution"]:
        """Best n solutions as a new batch (pareto-based for multi-objective
        when obj_index is None); with n omitted, the single best Solution."""
        if n is None:
            indices = self.argsort(obj_index=obj_index, min_assigned=1)
            return self[int(indices[0])]
        indices = self.argsort(obj_index=obj_index, min_assigned=int(n))[:n]
        return self.take(indices)

    def split(self, num_pieces: Optional[int] = None, *, max_size: Optional[int] = None) -> "SolutionBatchPieces":
        return SolutionBatchPieces(self, num_pieces=num_pieces, max_size=max_size)

    @staticmethod
    def cat(batches: Iterable["SolutionBatch"]) -> "SolutionBatch":
        return SolutionBatch(merging_of=batches)

    def concat(self, other: Union["SolutionBatch", Iterable["SolutionBatch"]]) -> "SolutionBatch":
        others = [other] if isinstance(other, SolutionBatch) else list(other)
        return SolutionBatch(merging_of=[self] + others)

    def to(self, device: Device) -> "SolutionBatch":
        device = torch.device(device)
        if (not isinstance(self._values, ObjectArray)) and self._values.device == device and self._evals.device == device:
            return self
        result = SolutionBatch(like=self, popsize=len(self), device=device)
        if isinstance(self._values, ObjectArray):
            result._values = self._values.clone()
        else:
            result._values.copy_(self._values.to(device))
        result._evals.copy_(self._evals.to(device))
        return result

    # -- sorting & utilities ---------------------------------------------------

    def _normalize_obj_index(self, obj_index: Optional[int]) -> int:
        if obj_index is None:
            if self._num_objs != 1:
                raise ValueError("obj_index must be given for a multi-objective batch")
            return 0
        obj_index = int(obj_index)
        if obj_index < 0:
            obj_index += self._num_objs
        if not (0 <= obj_index < self._num_objs):
            raise IndexError(f"Invalid obj_index {obj_index}")
        return obj_index

    def _utils_for_sorting(self) -> torch.Tensor:
        """Evals folded so higher is better for every objective."""
        utils = self._evals[:, : self._num_objs].clone()
        for j, sense in enumerate(self._senses):
            if sense == "min":
                utils[:, j] = -utils[:, j]
        return utils

    def argsort(self, obj_index: Optional[int] = None, *, min_assigned: Optional[int] = None) -> torch.Tensor:
        """Indices from best to worst. For a multi-objective batch without
        obj_index, sorts by (pareto rank, -crowding distance);
        `min_assigned` lets the non-dominated sort stop after the fronts
        covering that many solutions (exact for head-of-order uses)."""
        if self._num_objs > 1 and obj_index is None:
            ranks, crowd = self.compute_pareto_ranks(crowdsort=True, min_assigned=min_assigned)
            # lexicographic (pareto rank asc, crowding distance desc) via
            # integer keys: position-in-crowd-order breaks ties within a front
            n = ranks.shape[0]
            crowd_order = torch.nan_to_num(crowd.to(torch.float64), posinf=1e300).argsort(descending=True)
            crowd_pos = torch.empty_like(crowd_order)
            crowd_pos.scatter_(0, crowd_order, torch.arange(n, device=ranks.device))
            key = ranks * (n + 1) + crowd_pos
            return key.argsort()
        j = self._normalize_obj_index(obj_index)
        evals = self._evals[:, j]
        descending = self._senses[j] == "max"
        return evals.argsort(descending=descending)

    def argbest(self, obj_index: Optional[int] = None) -> torch.Tensor:
        j = self._normalize_obj_index(obj_index)
        evals = self._evals[:, j]
        return evals.argmax() if self._senses[j] == "max" else evals.argmin()

    def argworst(self, obj_index: Optional[int] = None) -> torch.Tensor:
        j = self._normalize_obj_index(obj_index)
        evals = self._evals[:, j]
        return evals.argmin() if self._senses[j] == "max" else evals.argmax()

    def compute_pareto_ranks(self, crowdsort: bool = True, min_assigned: Optional[int] = None):
        utils = self._utils_for_sorting()
        return _compute_pareto_ranks(utils, crowdsort=crowdsort, min_assigned=min_assigned)

    def arg_pareto_sort(self, crowdsort: bool = True):
        """List of fronts (each a tensor of indices), best front first
        (reference core.py:3554)."""
        ranks, crowd = self.compute_pareto_ranks(crowdsort=crowdsort)
        fronts = []
        for front in torch.unique(ranks, sorted=True):
            idx = torch.nonzero(ranks == front, as_tuple=True)[0]
            if crowdsort and crowd is not None and len(idx) > 1:
                idx = idx[crowd[idx].argsort(descending=True)]
            fronts.append(idx)
        return fronts

    def utility(self, obj_index: Optional[int] = None, *, ranking_method: Optional[str] = None) -> torch.Tensor:
        """Fitness-shaped utilities for one objective (reference
        core.py:4207); higher utility = better regardless of sense."""
        from .utils import ranking

        j = self._normalize_obj_index(obj_index)
        return ranking.rank(self._evals[:, j], ranking_method or "raw", higher_is_better=(self._senses[j] == "max"))

    def utils(self, *, ranking_method: Optional[str] = None) -> torch.Tensor:
        """Utilities for all objectives, shape (popsize, num_objectives)."""
        cols = [self.utility(j, ranking_method=ranking_method) for j in range(self._num_objs)]
        return torch.stack(cols, dim=-1)

    # -- cloning / serialization ----------------------------------------------

    def _get_cloned_state(self, *, memo: dict) -> dict:
        return {
            "_values": self._values.clone() if not isinstance(self._values, ObjectArray) else self._values.clone(),
            "_evals": self._evals.clone(),
            "_num_objs": self._num_objs,
            "_eval_data_length": self._eval_data_length,
            "_senses": list(self._senses),
        }

    def __repr__(self) -> str:
        return f"<SolutionBatch popsize={len(self)} length={self.solution_length} device={self.device}>"


class SolutionBatchPieces:
    """Lazy view of a batch split into contiguous sub-batches (reference
    core.py:4603-4729)."""

    def __init__(self, batch: SolutionBatch, *, num_pieces: Optional[int] = None, max_size: Optional[int] = None):
        self._batch = batch
        n = len(batch)
        if (num_pieces is None) == (max_size is None):
            raise ValueError("Provide exactly one of num_pieces, max_size")
        if max_size is not None:
            num_pieces = max(1, math.ceil(n / int(max_size)))
        sizes = split_workload(n, int(num_pieces))
        self._ranges = []
        start = 0
        for s in sizes:
            self._ranges.append((start, start + s))
            start += s

    def __len__(self) -> int:
        return len(self._ranges)

    def __getitem__(self, i: int) -> SolutionBatch:
        start, stop = self._ranges[i]
        return SolutionBatch(slice_of=(self._batch, (start, stop)))

    def indices_of(self, i: int) -> tuple:
        return self._ranges[i]

    def __iter__(self):
        for i in range(len(self)):
            yield self[i]


# ============================================================================
# Solution
# ============================================================================


class Solution(Serializable):
    """A single row view into a parent SolutionBatch (reference
    core.py:4742-5107)."""

    def __init__(self, parent: SolutionBatch, index: int):
        if index < 0:
            index += len(parent)
        if not (0 <= index < len(parent)):
            raise IndexError(f"Solution index {index} out of range")
        self._batch = parent
        self._index = index

    @property
    def values(self):
        v = self._batch._values
        if isinstance(v, ObjectArray):
            return v[self._index]
        return as_read_only_tensor(v[self._index])

    @property
    def evals(self) -> ReadOnlyTensor:
        return as_read_only_tensor(self._batch._evals[self._index])

    @property
    def evaluation(self) -> torch.Tensor:
        return as_read_only_tensor(self._batch._evals[self._index, 0])

    def access_values(self, *, keep_evals: bool = False):
        if not keep_evals:
            self._batch.forget_evals(solutions=self._index)
        v = self._batch._values
        if isinstance(v, ObjectArray):
            return v[self._index]
        return v[self._index]

    def set_values(self, values):
        self._batch.set_values(values, solutions=self._index)

    def access_evals(self) -> torch.Tensor:
        """Mutable view of this solution's evals row (reference
        core.py:4858)."""
        return self._batch.access_evals()[self._index]

    @property
    def shape(self) -> torch.Size:
        return torch.Size([self._batch.solution_length]) if self._batch.solution_length is not None else torch.Size([1])

    def size(self) -> torch.Size:
        return self.shape

    @property
    def ndim(self) -> int:
        return 1

    def dim(self) -> int:
        return self.ndim

    @property
    def eval_shape(self) -> torch.Size:
        return torch.Size([self._batch._evals.shape[1]])

    @property
    def eval_dtype(self):
        return self._batch.eval_dtype

    def to(self, device) -> "Solution":
        """Copy of this solution (as a 1-element batch view) on `device`
        (reference core.py:5010)."""
        return self.to_batch().to(device)[0]

    def set_evals(self, evals, eval_data=None):
        evals = torch.as_tensor(evals, dtype=self._batch._evals.dtype, device=self._batch._evals.device).reshape(-1)
        no = self._batch._num_objs
        if len(evals) == no + self._batch._eval_data_length:
            self._batch._evals[self._index] = evals
            return
        self._batch._evals[self._index, :no] = evals
        if eval_data is not None:
            self._batch._evals[self._index, no:] = torch.as_tensor(eval_data, dtype=self._batch._evals.dtype, device=self._batch._evals.device)

    def set_evaluation(self, evaluation: float, eval_data=None):
        self._batch._evals[self._index, 0] = float(evaluation)
        if eval_data is not None:
            self._batch._evals[self._index, self._batch._num_objs :] = torch.as_tensor(eval_data, dtype=self._batch._evals.dtype, device=self._batch._evals.device)

    @property
    def is_evaluated(self) -> bool:
        return not bool(torch.isnan(self._batch._evals[self._index, : self._batch._num_objs]).any())

    @property
    def objective_sense(self):
        return self._batch.objective_sense

    @property
    def senses(self) -> List[str]:
        return self._batch.senses

    @property
    def dtype(self):
        return self._batch.dtype

    @property
    def device(self):
        return self._batch.device

    def to_batch(self) -> SolutionBatch:
        """1-row shared-memory view batch."""
        return SolutionBatch(slice_of=(self._batch, (self._index, self._index + 1)))

    def clone(self, *, memo: Optional[dict] = None) -> "Solution":
        batch = self.to_batch()
        new_batch = SolutionBatch(like=batch, popsize=1)
        if isinstance(batch._values, ObjectArray):
            new_batch._values[0] = batch._values[0]
        else:
            new_batch._values.copy_(batch._values)
        new_batch._evals.copy_(batch._evals)
        return Solution(parent=new_batch, index=0)

    def _get_cloned_state(self, *, memo: dict) -> dict:
        c = self.clone()
        return {"_batch": c._batch, "_index": 0}

    def __len__(self) -> int:
        v = self._batch._values
        return 0 if isinstance(v, ObjectArray) else v.shape[1]

    def __getitem__(self, i):
        return self.values[i]

    def __repr__(self) -> str:
        ev = "evaluated" if self.is_evaluated else "not evaluated"
        return f"<Solution index={self._index} ({ev})>"


# ============================================================================
# Problem
# ============================================================================


class Problem(TensorMakerMixin, Serializable):
    """The central object: objective sense(s), solution geometry, dtypes,
    device, RNG, bounds, evaluation dispatch, and (through an attached
    `evotorch_amd.parallel.Comm`) population-parallel evaluation and
    distributed ES gradients.

    Reference parity: /root/reference/src/evotorch/core.py:365-3415. The Ray
    actor-pool constructor knobs (`num_actors`, `num_gpus_per_actor`, ...)
    are intentionally absent; SPMD rank topology comes from the launcher
    (torchrun) and `evotorch_amd.parallel`.
    """

    def __init__(
        self,
        objective_sense: ObjectiveSense,
        objective_func: Optional[Callable] = None,
        *,
        initial_bounds: Optional[BoundsPair] = None,
        bounds: Optional[BoundsPair] = None,
        solution_length: Optional[int] = None,
        dtype: Optional[DType] = None,
        eval_dtype: Optional[DType] = None,
        device: Optional[Device] = None,
        eval_data_length: int = 0,
        seed: Optional[int] = None,
        store_solution_stats: Optional[bool] = None,
        vectorized: Optional[bool] = None,
        num_actors=None,
        actor_config=None,
        num_gpus_per_actor=None,
        num_subbatches=None,
        subbatch_size=None,
    ):
        # num_actors / num_subbatches / subbatch_size are honored: they
        # drive a multiprocessing evaluation pool (parallel/evalpool.py —
        # the reference's Ray actor system on the standard library) for
        # CPU-side per-solution fitness. GPU/actor-placement knobs are
        # meaningless here (GPU scaling is torchrun + RCCL,
        # docs/parallelism.md) and are ignored with a warning.
        _gpu_knobs = {"actor_config": actor_config, "num_gpus_per_actor": num_gpus_per_actor}
        _given = [k for k, v in _gpu_knobs.items() if v is not None]
        if _given:
            import warnings

            warnings.warn(
                f"Ignoring Ray-era argument(s) {_given}: GPU parallelism here comes from "
                "torchrun + RCCL (problem.use_comm(init_comm())); see docs/migrating_from_evotorch.md",
                stacklevel=2,
            )
        if num_actors in ("max", "num_cpus"):
            num_actors = os.cpu_count() or 1
        self._num_actors = None if num_actors is None else max(0, int(num_actors))
        self._num_subbatches = None if num_subbatches is None else int(num_subbatches)
        self._subbatch_size = None if subbatch_size is None else int(subbatch_size)
        self._eval_pool = None
        self._senses = _normalize_sense(objective_sense)
        self._objective_func = objective_func
        self._vectorized = bool(getattr(objective_func, "__evotorch_vectorized__", False)) if vectorized is None else bool(vectorized)
        fn_device = getattr(objective_func, "__evotorch_device__", None)
        self._fitness_device = torch.device(fn_device) if fn_device is not None else None

        self._dtype = to_torch_dtype(dtype) if (dtype is not None and not is_dtype_object(dtype)) else (object if dtype is not None else torch.float32)
        if eval_dtype is not None:
            self._eval_dtype = to_torch_dtype(eval_dtype)
        else:
            self._eval_dtype = self._dtype if (self._dtype != object and to_torch_dtype(self._dtype).is_floating_point) else torch.float32
        self._device = torch.device(device) if device is not None else torch.device("cpu")
        if self._fitness_device is None and getattr(objective_func, "__evotorch_on_aux_device__", False):
            # @on_aux_device: run fitness on the first accelerator (resolved
            # now that the main device is known) — reference decorators.py:440
            self._fitness_device = self.aux_device
        if self.dtype_is_object and self._device.type != "cpu":
            raise ValueError("object-dtype problems must live on cpu")

        self._solution_length = None if solution_length is None else int(solution_length)
        if self._solution_length is None and not self.dtype_is_object:
            if initial_bounds is None and bounds is None:
                raise ValueError("Provide solution_length (and bounds or initial_bounds) for numeric problems")
        self._eval_data_length = int(eval_data_length)

        # Bounds
        self._initial_lower_bounds = self._initial_upper_bounds = None
        self._lower_bounds = self._upper_bounds = None
        if bounds is not None:
            self._lower_bounds, self._upper_bounds = self._normalize_bounds(bounds)
        if initial_bounds is not None:
            self._initial_lower_bounds, self._initial_upper_bounds = self._normalize_bounds(initial_bounds)
        elif bounds is not None:
            self._initial_lower_bounds, self._initial_upper_bounds = self._lower_bounds, self._upper_bounds

        # RNG
        self._seed = seed
        self._generator: Optional[torch.Generator] = None
        if seed is not None:
            self._generator = torch.Generator(device=self._device)
            self._generator.manual_seed(int(seed))

        # Solution stat tracking
        if store_solution_stats is None:
            store_solution_stats = self._device.type == "cpu"
        self._store_solution_stats = bool(store_solution_stats)
        self._best: Optional[List[Optional[Solution]]] = None
        self._worst: Optional[List[Optional[Solution]]] = None
        self._best_evals: Optional[torch.Tensor] = None
        self._worst_evals: Optional[torch.Tensor] = None

        # Hooks (reference core.py:2175-2238)
        self._before_eval_hook = Hook()
        self._after_eval_hook = Hook()
        self._before_grad_hook = Hook()
        self._after_grad_hook = Hook()
        self._remote_hook = Hook()

        # Counters
        self._after_eval_status: dict = {}
        self._after_grad_status: dict = {}

        # SPMD comm (attached lazily; replaces the reference's actor pool)
        self._comm = None
        self._spmd_seed_base = 0
        self._spmd_gen_counter = 0
        self._defer_fused_reduce = False
        self._fused_reduce_requests: list = []

    # -- configuration properties ---------------------------------------------

    def _normalize_bounds(self, bounds: BoundsPair):
        if self.dtype_is_object:
            raise ValueError("Bounds are not supported for object-dtype problems")
        try:
            lb, ub = bounds
        except Exception:
            raise ValueError(f"Bounds must be a pair (lb, ub); got {bounds!r}") from None
        length = self._solution_length
        lb = ensure_tensor_length_and_dtype(lb, length, self._dtype, about="lower bound", device=self._device)
        ub = ensure_tensor_length_and_dtype(ub, length, self._dtype, about="upper bound", device=self._device)
        if bool((lb > ub).any()):
            raise ValueError(f"Invalid bounds: lower bound exceeds upper bound ({lb} > {ub})")
        return lb, ub

    @property
    def senses(self) -> List[str]:
        return list(self._senses)

    @property
    def objective_sense(self) -> Union[str, List[str]]:
        return self._senses[0] if len(self._senses) == 1 else list(self._senses)

    @property
    def is_multi_objective(self) -> bool:
        return len(self._senses) > 1

    @property
    def is_single_objective(self) -> bool:
        return len(self._senses) == 1

    def get_obj_order_descending(self) -> List[bool]:
        """Per objective: True iff higher is better (reference
        core.py:1763)."""
        return [s == "max" for s in self._senses]

    def normalize_obj_index(self, obj_index: Optional[int] = None) -> int:
        """Resolve an objective index: None is allowed only for
        single-objective problems; negative indices wrap (reference
        core.py:2370)."""
        if obj_index is None:
            if len(self._senses) > 1:
                raise ValueError("obj_index must be given for a multi-objective problem")
            return 0
        obj_index = int(obj_index)
        if obj_index < 0:
            obj_index += len(self._senses)
        if not (0 <= obj_index < len(self._senses)):
            raise IndexError(f"obj_index out of range for {len(self._senses)} objectives")
        return obj_index

    def compare_solutions(self, a: "Solution", b: "Solution", obj_index: Optional[int] = None) -> float:
        """Positive if `a` is better, negative if `b` is better, 0 on a tie
        (both must be evaluated; reference core.py:2402)."""
        j = self.normalize_obj_index(obj_index)
        ea = float(a.evals[j])
        eb = float(b.evals[j])
        diff = ea - eb
        return diff if self._senses[j] == "max" else -diff

    def is_better(self, a: "Solution", b: "Solution", obj_index: Optional[int] = None) -> bool:
        """True iff `a` strictly beats `b` (reference core.py:2432)."""
        return self.compare_solutions(a, b, obj_index) > 0

    def is_worse(self, a: "Solution", b: "Solution", obj_index: Optional[int] = None) -> bool:
        """True iff `a` is strictly beaten by `b` (reference core.py:2448)."""
        return self.compare_solutions(a, b, obj_index) < 0

    @property
    def is_on_cpu(self) -> bool:
        """True if the problem's main device is the CPU (reference
        core.py:1727)."""
        return self._device.type == "cpu"

    def ensure_tensor_length_and_dtype(self, t, *, about: Optional[str] = None, allow_scalar: bool = False):
        """Coerce `t` into a solution-length vector of this problem's dtype
        and device (reference exposes this as a Problem method)."""
        return ensure_tensor_length_and_dtype(
            t, self._solution_length, self._dtype, about=about, allow_scalar=allow_scalar, device=self._device
        )

    @property
    def solution_length(self) -> Optional[int]:
        return self._solution_length

    @property
    def eval_data_length(self) -> int:
        return self._eval_data_length

    @property
    def dtype(self):
        return self._dtype

    @property
    def dtype_is_object(self) -> bool:
        return self._dtype == object

    @property
    def eval_dtype(self) -> torch.dtype:
        return self._eval_dtype

    @property
    def device(self) -> torch.device:
        return self._device

    @property
    def aux_device(self) -> torch.device:
        """The device for heavy fitness computation: the first visible
        accelerator if there is one, else the problem's own device
        (reference core.py:1656-1692)."""
        if torch.cuda.is_available():
            return torch.device("cuda", torch.cuda.current_device())
        return self._device

    @property
    def generator(self) -> Optional[torch.Generator]:
        return self._generator

    @property
    def has_own_generator(self) -> bool:
        """True iff this Problem was given a seed (and so owns a private
        RNG stream instead of using torch's global one)."""
        return self._generator is not None

    def manual_seed(self, seed: Optional[int] = None):
        if seed is None:
            self._generator = None
        else:
            self._generator = torch.Generator(device=self._device)
            self._generator.manual_seed(int(seed))
        return self

    @property
    def initial_lower_bounds(self):
        return self._initial_lower_bounds

    @property
    def initial_upper_bounds(self):
        return self._initial_upper_bounds

    @property
    def lower_bounds(self):
        return self._lower_bounds

    @property
    def upper_bounds(self):
        return self._upper_bounds

    @property
    def before_eval_hook(self) -> Hook:
        return self._before_eval_hook

    @property
    def after_eval_hook(self) -> Hook:
        return self._after_eval_hook

    @property
    def before_grad_hook(self) -> Hook:
        return self._before_grad_hook

    @property
    def after_grad_hook(self) -> Hook:
        return self._after_grad_hook

    @property
    def remote_hook(self) -> Hook:
        """Runs once on each evaluation worker's problem clone right after
        it is constructed (reference core.py:142, :2229 — the actor-side
        initialization hook). Register callables taking the problem; they
        execute inside the `num_actors` pool workers, never on the main
        process."""
        return self._remote_hook

    @property
    def status(self) -> dict:
        return {**self._after_eval_status, **self._after_grad_status}

    # -- SPMD comm -------------------------------------------------------------

    def use_comm(self, comm) -> "Problem":
        """Attach an `evotorch_amd.parallel.Comm`; subsequent `evaluate`
        calls shard the population rows across its ranks and all-gather the
        evals, and `sample_and_compute_gradients` runs in SPMD mode.

        A shared seed chain is established here (one int64 broadcast from
        rank 0): SPMD population sampling is counter-addressed from this
        chain, so the virtual population — and the whole search trajectory
        for a deterministic fitness function — is IDENTICAL for any world
        size that divides the popsize (SURVEY.md §7 "RNG discipline"). Rank
        seeds need not match; rank 0's generator decides."""
        from .ops.dispatch import _seed_from_generator

        self._comm = comm
        base = _seed_from_generator(self._generator, self._device)
        t = torch.tensor([base], dtype=torch.int64)
        comm.broadcast_(t, src=0)
        self._spmd_seed_base = int(t.item()) & 0x7FFFFFFFFFFFFFFF
        self._spmd_gen_counter = 0
        return self

    def _next_spmd_seed(self) -> int:
        """Next seed of the shared chain (all ranks advance in lockstep)."""
        from .ops.dispatch import _splitmix64

        self._spmd_gen_counter += 1
        return _splitmix64(self._spmd_seed_base ^ (self._spmd_gen_counter * 0x9E3779B97F4A7C15))

    def _peek_spmd_seed(self, ahead: int = 1) -> int:
        """Look ahead in the shared seed chain without advancing it (the
        side-stream noise pre-generation needs next generation's seed)."""
        from .ops.dispatch import _splitmix64

        return _splitmix64(self._spmd_seed_base ^ ((self._spmd_gen_counter + ahead) * 0x9E3779B97F4A7C15))

    # -- side-stream noise pre-generation (SURVEY.md §2.8 P2 overlap) --------
    #
    # The philox noise of generation g+1 does not depend on the updated
    # distribution parameters, so it is generated on a side HIP stream
    # CONCURRENTLY with generation g's evaluation and gradient all-reduce;
    # after the parameter update only a cheap bandwidth-bound affine pass
    # (x = mu ± sigma·z) sits on the critical path. Bitwise-identical to
    # direct counter-addressed sampling (both run fmaf(sigma, z, mu)).

    def _sample_with_pregen(self, distribution, values: torch.Tensor, seed: int, row_offset: int, dirs: int):
        from . import ops

        pg = getattr(self, "_pregen", None)
        if (
            pg is not None
            and pg["seed"] == seed
            and pg["row_offset"] == row_offset
            and pg["z"].shape == (dirs, self._solution_length)
        ):
            torch.cuda.current_stream().wait_event(pg["ready"])
            distribution.fill_from_noise(values, pg["z"])
        else:
            distribution.fill_counter_addressed(values, seed=seed, row_offset=row_offset)

    def _schedule_pregen(self, dirs: int, row_offset: int, seed: int):
        from . import ops

        length = self._solution_length
        if getattr(self, "_pregen_stream", None) is None:
            self._pregen_stream = torch.cuda.Stream(device=self._device)
        stream = self._pregen_stream
        pg = getattr(self, "_pregen", None)
        if pg is None or pg["z"].shape != (dirs, length):
            pg = {
                "z": torch.empty((dirs, length), dtype=torch.float32, device=self._device),
                "zero": torch.zeros(length, dtype=torch.float32, device=self._device),
                "one": torch.ones(length, dtype=torch.float32, device=self._device),
                "ready": torch.cuda.Event(),
            }
        # the side stream must not overwrite z while the main stream's
        # affine pass is still reading it
        consumed = torch.cuda.Event()
        consumed.record()
        with torch.cuda.stream(stream):
            stream.wait_event(consumed)
            ops.sample_gaussian(pg["z"], pg["zero"], pg["one"], symmetric=False, seed=seed, row_offset=row_offset)
            pg["ready"].record()
        pg["seed"] = seed
        pg["row_offset"] = row_offset
        self._pregen = pg

    def request_fused_reduce(self, tensor: torch.Tensor, callback) -> None:
        """All-reduce `tensor` across ranks and hand the result to
        `callback`. During a sharded gradient generation the reduction is
        DEFERRED and fused into the gradient all-reduce (one collective
        carries gradients + obs-norm stats — the per-generation collective
        count stays at 2); outside that window it runs immediately."""
        comm = self._comm
        if comm is None or comm.world_size <= 1:
            callback(tensor)
            return
        if getattr(self, "_defer_fused_reduce", False):
            self._fused_reduce_requests.append((tensor, callback))
        else:
            comm.all_reduce_(tensor)
            callback(tensor)

    @property
    def comm(self):
        return self._comm

    @property
    def is_main(self) -> bool:
        return self._comm is None or self._comm.rank == 0

    # -- validation helpers ------------------------------------------------------

    def ensure_numeric(self):
        if self.dtype_is_object:
            raise ValueError("This operation requires a numeric (non-object) dtype problem")

    def ensure_single_objective(self):
        if self.is_multi_objective:
            raise ValueError("This operation requires a single-objective problem")

    def ensure_unbounded(self):
        if self._lower_bounds is not None or self._upper_bounds is not None:
            raise ValueError("This operation requires an unbounded problem")

    # -- population generation ---------------------------------------------------

    def generate_values(self, num_solutions: int):
        """Fresh decision-variable rows, shape (num_solutions, L). Uses
        `_fill` (uniform within initial bounds by default; subclasses
        override)."""
        if self.dtype_is_object:
            result = ObjectArray(num_solutions)
        else:
            result = torch.empty((int(num_solutions), self._solution_length), dtype=self._dtype, device=self._device)
        self._fill(result)
        return result

    def _fill(self, values):
        """Fill a pre-allocated values container with fresh solutions
        (reference core.py:1874). Default: uniform in initial bounds."""
        if self.dtype_is_object:
            raise NotImplementedError("object-dtype problems must override _fill or generate_values")
        if self._initial_lower_bounds is None:
            raise RuntimeError("Problem has no initial_bounds; cannot generate random solutions. Override _fill().")
        from .utils.misc import make_uniform

        make_uniform(lb=self._initial_lower_bounds, ub=self._initial_upper_bounds, generator=self._generator, out=values)

    def generate_batch(
        self,
        popsize: Optional[int] = None,
        *,
        empty: bool = False,
        device: Optional[Device] = None,
    ) -> SolutionBatch:
        return SolutionBatch(self, popsize, device=device, empty=empty)

    def populate(self, popsize: int) -> SolutionBatch:
        return self.generate_batch(popsize)

    # -- evaluation ---------------------------------------------------------------

    def _evaluate_batch(self, batch: SolutionBatch):
        """Evaluate every solution of `batch`, writing into its evals.
        Default: vectorized objective_func if declared, else per-solution
        `_evaluate` (reference core.py:2602-2621)."""
        if self._vectorized and self._objective_func is not None:
            self._evaluate_vectorized(batch, self._objective_func)
            return
        if self._objective_func is not None:
            for sln in batch:
                result = self._objective_func(sln.values)
                self._write_solution_result(sln, result)
            return
        for sln in batch:
            self._evaluate(sln)

    def _evaluate_vectorized(self, batch: SolutionBatch, fn: Callable):
        values = batch.access_values(keep_evals=True)
        target_device = self._fitness_device
        moved = values.to(target_device) if (target_device is not None and not isinstance(values, ObjectArray)) else values
        result = fn(moved)
        if isinstance(result, tuple):
            evals, eval_data = result
            batch.set_evals(torch.as_tensor(evals).to(batch.device), torch.as_tensor(eval_data).to(batch.device))
        else:
            batch.set_evals(torch.as_tensor(result).to(batch.device))

    def _write_solution_result(self, sln: Solution, result):
        if isinstance(result, tuple):
            evals, eval_data = result
            sln.set_evals(torch.as_tensor(evals, dtype=self._eval_dtype).reshape(-1), torch.as_tensor(eval_data, dtype=self._eval_dtype).reshape(-1))
        elif isinstance(result, torch.Tensor) and result.ndim >= 1 and result.numel() == len(self._senses):
            sln.set_evals(result.reshape(-1))
        elif result is None:
            if not sln.is_evaluated:
                raise RuntimeError("Fitness function returned None but did not fill the solution's evals")
        else:
            sln.set_evaluation(float(result))

    def _evaluate(self, solution: Solution):
        """Per-solution evaluation; override in subclasses when no
        objective_func is given (reference core.py:2613)."""
        raise NotImplementedError("Either provide objective_func or override _evaluate/_evaluate_batch")

    def evaluate(self, batch: Union[SolutionBatch, Solution]):
        """Evaluate a batch. With an attached Comm and world_size > 1, each
        rank evaluates a contiguous shard of the rows and the evals are
        joined with one all_gather over xGMI (P1 in SURVEY.md §2.8)."""
        if isinstance(batch, Solution):
            batch = batch.to_batch()
        self._before_eval_hook(batch)
        comm = self._comm
        if comm is not None and comm.world_size > 1 and len(batch) >= comm.world_size:
            self._evaluate_sharded(batch, comm)
        elif self._use_eval_pool(batch):
            self._eval_pool.evaluate_into(self, batch)
        else:
            self._evaluate_batch(batch)
        if self._store_solution_stats:
            self._update_solution_stats(batch)
        self._after_eval_status = self._after_eval_hook.accumulate_dict(batch)

    def _evaluate_sharded(self, batch: SolutionBatch, comm):
        pieces = batch.split(comm.world_size)
        my_piece = pieces[comm.rank]
        self._evaluate_batch(my_piece)
        comm.all_gather_rows(batch.unsafe_evals, [pieces.indices_of(i) for i in range(len(pieces))])

    # -- process-pool evaluation (the reference's Ray actor system) -----------

    def _use_eval_pool(self, batch: SolutionBatch) -> bool:
        if not self._num_actors or self._num_actors < 2 or len(batch) < 2:
            return False
        if self._device.type != "cpu":
            return False  # GPU problems scale via torchrun + RCCL instead
        if self._eval_pool is None:
            from .parallel.evalpool import EvalPool

            self._eval_pool = EvalPool(
                self, self._num_actors, num_subbatches=self._num_subbatches, subbatch_size=self._subbatch_size
            )
        return True

    def kill_actors(self) -> None:
        """Shut down the evaluation worker pool (reference core.py:2117)."""
        pool, self._eval_pool = self._eval_pool, None
        if pool is not None:
            pool.close()

    def _update_solution_stats(self, batch: SolutionBatch):
        nobj = len(self._senses)
        if self._best is None:
            self._best = [None] * nobj
            self._worst = [None] * nobj
        for j, sense in enumerate(self._senses):
            evals = batch._evals[:, j]
            if bool(torch.isnan(evals).all()):
                continue
            bi = int(batch.argbest(j))
            wi = int(batch.argworst(j))
            cand_best = batch[bi]
            cand_worst = batch[wi]
            def better(a: float, b: float) -> bool:
                return a > b if sense == "max" else a < b
            if self._best[j] is None or better(float(cand_best.evals[j]), float(self._best[j].evals[j])):
                self._best[j] = cand_best.clone()
            if self._worst[j] is None or better(float(self._worst[j].evals[j]), float(cand_worst.evals[j])):
                self._worst[j] = cand_worst.clone()

    @property
    def stores_solution_stats(self) -> bool:
        return self._store_solution_stats

    @property
    def best(self):
        if self._best is None:
            return None
        return self._best[0] if len(self._senses) == 1 else list(self._best)

    @property
    def worst(self):
        if self._worst is None:
            return None
        return self._worst[0] if len(self._senses) == 1 else list(self._worst)

    # -- distribution-gradient machinery (P2) --------------------------------------

    def sample_and_compute_gradients(
        self,
        distribution,
        popsize: int,
        *,
        obj_index: Optional[int] = None,
        ranking_method: Optional[str] = None,
        num_interactions: Optional[int] = None,
        popsize_max: Optional[int] = None,
        ensure_even_popsize: bool = False,
        chunk_rows: Optional[int] = None,
    ) -> dict:
        """Sample a population from `distribution`, evaluate it, and return
        ``{"gradients": ..., "num_solutions": ..., "mean_eval": ...}``.

        Streaming mode (`chunk_rows` given): the N×L population is never
        materialized. Pass 1 samples and evaluates `chunk_rows` directions
        at a time keeping only the N fitnesses; utilities are ranked over
        the full fitness vector; pass 2 REGENERATES each chunk's noise from
        the same counter-addressed philox stream and accumulates the exact
        gradient. Memory is O(chunk_rows × L), so separable-Gaussian ES
        scales to parameter vectors far beyond HBM (L ≈ 10⁸⁺ with small
        chunks) — a capability the reference does not have (its population
        is always materialized). Requires a
        SeparableGaussian-family distribution; `num_interactions` adaptive
        popsize is not supported in this mode.

        SPMD mode (Comm attached, world > 1): each rank samples and
        evaluates ``popsize / world`` solutions using its own slice of the
        counter-based RNG stream; the utilities are ranked *globally* via an
        all-gather of the (tiny) fitness vector, after which each rank
        computes a partial gradient that is merged by a single all-reduce —
        the RCCL collapse of the reference's actor round-trip
        (core.py:2762-3074 → SURVEY.md §3.3).
        """
        obj_index = 0 if obj_index is None else int(obj_index)
        comm = self._comm
        self._before_grad_hook()
        if chunk_rows is not None:
            if int(chunk_rows) < 1:
                raise ValueError(f"chunk_rows must be >= 1, got {chunk_rows}")
            if num_interactions is not None:
                raise ValueError("chunk_rows (streaming gradients) does not support num_interactions adaptive popsize")
            result = self._sample_and_compute_gradients_streamed(distribution, int(popsize), obj_index, ranking_method, int(chunk_rows), comm)
        elif comm is not None:
            # SPMD path for ANY world size (world 1 degenerates to local
            # no-op collectives): one code path from the 1-GPU bench to the
            # 8-GPU scaling run, counter-addressed so the trajectory is
            # world-size invariant.
            result = self._sample_and_compute_gradients_sharded(distribution, int(popsize), obj_index, ranking_method, comm, num_interactions, popsize_max, ensure_even_popsize)
        else:
            result = self._sample_and_compute_gradients(distribution, int(popsize), obj_index, ranking_method, num_interactions, popsize_max, ensure_even_popsize)
        self._after_grad_status = self._after_grad_hook.accumulate_dict(result)
        return result

    def _sample_popsize(self, distribution, popsize: int, num_interactions, popsize_max, ensure_even: bool):
        if ensure_even and (popsize % 2 != 0):
            popsize += 1
        return popsize

    def _get_cached_grad_batch(self, popsize: int, device) -> "SolutionBatch":
        """Reusable sampling batch for the distribution-gradient path: a
        fresh 50 GB population tensor every generation would churn the
        caching allocator (measured: multi-second hipMalloc stalls at
        L=1e6, N=1e4); the batch is allocated once and re-sampled
        in place."""
        cached = getattr(self, "_grad_batch_cache", None)
        if cached is not None and len(cached) == popsize and cached.device == torch.device(device):
            cached.forget_evals()
            return cached
        batch = self.generate_batch(popsize, empty=True)
        self._grad_batch_cache = batch
        return batch

    def _sample_and_compute_gradients(
        self, distribution, popsize: int, obj_index: int, ranking_method, num_interactions=None, popsize_max=None, ensure_even_popsize: bool = False
    ) -> dict:
        popsize = self._sample_popsize(distribution, popsize, num_interactions, popsize_max, ensure_even_popsize)
        batch = self._get_cached_grad_batch(popsize, self._device)
        with record_range("sample"):
            distribution.sample(out=batch.access_values(), generator=self._generator)
        with record_range("evaluate"):
            self.evaluate(batch)
        values = batch._values
        fitnesses = batch._evals[:, obj_index]
        total_popsize = popsize
        if num_interactions is not None and not hasattr(self, "last_eval_interaction_count") and not getattr(self, "_warned_no_interactions", False):
            self._warned_no_interactions = True
            _logger.warning(
                "num_interactions=%s is set but %s does not report interaction counts "
                "(no `last_eval_interaction_count`); the adaptive-popsize loop is disabled.",
                num_interactions, type(self).__name__,
            )
        if num_interactions is not None and hasattr(self, "last_eval_interaction_count"):
            # adaptive popsize (reference core.py:3239-3274): keep sampling
            # extra sub-batches until the interaction threshold is reached
            interactions = int(getattr(self, "last_eval_interaction_count", 0) or 0)
            extra_values, extra_fits = [values], [fitnesses]
            while interactions < int(num_interactions) and not (popsize_max is not None and total_popsize >= int(popsize_max)):
                more = SolutionBatch(self, popsize=popsize, device=self._device, empty=True)
                distribution.sample(out=more.access_values(), generator=self._generator)
                self.evaluate(more)
                interactions += int(getattr(self, "last_eval_interaction_count", 0) or 0)
                extra_values.append(more._values)
                extra_fits.append(more._evals[:, obj_index])
                total_popsize += popsize
            if len(extra_values) > 1:
                values = torch.cat(extra_values)
                fitnesses = torch.cat(extra_fits)
        sense = self._senses[obj_index]
        grads = distribution.compute_gradients(values, fitnesses, objective_sense=sense, ranking_method=ranking_method)
        return {
            "gradients": grads,
            "num_solutions": total_popsize,
            "mean_eval": torch.nanmean(fitnesses),  # 0-dim tensor: no host sync
        }

    def _eval_shard_batch(self, batch: "SolutionBatch"):
        """Evaluate a rank-local shard with hooks/stats, deferring any
        cross-rank stat reductions into the gradient all-reduce."""
        self._before_eval_hook(batch)
        self._defer_fused_reduce = True
        try:
            self._evaluate_batch(batch)
            if self._store_solution_stats:
                self._update_solution_stats(batch)
            self._after_eval_status = self._after_eval_hook.accumulate_dict(batch)
        finally:
            self._defer_fused_reduce = False

    def _fused_allreduce_and_finalize(self, comm, sums: dict) -> dict:
        """ONE all-reduce carrying the gradient sums plus any deferred stat
        reductions (obs-norm (count, Σ, Σ²) etc.) — per-generation
        collective count stays at 2: fitness gather + this."""
        pending = self._fused_reduce_requests
        self._fused_reduce_requests = []
        container = dict(sums)
        for i, (tensor, _cb) in enumerate(pending):
            container[f"__fused{i}"] = tensor
        comm.all_reduce_container(container)
        for i, (_tensor, cb) in enumerate(pending):
            cb(container.pop(f"__fused{i}"))
        return container

    def _sample_and_compute_gradients_sharded(
        self, distribution, popsize: int, obj_index: int, ranking_method, comm, num_interactions=None, popsize_max=None, ensure_even_popsize: bool = False
    ) -> dict:
        """SPMD ES generation (the RCCL collapse of the reference's actor
        round-trip, core.py:2762-3074 → SURVEY.md §3.3).

        Rank r samples directions [r·D/W, (r+1)·D/W) of a GLOBAL virtual
        population, counter-addressed from the shared seed chain — the
        population (hence the trajectory, for deterministic fitness fns) is
        identical for every world size W dividing the popsize. Utilities
        are ranked globally after one all-gather of the fitness vector;
        gradients are raw per-rank partial sums merged by one fused
        all-reduce, with every weight-dependent normalization computed from
        the GLOBAL utility vector (round-1 ADVICE medium fix).

        Adaptive popsize (`num_interactions`, reference core.py:3239-3274):
        all ranks keep sampling synchronized extra rounds until the GLOBAL
        interaction count (one scalar all-reduce per round) reaches the
        threshold or `popsize_max` is hit.
        """
        world = comm.world_size
        rank = comm.rank
        symmetric = bool(getattr(distribution, "_symmetric", False))
        local_popsize = popsize // world
        if (ensure_even_popsize or symmetric) and local_popsize % 2 != 0:
            local_popsize += 1
        dirs_local = local_popsize // 2 if symmetric else local_popsize
        dirs_global = dirs_local * world
        ranking_used = ranking_method or "raw"
        sense = self._senses[obj_index]
        from .utils import ranking as ranking_mod

        if num_interactions is not None and not hasattr(self, "last_eval_interaction_count") and not getattr(self, "_warned_no_interactions", False):
            self._warned_no_interactions = True
            _logger.warning(
                "num_interactions=%s is set but %s does not report interaction counts "
                "(no `last_eval_interaction_count`); the adaptive-popsize loop is disabled.",
                num_interactions, type(self).__name__,
            )
        if num_interactions is not None and not hasattr(self, "last_eval_interaction_count"):
            num_interactions = None

        round_values: List[torch.Tensor] = []
        round_fits: List[torch.Tensor] = []
        total_popsize = 0
        interactions = 0
        while True:
            seed = self._next_spmd_seed()
            if len(round_values) == 0:
                batch = self._get_cached_grad_batch(local_popsize, self._device)
            else:
                batch = SolutionBatch(self, popsize=local_popsize, device=self._device, empty=True)
            use_overlap = (
                torch.device(self._device).type == "cuda"
                and hasattr(distribution, "fill_from_noise")
                and num_interactions is None  # adaptive rounds consume extra seeds
                # The overlap trades ~1.5x sampling HBM traffic (pregen z
                # write + affine read/write vs one fused write) for hiding
                # the philox latency — profitable when sampling is
                # latency-bound (small/medium populations), a net LOSS when
                # the population is tens of GB and everything is
                # bandwidth-bound (measured 82 vs 69 ms/gen at 50 GB).
                and local_popsize * self._solution_length * 4 <= (1 << 30)
            )
            with record_range("sample"):
                if use_overlap:
                    self._sample_with_pregen(distribution, batch.access_values(), seed, rank * dirs_local, dirs_local)
                    # pre-generate NEXT generation's noise on the side
                    # stream; it runs underneath the evaluation below and
                    # the gradient all-reduce after it
                    self._schedule_pregen(dirs_local, rank * dirs_local, self._peek_spmd_seed(1))
                else:
                    distribution.fill_counter_addressed(batch.access_values(), seed=seed, row_offset=rank * dirs_local)
            with record_range("evaluate"):
                self._eval_shard_batch(batch)
            round_values.append(batch._values)
            with record_range("fit_gather"):
                round_fits.append(comm.all_gather_vector(batch._evals[:, obj_index].contiguous()))
            total_popsize += local_popsize * world
            if num_interactions is None:
                break
            local_count = int(getattr(self, "last_eval_interaction_count", 0) or 0)
            count_t = torch.tensor([local_count], dtype=torch.int64)
            comm.all_reduce_(count_t)
            interactions += int(count_t.item())
            if interactions >= int(num_interactions):
                break
            if popsize_max is not None and total_popsize >= int(popsize_max):
                break

        all_fit = round_fits[0] if len(round_fits) == 1 else torch.cat(round_fits)
        all_utils = ranking_mod.rank(all_fit, ranking_used, higher_is_better=(sense == "max"))
        elite_mode = "parenthood_ratio" in getattr(distribution, "parameters", {})

        with record_range("partial_grads"):
            if elite_mode:
                num_elites = math.floor(total_popsize * float(distribution.parameters["parenthood_ratio"]))
                global_elite = torch.zeros(total_popsize, dtype=torch.bool, device=all_utils.device)
                global_elite[all_utils.argsort(descending=True)[:num_elites]] = True
                # local-row elite mask: global row of my local row i in round
                # k is k·local·world + rank·local + i (gather is rank-major)
                mask = torch.zeros(len(round_values) * local_popsize, dtype=torch.bool, device=all_utils.device)
                for k in range(len(round_values)):
                    base = k * local_popsize * world + rank * local_popsize
                    mask[k * local_popsize : (k + 1) * local_popsize] = global_elite[base : base + local_popsize]
                sum_x, sum_x2 = distribution.accumulate_elite_sums_streamed(
                    ((values, k * local_popsize, local_popsize) for k, values in enumerate(round_values)), mask
                )
                sums = {"x": sum_x, "x2": sum_x2}
            else:
                w_all = distribution.prepare_weights_global(all_utils, ranking_used)
                w_dtype = self._dtype if self._dtype.is_floating_point else torch.float32
                sums = None
                for k, values in enumerate(round_values):
                    base = k * local_popsize * world + rank * local_popsize
                    my_w = w_all[base : base + local_popsize].to(dtype=w_dtype)
                    part = distribution.partial_grad_sums(values, my_w)
                    if sums is None:
                        sums = part
                    else:
                        for key in sums:
                            sums[key] = sums[key] + part[key]

        with record_range("grad_allreduce"):
            sums = self._fused_allreduce_and_finalize(comm, sums)

        if elite_mode:
            grads = distribution.finalize_elite_gradients(sums["x"].to(torch.float64), sums["x2"].to(torch.float64), num_elites)
        else:
            grads = distribution.finalize_shard_gradients(sums, w_all, ranking_used)
        return {
            "gradients": grads,
            "num_solutions": total_popsize,
            "mean_eval": torch.nanmean(all_fit),  # 0-dim tensor: no host sync
        }

    def _get_stream_batch(self, popsize: int) -> "SolutionBatch":
        cache = getattr(self, "_stream_batch_cache", None)
        if cache is None:
            cache = self._stream_batch_cache = {}
        batch = cache.get(popsize)
        if batch is None or batch.device != torch.device(self._device):
            batch = self.generate_batch(popsize, empty=True)
            cache.clear()  # at most the main chunk size + one tail size alive
            cache[popsize] = batch
        batch.forget_evals()
        return batch

    def _sample_and_compute_gradients_streamed(
        self, distribution, popsize: int, obj_index: int, ranking_method, chunk_rows: int, comm
    ) -> dict:
        """Two-pass streaming ES gradients — see sample_and_compute_gradients.

        Uses the same shard-gradient protocol as the non-streamed SPMD path
        (prepare weights globally → raw partial sums per chunk → one fused
        all-reduce → finalize with global normalizers), with the chunk axis
        playing the role of extra shards. Rank r owns directions
        [r·D/W, (r+1)·D/W) of the global virtual population, addressed by
        per-row philox streams from the shared seed chain — regenerable in
        pass 2 without storing the N×L population."""
        if not hasattr(distribution, "fill_counter_addressed"):
            raise ValueError(f"{type(distribution).__name__} does not support streamed gradients")
        elite_mode = "parenthood_ratio" in getattr(distribution, "parameters", {})
        from .ops.dispatch import _seed_from_generator
        from .utils import ranking as ranking_mod

        world = comm.world_size if comm is not None else 1
        my_rank = comm.rank if comm is not None else 0
        local_popsize = popsize // world
        symmetric = bool(getattr(distribution, "_symmetric", False))
        if symmetric and local_popsize % 2 != 0:
            local_popsize += 1
        directions = local_popsize // 2 if symmetric else local_popsize
        my_dir0 = my_rank * directions
        # stream-per-row philox addressing: any chunk boundary is exact
        chunk_rows = min(chunk_rows, directions)
        seed = self._next_spmd_seed() if comm is not None else _seed_from_generator(self._generator, self._device)

        fits = torch.empty(local_popsize, dtype=self._eval_dtype, device=self._device)

        def chunk_spans():
            r0 = 0
            while r0 < directions:
                rows = min(chunk_rows, directions - r0)
                yield r0, rows
                r0 += rows

        # -- pass 1: sample + evaluate, keep fitnesses only ------------------
        for r0, rows in chunk_spans():
            batch = self._get_stream_batch(rows * 2 if symmetric else rows)
            with record_range("stream_sample"):
                distribution.fill_counter_addressed(batch.access_values(), seed=seed, row_offset=my_dir0 + r0)
            with record_range("stream_eval"):
                self._eval_shard_batch(batch)
            evals = batch._evals[:, obj_index]
            if symmetric:
                fits[r0 : r0 + rows] = evals[:rows]
                fits[directions + r0 : directions + r0 + rows] = evals[rows:]
            else:
                fits[r0 : r0 + rows] = evals

        # -- global ranking ---------------------------------------------------
        if comm is not None and world > 1:
            all_fit = comm.all_gather_vector(fits)
        else:
            all_fit = fits
        sense = self._senses[obj_index]
        ranking_used = ranking_method or "raw"
        all_utils = ranking_mod.rank(all_fit, ranking_used, higher_is_better=(sense == "max"))
        total = world * local_popsize

        # -- pass 2: regenerate noise per chunk, accumulate exact gradients --
        def regen_chunks():
            for r0, rows in chunk_spans():
                batch = self._get_stream_batch(rows * 2 if symmetric else rows)
                with record_range("stream_regen"):
                    distribution.fill_counter_addressed(batch.access_values(), seed=seed, row_offset=my_dir0 + r0)
                yield batch._values, r0, rows

        with record_range("stream_grad"):
            if elite_mode:
                # elite SET chosen from the GLOBAL utilities; each rank
                # accumulates its local elite members' (Σx, Σx²); the sums
                # (not a weighted gradient average — elite stats are
                # non-linear in the shards) go through the fused all-reduce
                num_elites = math.floor(total * float(distribution.parameters["parenthood_ratio"]))
                global_elite = torch.zeros(total, dtype=torch.bool, device=all_utils.device)
                global_elite[all_utils.argsort(descending=True)[:num_elites]] = True
                my_elite = global_elite[my_rank * local_popsize : (my_rank + 1) * local_popsize]
                sum_x, sum_x2 = distribution.accumulate_elite_sums_streamed(regen_chunks(), my_elite)
                sums = {"x": sum_x, "x2": sum_x2}
                if comm is not None:
                    sums = self._fused_allreduce_and_finalize(comm, sums)
                grads = distribution.finalize_elite_gradients(
                    sums["x"].to(torch.float64), sums["x2"].to(torch.float64), num_elites)
            else:
                w_all = distribution.prepare_weights_global(all_utils, ranking_used)
                w_dtype = self._dtype if self._dtype.is_floating_point else torch.float32
                my_w = w_all[my_rank * local_popsize : (my_rank + 1) * local_popsize].to(dtype=w_dtype)
                sums = None
                for values_chunk, r0, rows in regen_chunks():
                    if symmetric:
                        w_c = torch.cat([my_w[r0 : r0 + rows], my_w[directions + r0 : directions + r0 + rows]])
                    else:
                        w_c = my_w[r0 : r0 + rows]
                    part = distribution.partial_grad_sums(values_chunk, w_c)
                    if sums is None:
                        sums = part
                    else:
                        for key in sums:
                            sums[key] = sums[key] + part[key]
                if comm is not None:
                    sums = self._fused_allreduce_and_finalize(comm, sums)
                grads = distribution.finalize_shard_gradients(sums, w_all, ranking_used)
        return {
            "gradients": grads,
            "num_solutions": total,
            "mean_eval": torch.nanmean(all_fit),  # 0-dim tensor: no host sync
        }

    # -- functional adapter ---------------------------------------------------------

    def make_callable_evaluator(self, *, obj_index: Optional[int] = None) -> "ProblemBoundEvaluator":
        if is_dtype_object(self._dtype):
            return ObjectTypedProblemBoundEvaluator(self, obj_index=obj_index)
        return ProblemBoundEvaluator(self, obj_index=obj_index)

    # -- cloning / pickling -----------------------------------------------------------

    def _get_cloned_state(self, *, memo: dict) -> dict:
        state = {}
        for k, v in self.__dict__.items():
            if k in ("_generator", "_comm", "_grad_batch_cache", "_pregen", "_pregen_stream", "_graph_state", "_eval_pool"):
                state[k] = None
            else:
                state[k] = deep_clone(v, otherwise_deepcopy=True, memo=memo)
        return state

    def __setstate__(self, state: dict):
        super().__setstate__(state)
        if self._seed is not None and self._generator is None:
            self._generator = torch.Generator(device=self._device)
            self._generator.manual_seed(int(self._seed))

    def __repr__(self) -> str:
        return f"<{type(self).__name__} senses={self._senses} length={self._solution_length} dtype={self._dtype} device={self._device}>"


class ProblemBoundEvaluator:
    """Adapts a Problem into a pure callable ``f(values_2d) -> evals`` for
    the functional API (reference core.py:5109-5256). Supports an extra
    leading batch dimension (the batched-search axis)."""

    def __init__(self, problem: Problem, *, obj_index: Optional[int] = None):
        self._problem = problem
        self._obj_index = 0 if obj_index is None else int(obj_index)
        if not (0 <= self._obj_index < len(problem.senses)):
            raise IndexError(f"obj_index {obj_index} out of range for {len(problem.senses)} objectives")

    @property
    def problem(self) -> Problem:
        return self._problem

    def __call__(self, values: torch.Tensor) -> torch.Tensor:
        problem = self._problem
        if values.ndim == 2:
            batch = problem.generate_batch(values.shape[0], empty=True)
            batch.access_values().copy_(values.to(batch.device, dtype=batch.dtype))
            problem.evaluate(batch)
            return batch._evals[:, self._obj_index].to(values.device)
        if values.ndim > 2:
            lead = values.shape[:-2]
            flat = values.reshape(-1, values.shape[-1])
            # Evaluate each (batched search) population in one flat batch
            out = self(flat)
            return out.reshape(lead + (values.shape[-2],))
        raise ValueError(f"Expected values of ndim >= 2, got shape {tuple(values.shape)}")

class ObjectTypedProblemBoundEvaluator:
    """Callable evaluator for object-dtype problems (reference
    core.py:5201): values arrive as an ObjectArray (or any sequence) and
    evals come back as a 1-D tensor; no extra batch dimensions (object
    populations cannot be vmapped)."""

    def __init__(self, problem: "Problem", *, obj_index: Optional[int] = None):
        if not is_dtype_object(problem.dtype):
            raise TypeError(
                f"Expected an object-dtype problem, got dtype {problem.dtype}."
                " Hint: use ProblemBoundEvaluator (make_callable_evaluator) for tensor-dtype problems."
            )
        self._problem = problem
        self._obj_index = 0 if obj_index is None else int(obj_index)

    @property
    def problem(self) -> "Problem":
        return self._problem

    def __call__(self, values) -> torch.Tensor:
        problem = self._problem
        n = len(values)
        batch = SolutionBatch(problem, popsize=n, empty=True)
        for i in range(n):
            batch.access_values()[i] = values[i]
        problem.evaluate(batch)
        return batch._evals[:, self._obj_index].clone()
