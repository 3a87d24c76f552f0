"""SearchAlgorithm base, lazy status reporting, and the single-population
status mixin.

Reference parity: /root/reference/src/evotorch/algorithms/searchalgorithm.py
(LazyReporter :34, LazyStatusDict :185, SearchAlgorithm :240,
SinglePopulationAlgorithmMixin :450).
"""

from collections.abc import Mapping
from datetime import datetime
from typing import Any, Callable, Optional

import torch

from ..core import Problem
from ..utils import Hook, ReadOnlyTensor, as_read_only_tensor
from ..utils.recursiveprintable import RecursivePrintable

__all__ = ["LazyReporter", "LazyStatusDict", "SearchAlgorithm", "SinglePopulationAlgorithmMixin"]


class LazyReporter:
    """Maintains a status dict whose items are computed lazily by getter
    functions and cached until `clear_status()`."""

    def __init__(self, **getters: Callable):
        self._getters: dict = dict(getters)
        self._computed: dict = {}

    def add_status_getters(self, getters: Mapping):
        self._getters.update(getters)

    def get_status_value(self, key: str) -> Any:
        if key not in self._computed:
            if key not in self._getters:
                raise KeyError(key)
            value = self._getters[key]()
            if isinstance(value, torch.Tensor) and not isinstance(value, ReadOnlyTensor):
                value = as_read_only_tensor(value)
            self._computed[key] = value
        return self._computed[key]

    def has_status_key(self, key: str) -> bool:
        return key in self._getters or key in self._computed

    def is_status_computed(self, key: str) -> bool:
        """True if the status item is already materialized (reading it will
        not trigger its getter) — reference searchalgorithm.py:118."""
        return key in self._computed

    def iter_status_keys(self):
        seen = set()
        for k in self._computed:
            seen.add(k)
            yield k
        for k in self._getters:
            if k not in seen:
                yield k

    def clear_status(self):
        self._computed.clear()

    def update_status(self, additional_status: Mapping):
        for k, v in additional_status.items():
            if isinstance(v, torch.Tensor) and not isinstance(v, ReadOnlyTensor):
                v = as_read_only_tensor(v)
            self._computed[k] = v

    @property
    def status(self) -> "LazyStatusDict":
        return LazyStatusDict(self)


class LazyStatusDict(Mapping, RecursivePrintable):
    """Mapping view over a LazyReporter: looking up a key triggers its lazy
    computation."""

    def __init__(self, reporter: LazyReporter):
        self._reporter = reporter

    def __getitem__(self, key: str) -> Any:
        return self._reporter.get_status_value(key)

    def __iter__(self):
        return iter(list(self._reporter.iter_status_keys()))

    def __len__(self) -> int:
        return len(list(self._reporter.iter_status_keys()))

    def __contains__(self, key) -> bool:
        return self._reporter.has_status_key(key)

    def to_string(self, *, max_depth: int = 10) -> str:
        keys = ", ".join(repr(k) for k in self)
        return f"<LazyStatusDict keys=[{keys}]>"


class SearchAlgorithm(LazyReporter):
    """Base class of all searchers: `step()` advances one generation,
    `run(n)` loops, hooks observe. Subclasses implement `_step()`."""

    def __init__(self, problem: Problem, **kwargs: Callable):
        super().__init__(**kwargs)
        self._problem = problem
        self._before_step_hook = Hook()
        self._after_step_hook = Hook()
        self._log_hook = Hook()
        self._end_of_run_hook = Hook()
        self._steps_count = 0
        self._first_step_datetime: Optional[datetime] = None

    @property
    def problem(self) -> Problem:
        return self._problem

    @property
    def before_step_hook(self) -> Hook:
        return self._before_step_hook

    @property
    def after_step_hook(self) -> Hook:
        return self._after_step_hook

    @property
    def log_hook(self) -> Hook:
        return self._log_hook

    @property
    def end_of_run_hook(self) -> Hook:
        return self._end_of_run_hook

    @property
    def step_count(self) -> int:
        return self._steps_count

    @property
    def steps_count(self) -> int:  # reference-compatible alias
        return self._steps_count

    @property
    def first_step_datetime(self) -> Optional[datetime]:
        return self._first_step_datetime

    def _step(self):
        raise NotImplementedError

    def step(self):
        """One generation: clear status, run hooks, `_step()`, increment the
        counter, then feed the log hook with the fresh status."""
        self._before_step_hook()
        self.clear_status()
        if self._first_step_datetime is None:
            self._first_step_datetime = datetime.now()
        self._step()
        self._steps_count += 1
        self.update_status({"iter": self._steps_count})
        self.update_status(self._problem.status)
        extra = self._after_step_hook.accumulate_dict()
        if extra:
            self.update_status(extra)
        if len(self._log_hook) >= 1:
            self._log_hook(dict(self.status))

    def run(self, num_generations: int, *, reset_first_step_datetime: bool = True):
        if reset_first_step_datetime:
            self.reset_first_step_datetime()
        for _ in range(int(num_generations)):
            self.step()
        if len(self._end_of_run_hook) >= 1:
            self._end_of_run_hook(dict(self.status))

    def reset_first_step_datetime(self):
        self._first_step_datetime = None

    @property
    def is_terminated(self) -> bool:
        """Whether the searcher reached a terminal state (always False in
        the base; Restart wrappers consult this) — reference
        searchalgorithm.py:445."""
        return False

    # -- checkpoint / resume -------------------------------------------------
    # (a green-field addition relative to the reference, which only offers
    # whole-object pickling and PicklingLogger snapshots — SURVEY.md §5.4)

    def _state_items(self) -> dict:
        """Subclass hook: the tensors/values that define search progress."""
        return {}

    def _load_state_items(self, state: dict):
        raise NotImplementedError(f"{type(self).__name__} does not support load_state_dict")

    def state_dict(self) -> dict:
        """Resumable search state (cpu tensors): algorithm parameters plus
        any problem-side state (observation normalization, counters)."""
        import torch as _torch

        def to_cpu(x):
            if isinstance(x, _torch.Tensor):
                return x.detach().cpu().clone()
            if isinstance(x, dict):
                return {k: to_cpu(v) for k, v in x.items()}
            return x

        state = {"steps_count": self._steps_count, "algorithm": type(self).__name__}
        state["items"] = to_cpu(self._state_items())
        problem = self._problem
        if hasattr(problem, "obs_norm"):
            rn = problem.obs_norm
            c, s, ss = rn.stats_triple()
            state["obs_norm"] = {"count": c.cpu(), "sum": s.cpu(), "sum_sq": ss.cpu()}
        return state

    def load_state_dict(self, state: dict):
        if state.get("algorithm") not in (type(self).__name__, None):
            raise ValueError(f"Checkpoint is for {state.get('algorithm')}, not {type(self).__name__}")
        self._steps_count = int(state.get("steps_count", 0))
        self._load_state_items(state.get("items", {}))
        problem = self._problem
        if "obs_norm" in state and hasattr(problem, "obs_norm"):
            rn = problem.obs_norm
            rn.reset()
            o = state["obs_norm"]
            rn.update((o["count"], o["sum"], o["sum_sq"]))
        return self


class SinglePopulationAlgorithmMixin:
    """Adds pop_best / mean_eval / median_eval / pop_best_eval status items
    for searchers exposing a `population` property (reference
    searchalgorithm.py:450; per-objective variants for multi-objective)."""

    def __init__(self, *, exclude: Optional[set] = None, enable: bool = True):
        if not enable:
            return
        exclude = exclude or set()
        problem: Problem = self.problem
        is_multi = problem.is_multi_objective

        def add(name: str, fn: Callable):
            if name not in exclude:
                self.add_status_getters({name: fn})

        if is_multi:
            for j, sense in enumerate(problem.senses):
                add(f"obj{j}_pop_best", self._make_pop_best_getter(j))
                add(f"obj{j}_pop_best_eval", self._make_pop_best_eval_getter(j))
                add(f"obj{j}_mean_eval", self._make_mean_eval_getter(j))
                add(f"obj{j}_median_eval", self._make_median_eval_getter(j))
        else:
            add("pop_best", self._make_pop_best_getter(0))
            add("pop_best_eval", self._make_pop_best_eval_getter(0))
            add("mean_eval", self._make_mean_eval_getter(0))
            add("median_eval", self._make_median_eval_getter(0))
            add("best", lambda: problem.best)
            add("worst", lambda: problem.worst)
            if problem.stores_solution_stats:
                add("best_eval", lambda: (None if problem.best is None else float(problem.best.evals[0])))
                add("worst_eval", lambda: (None if problem.worst is None else float(problem.worst.evals[0])))

    def _make_pop_best_getter(self, obj_index: int):
        def getter():
            pop = self.population
            if pop is None:
                return None
            return pop[int(pop.argbest(obj_index))].clone()

        return getter

    def _make_pop_best_eval_getter(self, obj_index: int):
        def getter():
            pop = self.population
            if pop is None:
                return None
            return float(pop.evals[int(pop.argbest(obj_index)), obj_index])

        return getter

    def _make_mean_eval_getter(self, obj_index: int):
        def getter():
            pop = self.population
            if pop is None:
                return None
            return float(torch.nanmean(torch.Tensor.as_subclass(pop.evals, torch.Tensor)[:, obj_index]))

        return getter

    def _make_median_eval_getter(self, obj_index: int):
        def getter():
            pop = self.population
            if pop is None:
                return None
            return float(torch.nanmedian(torch.Tensor.as_subclass(pop.evals, torch.Tensor)[:, obj_index]))

        return getter
