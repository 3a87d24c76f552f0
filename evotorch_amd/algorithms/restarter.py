"""Restart meta-algorithms: Restart, ModifyingRestart, IPOP.

Reference parity: /root/reference/src/evotorch/algorithms/restarter/
(restart.py:21, modify_restart.py:23,34). The inner searcher is
re-instantiated whenever it reports termination; IPOP doubles the
population size on each restart (Auger & Hansen 2005).
"""

from typing import Callable, Optional

import torch

from ..core import Problem
from .searchalgorithm import SearchAlgorithm

__all__ = ["BIPOP", "IPOP", "ModifyingRestart", "Restart"]


class Restart(SearchAlgorithm):
    """Repeatedly runs a searcher built by `algorithm_factory`; when
    `termination_criterion(searcher) -> bool` fires (default: the inner
    searcher exposes a truthy status item "terminated", or its stdev
    collapsed below `min_stdev`), a fresh searcher is created."""

    def __init__(
        self,
        problem: Problem,
        algorithm_factory: Callable[..., SearchAlgorithm],
        *,
        algorithm_args: Optional[dict] = None,
        termination_criterion: Optional[Callable[[SearchAlgorithm], bool]] = None,
        min_stdev: Optional[float] = None,
        max_inner_steps: Optional[int] = None,
    ):
        super().__init__(
            problem,
            num_restarts=lambda: self._num_restarts,
            inner_iter=lambda: self._inner_steps,
        )
        self._factory = algorithm_factory
        self._algorithm_args = dict(algorithm_args or {})
        self._criterion = termination_criterion
        self._min_stdev = min_stdev
        self._max_inner_steps = max_inner_steps
        self._num_restarts = 0
        self._inner_steps = 0
        self._searcher: Optional[SearchAlgorithm] = None

    @property
    def search_algorithm(self) -> Optional[SearchAlgorithm]:
        return self._searcher

    @property
    def num_restarts(self) -> int:
        return self._num_restarts

    @property
    def algorithm_args(self) -> dict:
        """The LIVE keyword arguments the next restart will construct the
        inner searcher with — `ModifyingRestart`'s modify callback (and
        subclasses like IPOP) mutate this dict in place."""
        return self._algorithm_args

    def _current_args(self) -> dict:
        return dict(self._algorithm_args)

    def _make_searcher(self) -> SearchAlgorithm:
        return self._factory(self.problem, **self._current_args())

    def _terminated(self) -> bool:
        s = self._searcher
        if self._criterion is not None and self._criterion(s):
            return True
        if "terminated" in s.status and bool(s.status["terminated"]):
            return True
        if self._min_stdev is not None and "stdev" in s.status:
            stdev = s.status["stdev"]
            if isinstance(stdev, torch.Tensor) and float(torch.Tensor.as_subclass(stdev, torch.Tensor).max()) < self._min_stdev:
                return True
        if self._max_inner_steps is not None and self._inner_steps >= self._max_inner_steps:
            return True
        return False

    def _on_restart(self):
        """Hook for subclasses: adjust `self._algorithm_args` before the
        next inner searcher is created."""

    def _step(self):
        if self._searcher is None:
            self._searcher = self._make_searcher()
            self._inner_steps = 0
        self._searcher.step()
        self._inner_steps += 1
        for k, v in self._searcher.status.items():
            if k != "iter":
                self.update_status({k: v})
        if self._terminated():
            self._num_restarts += 1
            self._on_restart()
            self._searcher = None


class ModifyingRestart(Restart):
    """Restart variant that calls `modify(self)` before each restart to
    adjust the inner algorithm's arguments (reference
    modify_restart.py:23)."""

    def __init__(self, problem, algorithm_factory, *, modify: Optional[Callable[["ModifyingRestart"], None]] = None, **kwargs):
        super().__init__(problem, algorithm_factory, **kwargs)
        self._modify = modify

    def _on_restart(self):
        if self._modify is not None:
            self._modify(self)


class IPOP(ModifyingRestart):
    """IPOP restart strategy: double `popsize` on every restart
    (reference modify_restart.py:34)."""

    def __init__(self, problem, algorithm_factory, *, popsize_multiplier: float = 2.0, **kwargs):
        super().__init__(problem, algorithm_factory, **kwargs)
        self._popsize_multiplier = float(popsize_multiplier)

    def _on_restart(self):
        if "popsize" in self._algorithm_args and self._algorithm_args["popsize"]:
            self._algorithm_args["popsize"] = int(self._algorithm_args["popsize"] * self._popsize_multiplier)
        super()._on_restart()


class BIPOP(ModifyingRestart):
    """BIPOP restart strategy (Hansen 2009): restarts alternate between a
    LARGE regime whose population doubles each time (IPOP-style) and a
    SMALL regime whose population is drawn as
    ``λ_default · (λ_large / (2·λ_default)) ** U[0,1]²`` with at most half
    the large regime's evaluation budget. Each restart enters the regime
    that has consumed fewer evaluations so far.

    The reference only reaches BIPOP through the external `cma` package
    (`PyCMAES(..., cma_options)`); this is a native implementation usable
    with any popsize-parameterized searcher (typically `CMAES`).
    """

    def __init__(
        self,
        problem: Problem,
        algorithm_factory: Callable[..., SearchAlgorithm],
        *,
        algorithm_args: Optional[dict] = None,
        seed: Optional[int] = None,
        **kwargs,
    ):
        algorithm_args = dict(algorithm_args or {})
        if not algorithm_args.get("popsize"):
            import math

            algorithm_args["popsize"] = int(4 + math.floor(3 * math.log(problem.solution_length)))
        super().__init__(problem, algorithm_factory, algorithm_args=algorithm_args, **kwargs)
        self._default_popsize = int(algorithm_args["popsize"])
        self._large_popsize = self._default_popsize
        self._large_evals = 0
        self._small_evals = 0
        self._regime = "large"  # the first run counts as a large run
        self._rng = torch.Generator().manual_seed(seed if seed is not None else 0)
        self.add_status_getters({
            "regime": lambda: self._regime,
            "current_popsize": lambda: int(self._algorithm_args.get("popsize", 0)),
        })

    def _on_restart(self):
        spent = self._inner_steps * int(self._algorithm_args.get("popsize", self._default_popsize))
        if self._regime == "large":
            self._large_evals += spent
        else:
            self._small_evals += spent
        if self._small_evals < self._large_evals:
            # small regime: randomized popsize in [λ_default, λ_large/2]
            self._regime = "small"
            u = float(torch.rand((), generator=self._rng))
            ratio = max(1.0, self._large_popsize / (2.0 * self._default_popsize))
            popsize = int(self._default_popsize * (ratio ** (u * u)))
            self._algorithm_args["popsize"] = max(popsize, 4)
        else:
            self._regime = "large"
            self._large_popsize = int(self._large_popsize * 2)
            self._algorithm_args["popsize"] = self._large_popsize
        super()._on_restart()
