"""Distribution-based searchers: PGPE, SNES, CEM, XNES on a common
GaussianSearchAlgorithm base.

Reference parity: /root/reference/src/evotorch/algorithms/distributed/
gaussian.py:35-1404. Two execution modes:

* non-distributed: sample → evaluate → rank → gradient → update, all on one
  device (the per-generation HIP kernel chain K1→eval→K2→K3→K4).
* distributed=True: delegates to `Problem.sample_and_compute_gradients`,
  which in SPMD mode (Comm attached) runs per-rank sampling/evaluation and
  merges the gradients with a single RCCL all-reduce — the collapse of the
  reference's actor round-trip (SURVEY.md §3.3 → §2.8 P2).
"""

import math
from typing import Optional

import torch

from ..core import Problem, SolutionBatch
from ..distributions import (
    Distribution,
    ExpGaussian,
    ExpSeparableGaussian,
    SeparableGaussian,
    SymmetricSeparableGaussian,
)
from ..optimizers import get_optimizer_class
from ..utils import RealOrVector, modify_tensor, to_stdev_init
from ..utils.misc import ensure_tensor_length_and_dtype
from ..utils.profiling import record_range
from .searchalgorithm import SearchAlgorithm, SinglePopulationAlgorithmMixin

__all__ = ["GaussianSearchAlgorithm", "PGPE", "SNES", "CEM", "XNES"]


class GaussianSearchAlgorithm(SearchAlgorithm, SinglePopulationAlgorithmMixin):
    """Base for searchers driving a Gaussian-family Distribution."""

    DISTRIBUTION_TYPE = NotImplemented
    DISTRIBUTION_PARAMS: Optional[dict] = NotImplemented

    def __init__(
        self,
        problem: Problem,
        *,
        popsize: int,
        center_learning_rate: float,
        stdev_learning_rate: float,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        num_interactions: Optional[int] = None,
        popsize_max: Optional[int] = None,
        optimizer=None,
        optimizer_config: Optional[dict] = None,
        ranking_method: Optional[str] = None,
        center_init: Optional[RealOrVector] = None,
        stdev_min: Optional[RealOrVector] = None,
        stdev_max: Optional[RealOrVector] = None,
        stdev_max_change: Optional[RealOrVector] = None,
        obj_index: Optional[int] = None,
        distributed: bool = False,
        popsize_weighted_grad_avg: Optional[bool] = None,
        ensure_even_popsize: bool = False,
        grad_chunk_rows: Optional[int] = None,
    ):
        problem.ensure_numeric()
        problem.ensure_unbounded()

        SearchAlgorithm.__init__(
            self,
            problem,
            center=self._get_mu,
            stdev=self._get_sigma,
            mean_eval=self._get_mean_eval,
        )

        self._ensure_even_popsize = bool(ensure_even_popsize)
        if self._ensure_even_popsize and popsize % 2 != 0:
            raise ValueError(f"popsize must be even for this searcher, got {popsize}")

        if center_init is None:
            mu = problem.generate_values(1).reshape(-1)
        else:
            mu = ensure_tensor_length_and_dtype(center_init, problem.solution_length, problem.dtype, about="center_init", device=problem.device)

        stdev_spec = to_stdev_init(solution_length=problem.solution_length, stdev_init=stdev_init, radius_init=radius_init)
        sigma = ensure_tensor_length_and_dtype(stdev_spec, problem.solution_length, problem.dtype, about="stdev_init", device=problem.device)

        dist_cls = self.DISTRIBUTION_TYPE
        dist_params = dict(self.DISTRIBUTION_PARAMS or {})
        dist_params.update({"mu": mu, "sigma": sigma})
        self._distribution: Distribution = dist_cls(dist_params, dtype=problem.dtype, device=problem.device)

        self._popsize = int(popsize)
        self._popsize_max = None if popsize_max is None else int(popsize_max)
        self._num_interactions = None if num_interactions is None else int(num_interactions)
        self._center_learning_rate = float(center_learning_rate)
        self._stdev_learning_rate = float(stdev_learning_rate)
        self._optimizer = self._initialize_optimizer(self._center_learning_rate, optimizer, optimizer_config)
        self._ranking_method = None if ranking_method is None else str(ranking_method)
        if obj_index is None and problem.is_multi_objective:
            raise ValueError("For a multi-objective problem, a Gaussian searcher needs an explicit obj_index")
        self._obj_index = 0 if obj_index is None else int(obj_index)

        def opt_length_tensor(x, about):
            if x is None:
                return None
            return ensure_tensor_length_and_dtype(x, problem.solution_length, problem.dtype, about=about, allow_scalar=True, device=problem.device)

        self._stdev_min = opt_length_tensor(stdev_min, "stdev_min")
        self._stdev_max = opt_length_tensor(stdev_max, "stdev_max")
        self._stdev_max_change = opt_length_tensor(stdev_max_change, "stdev_max_change")

        self._distributed = bool(distributed)
        self._grad_chunk_rows = None if grad_chunk_rows is None else int(grad_chunk_rows)
        if self._grad_chunk_rows is not None and not self._distributed:
            raise ValueError("grad_chunk_rows (streaming gradients) requires distributed=True")
        if not self._distributed and popsize_weighted_grad_avg is not None:
            raise ValueError("popsize_weighted_grad_avg is only meaningful with distributed=True")
        # Accepted for reference compatibility. In the reference, actors may
        # evaluate different popsizes, making weighted vs unweighted
        # averaging distinct (gaussian.py:246-271); here every rank owns an
        # equal popsize/world shard, so the two averages coincide and the
        # sharded path's local/total scaling is already the weighted form.
        self._popsize_weighted_grad_avg = popsize_weighted_grad_avg

        self._population: Optional[SolutionBatch] = None
        self._mean_eval: Optional[float] = None
        self._first_iter = True

        SinglePopulationAlgorithmMixin.__init__(self, enable=not self._distributed)
        if self._num_interactions is not None and not self._distributed:
            self.add_status_getters({"popsize": lambda: 0 if self._population is None else len(self._population)})

    def _initialize_optimizer(self, learning_rate: float, optimizer=None, optimizer_config: Optional[dict] = None):
        if optimizer is None:
            return None
        if isinstance(optimizer, str):
            cls = get_optimizer_class(optimizer, optimizer_config)
            return cls(
                stepsize=float(learning_rate),
                solution_length=self.problem.solution_length,
                dtype=self.problem.dtype,
                device=self.problem.device,
            )
        if callable(optimizer) and not hasattr(optimizer, "ascent"):
            return optimizer(
                stepsize=float(learning_rate),
                solution_length=self.problem.solution_length,
                dtype=self.problem.dtype,
                device=self.problem.device,
                **(optimizer_config or {}),
            )
        return optimizer

    # -- stepping -----------------------------------------------------------

    def _step(self):
        if self._distributed:
            self._step_distributed()
        else:
            self._step_non_distributed()


    def _step_distributed(self):
        fetched = self.problem.sample_and_compute_gradients(
            self._distribution,
            self._popsize,
            popsize_max=self._popsize_max,
            obj_index=self._obj_index,
            num_interactions=self._num_interactions,
            ranking_method=self._ranking_method,
            ensure_even_popsize=self._ensure_even_popsize,
            chunk_rows=self._grad_chunk_rows,
        )
        self._update_distribution(fetched["gradients"])
        self._mean_eval = fetched["mean_eval"]
        self.update_status({"num_solutions": int(fetched["num_solutions"])})

    def _fill_and_eval_pop(self):
        problem = self.problem
        if self._num_interactions is None:
            if self._population is None:
                self._population = SolutionBatch(problem, popsize=self._popsize, device=self._distribution.device, empty=True)
            self._distribution.sample(out=self._population.access_values(), generator=problem)
            problem.evaluate(self._population)
            return
        # Adaptive popsize (Toklu et al. 2020): keep sampling sub-batches
        # until the interaction threshold is reached or popsize_max is hit
        # (reference core.py:3239-3274).
        if not hasattr(problem, "last_eval_interaction_count"):
            import warnings

            warnings.warn(
                f"num_interactions={self._num_interactions} is set but {type(problem).__name__} does not "
                "report interaction counts (no `last_eval_interaction_count`); adaptive popsize is disabled.",
                stacklevel=2,
            )
        interactions = 0
        batches = []
        total = 0
        while True:
            n = self._popsize
            if self._ensure_even_popsize and n % 2 != 0:
                n += 1
            batch = SolutionBatch(problem, popsize=n, device=self._distribution.device, empty=True)
            self._distribution.sample(out=batch.access_values(), generator=problem)
            problem.evaluate(batch)
            batches.append(batch)
            total += n
            interactions += int(getattr(problem, "last_eval_interaction_count", 0) or 0)
            if self._num_interactions is not None and interactions >= self._num_interactions:
                break
            if self._popsize_max is not None and total >= self._popsize_max:
                break
            if not hasattr(problem, "last_eval_interaction_count"):
                break
        self._population = batches[0] if len(batches) == 1 else SolutionBatch.cat(batches)

    def _step_non_distributed(self):
        if self._population is None:
            self._fill_and_eval_pop()
        fitnesses = self._population.access_evals()[:, self._obj_index]
        obj_sense = self.problem.senses[self._obj_index]
        with record_range("rank+grad"):
            gradients = self._distribution.compute_gradients(
                self._population.unsafe_values, fitnesses, objective_sense=obj_sense, ranking_method=self._ranking_method
            )
        with record_range("update"):
            self._update_distribution(gradients)
        self._fill_and_eval_pop()

    def _update_distribution(self, gradients: dict):
        controlled = (self._stdev_min is not None) or (self._stdev_max is not None) or (self._stdev_max_change is not None)
        old_sigma = self._distribution.parameters.get("sigma", None) if controlled else None

        learning_rates = {"sigma": self._stdev_learning_rate}
        optimizers = {}
        if self._optimizer is not None:
            optimizers["mu"] = self._optimizer
        else:
            learning_rates["mu"] = self._center_learning_rate

        updated = self._distribution.update_parameters(gradients, learning_rates=learning_rates, optimizers=optimizers)

        if controlled:
            updated = updated.modified_copy(
                sigma=modify_tensor(
                    old_sigma,
                    updated.parameters["sigma"],
                    lb=self._stdev_min,
                    ub=self._stdev_max,
                    max_change=self._stdev_max_change,
                )
            )
        self._distribution = updated

    # -- status getters ------------------------------------------------------

    @property
    def population(self) -> Optional[SolutionBatch]:
        return self._population

    @property
    def distribution(self) -> Distribution:
        return self._distribution

    @property
    def optimizer(self):
        return None if self._optimizer is None else self._optimizer.contained_optimizer

    @property
    def obj_index(self) -> int:
        return self._obj_index

    def _state_items(self) -> dict:
        items = {"distribution": dict(self._distribution.parameters)}
        opt = self._optimizer
        if opt is not None:
            opt_state = {}
            for attr in ("_velocity", "_m", "_v"):
                if getattr(opt, attr, None) is not None:
                    opt_state[attr] = getattr(opt, attr)
            if hasattr(opt, "_t"):
                opt_state["_t"] = opt._t
            items["optimizer"] = opt_state
        return items

    def _load_state_items(self, items: dict):
        dist_params = items.get("distribution", {})
        params = {
            k: (torch.as_tensor(v).to(self._distribution.device, self._distribution.dtype) if isinstance(v, torch.Tensor) else v)
            for k, v in dist_params.items()
        }
        self._distribution = type(self._distribution)(params, dtype=self._distribution.dtype, device=self._distribution.device)
        opt = self._optimizer
        for attr, v in items.get("optimizer", {}).items():
            if attr == "_t":
                opt._t = int(v)
            elif getattr(opt, attr, None) is not None:
                getattr(opt, attr).copy_(torch.as_tensor(v).to(getattr(opt, attr).device))

    def _get_mu(self) -> torch.Tensor:
        return self._distribution.parameters["mu"]

    def _get_sigma(self) -> torch.Tensor:
        return self._distribution.parameters["sigma"]

    def _get_mean_eval(self) -> Optional[float]:
        if self._population is None:
            me = self._mean_eval
            return float(me) if isinstance(me, torch.Tensor) else me
        return float(torch.mean(torch.Tensor.as_subclass(self._population.evals, torch.Tensor)[:, self._obj_index]))


class PGPE(GaussianSearchAlgorithm):
    """PGPE with ClipUp (default), antithetic sampling and centered
    ranking — the configuration of Toklu et al. 2020 (reference
    gaussian.py:503-744)."""

    DISTRIBUTION_TYPE = NotImplemented
    DISTRIBUTION_PARAMS = NotImplemented

    def __init__(
        self,
        problem: Problem,
        *,
        popsize: int,
        center_learning_rate: float,
        stdev_learning_rate: float,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        num_interactions: Optional[int] = None,
        popsize_max: Optional[int] = None,
        optimizer="clipup",
        optimizer_config: Optional[dict] = None,
        ranking_method: Optional[str] = "centered",
        center_init: Optional[RealOrVector] = None,
        stdev_min: Optional[RealOrVector] = None,
        stdev_max: Optional[RealOrVector] = None,
        stdev_max_change: Optional[RealOrVector] = 0.2,
        symmetric: bool = True,
        obj_index: Optional[int] = None,
        distributed: bool = False,
        popsize_weighted_grad_avg: Optional[bool] = None,
        grad_chunk_rows: Optional[int] = None,
    ):
        if symmetric:
            self.DISTRIBUTION_TYPE = SymmetricSeparableGaussian
            divide_by = "num_directions"
        else:
            self.DISTRIBUTION_TYPE = SeparableGaussian
            divide_by = "num_solutions"
        self.DISTRIBUTION_PARAMS = {"divide_mu_grad_by": divide_by, "divide_sigma_grad_by": divide_by}
        super().__init__(
            problem,
            popsize=popsize,
            center_learning_rate=center_learning_rate,
            stdev_learning_rate=stdev_learning_rate,
            stdev_init=stdev_init,
            radius_init=radius_init,
            popsize_max=popsize_max,
            num_interactions=num_interactions,
            optimizer=optimizer,
            optimizer_config=optimizer_config,
            ranking_method=ranking_method,
            center_init=center_init,
            stdev_min=stdev_min,
            stdev_max=stdev_max,
            stdev_max_change=stdev_max_change,
            obj_index=obj_index,
            distributed=distributed,
            popsize_weighted_grad_avg=popsize_weighted_grad_avg,
            ensure_even_popsize=symmetric,
            grad_chunk_rows=grad_chunk_rows,
        )


class SNES(GaussianSearchAlgorithm):
    """Separable NES (Schaul et al. 2011). Default popsize
    4 + ⌊3·ln(L)⌋ and stdev lr 0.2·(3+ln L)/√L (reference
    gaussian.py:746-984)."""

    DISTRIBUTION_TYPE = ExpSeparableGaussian
    DISTRIBUTION_PARAMS = None

    def __init__(
        self,
        problem: Problem,
        *,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        popsize: Optional[int] = None,
        center_learning_rate: Optional[float] = None,
        stdev_learning_rate: Optional[float] = None,
        scale_learning_rate: bool = True,
        num_interactions: Optional[int] = None,
        popsize_max: Optional[int] = None,
        optimizer=None,
        optimizer_config: Optional[dict] = None,
        ranking_method: Optional[str] = "nes",
        center_init: Optional[RealOrVector] = None,
        stdev_min: Optional[RealOrVector] = None,
        stdev_max: Optional[RealOrVector] = None,
        stdev_max_change: Optional[RealOrVector] = None,
        obj_index: Optional[int] = None,
        distributed: bool = False,
        popsize_weighted_grad_avg: Optional[bool] = None,
        grad_chunk_rows: Optional[int] = None,
    ):
        if popsize is None:
            popsize = int(4 + math.floor(3 * math.log(problem.solution_length)))
        if center_learning_rate is None:
            center_learning_rate = 1.0

        def default_stdev_lr():
            n = problem.solution_length
            return 0.2 * (3 + math.log(n)) / math.sqrt(n)

        if stdev_learning_rate is None:
            stdev_learning_rate = default_stdev_lr()
        else:
            stdev_learning_rate = float(stdev_learning_rate)
            if scale_learning_rate:
                stdev_learning_rate *= default_stdev_lr()

        super().__init__(
            problem,
            popsize=popsize,
            center_learning_rate=center_learning_rate,
            stdev_learning_rate=stdev_learning_rate,
            stdev_init=stdev_init,
            radius_init=radius_init,
            popsize_max=popsize_max,
            num_interactions=num_interactions,
            optimizer=optimizer,
            optimizer_config=optimizer_config,
            ranking_method=ranking_method,
            center_init=center_init,
            stdev_min=stdev_min,
            stdev_max=stdev_max,
            stdev_max_change=stdev_max_change,
            obj_index=obj_index,
            distributed=distributed,
            popsize_weighted_grad_avg=popsize_weighted_grad_avg,
            grad_chunk_rows=grad_chunk_rows,
        )


class CEM(GaussianSearchAlgorithm):
    """Cross-entropy method, elite-mean/std variant (Duan et al. 2016;
    reference gaussian.py:986-1181)."""

    DISTRIBUTION_TYPE = SeparableGaussian
    DISTRIBUTION_PARAMS = NotImplemented

    def __init__(
        self,
        problem: Problem,
        *,
        popsize: int,
        parenthood_ratio: float,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        num_interactions: Optional[int] = None,
        popsize_max: Optional[int] = None,
        center_init: Optional[RealOrVector] = None,
        stdev_min: Optional[RealOrVector] = None,
        stdev_max: Optional[RealOrVector] = None,
        stdev_max_change: Optional[RealOrVector] = None,
        obj_index: Optional[int] = None,
        distributed: bool = False,
        popsize_weighted_grad_avg: Optional[bool] = None,
        grad_chunk_rows: Optional[int] = None,
    ):
        self.DISTRIBUTION_PARAMS = {"parenthood_ratio": float(parenthood_ratio)}
        super().__init__(
            problem,
            popsize=popsize,
            center_learning_rate=1.0,
            stdev_learning_rate=1.0,
            stdev_init=stdev_init,
            radius_init=radius_init,
            popsize_max=popsize_max,
            num_interactions=num_interactions,
            optimizer=None,
            optimizer_config=None,
            ranking_method=None,
            center_init=center_init,
            stdev_min=stdev_min,
            stdev_max=stdev_max,
            stdev_max_change=stdev_max_change,
            obj_index=obj_index,
            distributed=distributed,
            popsize_weighted_grad_avg=popsize_weighted_grad_avg,
            grad_chunk_rows=grad_chunk_rows,
        )


class XNES(GaussianSearchAlgorithm):
    """Exponential NES with full covariance (Glasmachers et al. 2010).
    Default stdev lr 0.6·(3+ln L)/(L·√L) (reference gaussian.py:1183-1404)."""

    DISTRIBUTION_TYPE = ExpGaussian
    DISTRIBUTION_PARAMS = None

    def __init__(
        self,
        problem: Problem,
        *,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        popsize: Optional[int] = None,
        center_learning_rate: Optional[float] = None,
        stdev_learning_rate: Optional[float] = None,
        scale_learning_rate: bool = True,
        num_interactions: Optional[int] = None,
        popsize_max: Optional[int] = None,
        optimizer=None,
        optimizer_config: Optional[dict] = None,
        ranking_method: Optional[str] = "nes",
        center_init: Optional[RealOrVector] = None,
        obj_index: Optional[int] = None,
        distributed: bool = False,
        popsize_weighted_grad_avg: Optional[bool] = None,
        grad_chunk_rows: Optional[int] = None,
    ):
        if popsize is None:
            popsize = int(4 + math.floor(3 * math.log(problem.solution_length)))
        if center_learning_rate is None:
            center_learning_rate = 1.0

        def default_stdev_lr():
            n = problem.solution_length
            return 0.6 * (3 + math.log(n)) / (n * math.sqrt(n))

        if stdev_learning_rate is None:
            stdev_learning_rate = default_stdev_lr()
        else:
            stdev_learning_rate = float(stdev_learning_rate)
            if scale_learning_rate:
                stdev_learning_rate *= default_stdev_lr()

        super().__init__(
            problem,
            popsize=popsize,
            center_learning_rate=center_learning_rate,
            stdev_learning_rate=stdev_learning_rate,
            stdev_init=stdev_init,
            radius_init=radius_init,
            popsize_max=popsize_max,
            num_interactions=num_interactions,
            optimizer=optimizer,
            optimizer_config=optimizer_config,
            ranking_method=ranking_method,
            center_init=center_init,
            stdev_min=None,
            stdev_max=None,
            stdev_max_change=None,
            obj_index=obj_index,
            distributed=distributed,
            popsize_weighted_grad_avg=popsize_weighted_grad_avg,
            grad_chunk_rows=grad_chunk_rows,
        )
