"""hipGraph-captured generation loops.

Small-population searchers (the SNES-on-Rastrigin regime of BASELINE.md
row 1) are launch-latency-bound on GPU: a generation is ~10 small kernels
whose combined launch + Python overhead dwarfs their device time. This
module captures ONE whole generation — philox sampling, fitness
evaluation, ranking, the fused gradient reduction and the in-place
distribution update — into a hipGraph (`torch.cuda.CUDAGraph` maps onto
hipGraph on ROCm) and replays it per generation: one graph launch instead
of ~10 kernel launches and zero Python in the loop.

(Green-field MI355X feature: the reference has no equivalent — its
nearest analogue is the torch.compile/vmap usage in the functional API,
/root/reference/src/evotorch/algorithms/functional/__init__.py:15-50.)

Graph-safe RNG: sampling uses `sample_gaussian_graphsafe`, which reads its
philox seed from a device buffer and advances it ON DEVICE (splitmix64
bump kernel) — a by-value seed would be frozen into the capture and every
replay would resample the same population.

Supported: non-distributed GaussianSearchAlgorithm searchers (PGPE / SNES
/ CEM — the elite top-k sort is shape-static and captures fine) on a ROCm
device, with a capture-safe (pure-tensor) fitness function, fixed
popsize, and a plain center learning rate, ClipUp, or Adam (the Adam step
count lives in a device buffer advanced on device, so each replay applies
the right bias correction). Do not interleave plain `searcher.step()`
calls after capture — the graph is bound to the distribution's parameter
buffers; re-create the GraphedSearch if you need to.
"""

from typing import Optional

import torch

from ..distributions import ExpSeparableGaussian, SeparableGaussian, SymmetricSeparableGaussian
from ..optimizers import Adam, ClipUp
from ..utils import ranking as _ranking
from ..utils.misc import modify_tensor
from .gaussian import GaussianSearchAlgorithm

__all__ = ["GraphedSearch"]


class GraphedSearch:
    def __init__(self, searcher: GaussianSearchAlgorithm, *, warmup: int = 3, generations_per_capture: int = 1):
        if not isinstance(searcher, GaussianSearchAlgorithm):
            raise TypeError("GraphedSearch supports GaussianSearchAlgorithm searchers")
        if searcher._distributed:
            raise ValueError("GraphedSearch requires a non-distributed searcher")
        if searcher._num_interactions is not None:
            raise ValueError("Adaptive popsize (num_interactions) cannot be graph-captured")
        dist = searcher._distribution
        if not isinstance(dist, (SeparableGaussian, SymmetricSeparableGaussian, ExpSeparableGaussian)):
            raise TypeError(f"Unsupported distribution for graph capture: {type(dist).__name__}")
        opt = searcher._optimizer
        if opt is not None and not isinstance(opt, (ClipUp, Adam)):
            raise TypeError("Graph capture supports no optimizer, ClipUp, or Adam")
        problem = searcher.problem
        if problem.device.type != "cuda":
            raise ValueError("GraphedSearch needs a ROCm device problem")
        comm = getattr(problem, "_comm", None)
        if comm is not None and comm.world_size > 1:
            raise ValueError("GraphedSearch cannot capture collectives; detach the Comm (world 1) first")
        # problems with host-derived per-eval randomness switch it to a
        # device seed chain so episode seeds vary across replays
        enable_graph_mode = getattr(problem, "enable_graph_mode", None)
        if enable_graph_mode is not None:
            enable_graph_mode()
        self._searcher = searcher
        self._problem = problem
        self._dist = dist
        self._opt = opt
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self._warmup = int(warmup)
        self._gens_per_capture = max(1, int(generations_per_capture))

        from .. import ops
        from ..core import SolutionBatch
        from ..ops.dispatch import _seed_from_generator

        self._C = ops.hip_required()
        seed0 = _seed_from_generator(problem.generator, problem.device)
        self._seed_buf = torch.tensor([seed0], dtype=torch.int64, device=problem.device)
        if searcher._population is None:
            searcher._population = SolutionBatch(problem, popsize=searcher._popsize, device=dist.device, empty=True)
        self._pop = searcher._population
        # scratch for the stdev-control clamp (needs the pre-update sigma)
        self._old_sigma = dist.parameters["sigma"].clone()
        self._mean_eval_buf = torch.zeros((), dtype=problem.eval_dtype, device=problem.device)
        self._adam_t_buf = None
        if isinstance(opt, Adam):
            # graph-safe device-side step counter (see adam_step_graphsafe)
            self._adam_t_buf = torch.tensor([int(opt._t)], dtype=torch.int64, device=problem.device)
            self._adam_step_out = torch.zeros_like(dist.parameters["mu"])

    # -- one in-place generation (everything stays in fixed buffers) --------

    def _step_body(self):
        searcher = self._searcher
        dist = self._dist
        problem = self._problem
        mu = dist.parameters["mu"]
        sigma = dist.parameters["sigma"]
        pop = self._pop
        values = pop.unsafe_values

        self._C.sample_gaussian_graphsafe(values, mu, sigma, dist._symmetric, self._seed_buf)
        problem._evaluate_batch(pop)
        merge_stats = getattr(problem, "_merge_pending_stats", None)
        if merge_stats is not None:
            merge_stats()  # world-1: in-place obs-norm accumulation, capture-safe
        fitnesses = pop.unsafe_evals[:, searcher._obj_index]
        self._mean_eval_buf.copy_(fitnesses.mean())
        sense = problem.senses[searcher._obj_index]
        method = searcher._ranking_method or "raw"
        weights = _ranking.rank(fitnesses, method, higher_is_better=(sense == "max")).to(values.dtype)
        grads = dist._compute_gradients(values, weights, ranking_used=method)

        # in-place center update
        if isinstance(self._opt, Adam):
            self._C.adam_step_graphsafe(
                self._adam_step_out,
                grads["mu"].to(self._adam_step_out.dtype),
                self._opt._m,
                self._opt._v,
                self._adam_t_buf,
                self._opt._stepsize,
                self._opt._beta1,
                self._opt._beta2,
                self._opt._epsilon,
            )
            mu.add_(self._adam_step_out)
        elif self._opt is not None:
            from .. import ops

            ops.clipup_step_(
                self._opt._velocity,
                grads["mu"].to(self._opt._dtype),
                step_size=self._opt._stepsize,
                max_speed=self._opt._max_speed,
                momentum=self._opt._momentum,
            )
            mu.add_(self._opt._velocity)
        else:
            mu.add_(grads["mu"], alpha=searcher._center_learning_rate)

        # in-place stdev update
        lr = searcher._stdev_learning_rate
        if isinstance(dist, ExpSeparableGaussian):
            sigma.mul_(torch.exp(0.5 * lr * grads["sigma"]))
        else:
            self._old_sigma.copy_(sigma)
            sigma.add_(grads["sigma"], alpha=lr)
            if searcher._stdev_min is not None or searcher._stdev_max is not None or searcher._stdev_max_change is not None:
                clamped = modify_tensor(
                    self._old_sigma, sigma, lb=searcher._stdev_min, ub=searcher._stdev_max, max_change=searcher._stdev_max_change
                )
                sigma.copy_(clamped)

    # -- capture & replay ----------------------------------------------------

    def capture(self):
        from ..neuroevolution.vecenv import _capture_stream

        # persistent per-device capture stream: keeps the BLAS workspace
        # OUT of the graph's private pool (see _capture_stream docstring)
        side = _capture_stream(self._problem.device)
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(self._warmup):
                self._step_body()
                self._searcher._steps_count += 1
        torch.cuda.current_stream().wait_stream(side)
        # no cyclic GC during capture: freeing stale CUDA garbage on
        # non-captured streams mid-capture aborts the HIP runtime
        import gc

        gc.collect()
        gc.disable()
        try:
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph, stream=side):
                for _ in range(self._gens_per_capture):
                    self._step_body()
        finally:
            gc.enable()
        self._searcher._steps_count += self._gens_per_capture
        return self

    def run(self, num_generations: int):
        """Replay `num_generations` captured generations (capturing first if
        needed), then refresh the searcher's status."""
        if self._graph is None:
            self.capture()
        g = self._graph
        replays = (int(num_generations) + self._gens_per_capture - 1) // self._gens_per_capture
        for _ in range(replays):
            g.replay()
        torch.cuda.synchronize()
        searcher = self._searcher
        searcher._steps_count += replays * self._gens_per_capture
        searcher.clear_status()
        searcher.update_status({"iter": searcher._steps_count, "mean_eval": float(self._mean_eval_buf)})
        return self

    @property
    def mean_eval(self) -> float:
        return float(self._mean_eval_buf)

    def state_dict(self) -> dict:
        """Checkpoint of the wrapped searcher's current (graph-updated)
        state — the in-place buffers ARE the distribution parameters, so
        the inner searcher's state_dict sees the latest values."""
        return self._searcher.state_dict()

    def load_state_dict(self, state: dict):
        self._searcher.load_state_dict(state)
        if self._graph is not None:
            raise RuntimeError("load_state_dict after capture() is not supported: the captured graph holds the old parameter buffers; re-create the GraphedSearch and capture again")
        return self
