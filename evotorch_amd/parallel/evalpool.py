"""Process-pool evaluation for CPU-bound problems: the reference's Ray
actor system (reference core.py:115-270 `EvaluationActor`, :1977
`_parallelize`, :2595-2600 actor-pool dispatch, :2239-2340 actor state
sync) rebuilt on the standard library.

Where the reference ships a pickled clone of the Problem to each Ray
actor and round-trips batch pieces through the object store, this module
spawns plain `multiprocessing` workers, hands each ONE cloudpickle
payload of the problem at startup (lambdas and closures pickle fine via
cloudpickle), and scatters contiguous row pieces of the batch per
evaluation. Worker-side auxiliary state — observation-normalization
statistics and interaction counters — returns with the evals and is
merged on the main process, mirroring the reference's `_sync_after`
round-trip.

Positioning (docs/parallelism.md): the SPMD path (torchrun + RCCL/gloo,
`problem.use_comm`) remains the way to scale GPU work and multi-node
runs; `num_actors` covers the reference's "one script, many CPU env
workers" use case (GymNE-style rollouts, per-solution python fitness)
without a launcher. When BOTH a Comm and a pool exist, the Comm wins —
each rank then evaluates its shard serially.
"""

import os
import pickle
from typing import Optional

import torch

__all__ = ["EvalPool"]

_WORKER_PROBLEM = None


def _worker_init(payload: bytes, base_seed: int, counter) -> None:
    """Runs once inside each spawned worker: unpickle the problem clone
    and seed every RNG domain per-worker (reference core.py:135-141
    seeds py/np/torch/problem per actor)."""
    global _WORKER_PROBLEM
    import random

    import cloudpickle
    import numpy as np

    with counter.get_lock():
        index = counter.value
        counter.value += 1
    seed = (int(base_seed) + 1000 * (index + 1)) & 0x7FFFFFFF
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    problem = cloudpickle.loads(payload)
    if getattr(problem, "_seed", None) is not None:
        problem._seed = seed
        problem._generator = torch.Generator(device=problem._device)
        problem._generator.manual_seed(seed)
    hook = getattr(problem, "remote_hook", None)
    if hook is not None:
        hook(problem)  # reference core.py:142 — actor-side init hook
    _WORKER_PROBLEM = problem


def _worker_ping():
    return _WORKER_PROBLEM is not None


def _worker_eval(args):
    """Evaluate one contiguous piece of the population; return (piece
    index, evals, aux sync data)."""
    piece_index, values, sync_stats = args
    problem = _WORKER_PROBLEM
    from ..core import SolutionBatch

    if sync_stats is not None:
        # main -> worker: normalize with the stats as of this generation's
        # start (the reference's _sync_before push)
        set_stats = getattr(problem, "set_observation_stats", None)
        if set_stats is not None:
            set_stats(sync_stats)
    pop_stats = getattr(problem, "pop_observation_stats", None)
    if pop_stats is not None:
        pop_stats()  # discard deltas left from a previous piece
    if hasattr(problem, "last_eval_interaction_count"):
        problem.last_eval_interaction_count = 0
    batch = SolutionBatch(problem, popsize=len(values), empty=True)
    bv = batch.access_values()
    bv[:] = values
    problem._evaluate_batch(batch)
    aux = {}
    pop_stats = getattr(problem, "pop_observation_stats", None)
    if pop_stats is not None:
        rn = pop_stats()
        if rn is not None:
            aux["obs_stats"] = tuple(t.cpu() for t in rn.stats_triple())
    interactions = getattr(problem, "last_eval_interaction_count", None)
    if interactions is not None:
        aux["interactions"] = int(interactions)
    return piece_index, batch.access_evals().cpu(), aux


class EvalPool:
    """A persistent pool of evaluation workers bound to one Problem.

    Created lazily by `Problem.evaluate` when the problem was constructed
    with `num_actors`; closed by `Problem.kill_actors()` (the reference's
    method name) or interpreter exit."""

    def __init__(self, problem, num_actors: int, *, num_subbatches: Optional[int] = None, subbatch_size: Optional[int] = None):
        import multiprocessing as mp

        import cloudpickle

        self._num_actors = max(1, int(num_actors))
        self._num_subbatches = num_subbatches
        self._subbatch_size = subbatch_size
        payload = cloudpickle.dumps(problem)
        # quick sanity: the payload must unpickle locally (catches problems
        # holding genuinely unpicklable state before workers die opaquely)
        cloudpickle.loads(payload)
        base_seed = problem._seed if getattr(problem, "_seed", None) is not None else torch.initial_seed()
        ctx = mp.get_context("spawn")
        counter = ctx.Value("i", 0)
        self._pool = ctx.Pool(self._num_actors, initializer=_worker_init, initargs=(payload, int(base_seed) & 0x7FFFFFFF, counter))
        # readiness probe: a worker whose spawn/init keeps failing makes the
        # pool respawn forever and evaluation would HANG; surface it instead.
        # (The classic cause: the launching script is not importable — run
        # pool-using code under `if __name__ == "__main__":`, like any
        # multiprocessing or DataLoader-workers program.)
        timeout_s = float(os.environ.get("EVOTORCH_AMD_POOL_INIT_TIMEOUT", "120"))
        try:
            self._pool.apply_async(_worker_ping).get(timeout=timeout_s)
        except mp.TimeoutError:
            self.close()
            raise RuntimeError(
                f"evaluation workers failed to come up within {timeout_s:.0f}s "
                "(num_actors pool). Most common cause: the launching script is not "
                "importable by spawned processes — guard the entry point with "
                "`if __name__ == '__main__':` (the standard multiprocessing requirement)."
            ) from None

    def _piece_size(self, n: int) -> int:
        if self._subbatch_size is not None:
            return max(1, int(self._subbatch_size))
        if self._num_subbatches is not None:
            return max(1, (n + int(self._num_subbatches) - 1) // int(self._num_subbatches))
        return max(1, (n + self._num_actors - 1) // self._num_actors)

    def evaluate_into(self, problem, batch) -> None:
        n = len(batch)
        size = self._piece_size(n)
        values = batch.access_values(keep_evals=True)
        sync_stats = None
        obs_norm = getattr(problem, "obs_norm", None)
        if obs_norm is not None and getattr(problem, "_obs_norm_enabled", True) and obs_norm.has_data:
            sync_stats = tuple(t.cpu() for t in obs_norm.stats_triple())
        tasks = []
        bounds = []
        for start in range(0, n, size):
            stop = min(start + size, n)
            piece = values[start:stop]
            piece = piece.cpu() if isinstance(piece, torch.Tensor) else piece
            tasks.append((len(bounds), piece, sync_stats))
            bounds.append((start, stop))
        total_interactions = 0
        merged_any_interactions = False
        for piece_index, evals, aux in self._pool.imap_unordered(_worker_eval, tasks):
            start, stop = bounds[piece_index]
            batch._evals[start:stop] = evals.to(batch._evals.device, batch._evals.dtype)
            if "obs_stats" in aux:
                update = getattr(problem, "update_observation_stats", None)
                obs_norm = getattr(problem, "obs_norm", None)
                if update is not None:
                    update(aux["obs_stats"])
                elif obs_norm is not None:
                    obs_norm.update(aux["obs_stats"])
            if "interactions" in aux:
                merged_any_interactions = True
                total_interactions += aux["interactions"]
        if merged_any_interactions:
            if hasattr(problem, "last_eval_interaction_count"):
                problem.last_eval_interaction_count = total_interactions
            if hasattr(problem, "_total_interactions"):
                problem._total_interactions += total_interactions
            if hasattr(problem, "_episode_count"):
                problem._episode_count += n * int(getattr(problem, "_num_episodes", 1) or 1)

    def close(self) -> None:
        pool, self._pool = self._pool, None
        if pool is not None:
            pool.terminate()
            pool.join()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
