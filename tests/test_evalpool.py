"""Process-pool evaluation (`num_actors`): the reference's Ray actor
system rebuilt on multiprocessing + cloudpickle (parallel/evalpool.py)."""

import pytest
import torch

from evotorch_amd import Problem


def test_pool_matches_serial_and_supports_lambdas():
    def make(num_actors=None):
        return Problem("min", lambda x: float((x - 0.25).pow(2).sum()),
                       solution_length=6, initial_bounds=(-1, 1), seed=11, num_actors=num_actors)

    serial = make()
    pooled = make(num_actors=2)
    vals = torch.randn(torch.Generator().manual_seed(3) and 9, 6)
    torch.manual_seed(3)
    vals = torch.randn(9, 6)

    def evaluate(prob):
        b = prob.generate_batch(9)
        b.access_values()[:] = vals
        prob.evaluate(b)
        return torch.Tensor.as_subclass(b.evals[:, 0], torch.Tensor).clone()

    try:
        e_serial = evaluate(serial)
        e_pooled = evaluate(pooled)
        torch.testing.assert_close(e_pooled, e_serial)
        assert pooled._eval_pool is not None
        # subsequent evaluations reuse the same pool
        pool = pooled._eval_pool
        evaluate(pooled)
        assert pooled._eval_pool is pool
    finally:
        pooled.kill_actors()
    assert pooled._eval_pool is None


def test_pool_subbatch_size_and_vectorized_fitness():
    from evotorch_amd.decorators import vectorized

    @vectorized
    def f(x):
        return (x**2).sum(-1)

    prob = Problem("min", f, solution_length=4, initial_bounds=(-1, 1), seed=7,
                   num_actors=2, subbatch_size=3)
    try:
        b = prob.generate_batch(10)
        prob.evaluate(b)
        vals = torch.Tensor.as_subclass(b.access_values(keep_evals=True), torch.Tensor)
        expected = (vals**2).sum(-1)
        torch.testing.assert_close(torch.Tensor.as_subclass(b.evals[:, 0], torch.Tensor), expected)
    finally:
        prob.kill_actors()


def test_pool_gymne_merges_counters_and_obs_stats():
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import GymNE

    prob = GymNE("CartPole-v1", "Linear(obs_length, act_length)", num_episodes=1,
                 episode_length=25, seed=5, num_actors=2, observation_normalization=True)
    try:
        searcher = PGPE(prob, popsize=8, center_learning_rate=0.2, stdev_learning_rate=0.1, radius_init=0.5)
        searcher.run(2)
        assert searcher.status["iter"] == 2
        assert prob._total_interactions > 0
        assert prob.obs_norm.has_data  # worker stats merged back
    finally:
        prob.kill_actors()


def test_pool_object_dtype_problem():
    """Variable-length (object dtype) problems evaluate through the pool:
    ObjectArray pieces ship via cloudpickle, per-solution `_evaluate` runs
    in the workers."""
    from evotorch_amd import Problem

    class VarLenProblem(Problem):
        def __init__(self, **kw):
            super().__init__("min", dtype=object, seed=3, **kw)

        def _fill(self, values):
            for i in range(len(values)):
                values[i] = [float(i), float(i + 1)]

        def _evaluate(self, solution):
            solution.set_evaluation(float(sum(solution.values)))

    prob = VarLenProblem(num_actors=2)
    try:
        b = prob.generate_batch(6)
        prob.evaluate(b)
        got = [float(x) for x in b.evals[:, 0]]
        assert got == [1.0, 3.0, 5.0, 7.0, 9.0, 11.0]
    finally:
        prob.kill_actors()


def test_remote_hook_runs_on_workers():
    """remote_hook executes on each worker's problem clone (not on the
    main process) — observable through the fitness it configures."""
    from evotorch_amd import Problem

    class OffsetProblem(Problem):
        def __init__(self, **kw):
            super().__init__("min", solution_length=3, initial_bounds=(-1, 1), seed=2, **kw)
            self.offset = 0.0

        def _evaluate(self, solution):
            solution.set_evaluation(float(solution.values.sum()) + self.offset)

    def set_offset(problem):
        problem.offset = 100.0

    prob = OffsetProblem(num_actors=2)
    prob.remote_hook.append(set_offset)
    try:
        b = prob.generate_batch(4)
        vals = torch.Tensor.as_subclass(b.access_values(keep_evals=True), torch.Tensor).clone()
        prob.evaluate(b)
        expected = vals.sum(-1) + 100.0
        torch.testing.assert_close(torch.Tensor.as_subclass(b.evals[:, 0], torch.Tensor), expected)
        assert prob.offset == 0.0  # main-process problem untouched
    finally:
        prob.kill_actors()


def test_pool_parallelizes_distributed_gradient_evaluation():
    """PGPE(distributed=True) without a Comm routes its population
    evaluation through Problem.evaluate — so the num_actors pool
    parallelizes the expensive part of the gradient step too (the
    reference's actor-parallel `sample_and_compute_gradients` use case)."""
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import GymNE

    prob = GymNE("CartPole-v1", "Linear(obs_length, act_length)", num_episodes=1,
                 episode_length=20, seed=6, num_actors=2)
    try:
        s = PGPE(prob, popsize=8, center_learning_rate=0.2, stdev_learning_rate=0.1,
                 radius_init=0.5, distributed=True)
        s.run(2)
        assert s.status["iter"] == 2
        assert prob._eval_pool is not None  # the pool actually engaged
        assert prob._total_interactions > 0
    finally:
        prob.kill_actors()


def test_pool_survives_worker_exception():
    """A fitness error inside a worker propagates to the caller, and the
    pool stays usable for the next evaluation."""
    def touchy(x):
        if float(x[0]) > 1e8:
            raise ValueError("boom")
        return float((x**2).sum())

    prob = Problem("min", touchy, solution_length=3, initial_bounds=(-1, 1), seed=4, num_actors=2)
    try:
        b = prob.generate_batch(6)
        prob.evaluate(b)  # normal values: fine
        bad = prob.generate_batch(4)
        bad.access_values()[0, 0] = 1e9
        with pytest.raises(Exception):
            prob.evaluate(bad)
        ok = prob.generate_batch(6)
        prob.evaluate(ok)  # pool still healthy
        assert bool(torch.isfinite(torch.Tensor.as_subclass(ok.evals[:, 0], torch.Tensor)).all())
    finally:
        prob.kill_actors()
