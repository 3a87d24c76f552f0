"""Core container + Problem tests (mirrors reference tests/test_core.py:
slicing aliasing via storage_ptr, cloning, hooks, multi-objective, object
dtype)."""

import math

import numpy as np
import pytest
import torch

from evotorch_amd import Problem, Solution, SolutionBatch
from evotorch_amd.decorators import vectorized
from evotorch_amd.utils import storage_ptr


@vectorized
def sphere(x: torch.Tensor) -> torch.Tensor:
    return (x**2).sum(-1)


def make_problem(**kwargs):
    defaults = dict(objective_sense="min", objective_func=sphere, solution_length=8, initial_bounds=(-1.0, 1.0), seed=0)
    defaults.update(kwargs)
    return Problem(**defaults)


def test_generate_batch_and_evaluate():
    prob = make_problem()
    batch = prob.generate_batch(10)
    assert len(batch) == 10
    assert batch.values_shape == (10, 8)
    assert not batch.evals_are_ready
    prob.evaluate(batch)
    assert batch.evals_are_ready
    expected = (torch.Tensor.as_subclass(batch.values, torch.Tensor) ** 2).sum(-1)
    assert torch.allclose(torch.Tensor.as_subclass(batch.evals, torch.Tensor)[:, 0], expected)


def test_batch_slicing_shares_memory():
    prob = make_problem()
    batch = prob.generate_batch(10)
    sub = batch[2:5]
    assert isinstance(sub, SolutionBatch)
    assert len(sub) == 3
    assert storage_ptr(sub.unsafe_values) == storage_ptr(batch.unsafe_values)
    sub.access_values()[0, 0] = 123.0
    assert float(batch.unsafe_values[2, 0]) == 123.0


def test_take_copies():
    prob = make_problem()
    batch = prob.generate_batch(10)
    prob.evaluate(batch)
    taken = batch.take([1, 3, 5])
    assert len(taken) == 3
    assert storage_ptr(taken.unsafe_values) != storage_ptr(batch.unsafe_values)
    assert torch.allclose(taken.unsafe_values[0], batch.unsafe_values[1])
    assert torch.allclose(taken.unsafe_evals[2], batch.unsafe_evals[5])


def test_access_values_resets_evals():
    prob = make_problem()
    batch = prob.generate_batch(4)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    batch.access_values()
    assert not batch.evals_are_ready


def test_solution_view():
    prob = make_problem()
    batch = prob.generate_batch(5)
    prob.evaluate(batch)
    sln = batch[2]
    assert isinstance(sln, Solution)
    assert sln.is_evaluated
    assert float(sln.evaluation) == pytest.approx(float(batch.unsafe_evals[2, 0]))
    sln.set_values(torch.zeros(8))
    assert not sln.is_evaluated
    assert torch.allclose(batch.unsafe_values[2], torch.zeros(8))


def test_argsort_and_take_best():
    prob = make_problem()
    batch = prob.generate_batch(20)
    prob.evaluate(batch)
    order = batch.argsort()
    evals = batch.unsafe_evals[:, 0]
    assert float(evals[order[0]]) == float(evals.min())  # "min" sense: best first
    best2 = batch.take_best(2)
    assert len(best2) == 2
    assert float(best2.unsafe_evals[0, 0]) <= float(best2.unsafe_evals[1, 0])
    single = batch.take_best()
    assert isinstance(single, Solution)


def test_split_and_merge():
    prob = make_problem()
    batch = prob.generate_batch(10)
    pieces = batch.split(3)
    assert len(pieces) == 3
    assert [len(p) for p in pieces] == [4, 3, 3]
    assert pieces.indices_of(1) == (4, 7)
    # piece shares memory
    pieces[0].access_values()[0, 0] = 7.0
    assert float(batch.unsafe_values[0, 0]) == 7.0
    merged = SolutionBatch.cat([pieces[i] for i in range(3)])
    assert len(merged) == 10
    assert torch.allclose(merged.unsafe_values, batch.unsafe_values)


def test_multiobjective_pareto():
    prob = Problem(["min", "min"], solution_length=2, initial_bounds=(0.0, 1.0), vectorized=True,
                   objective_func=lambda x: torch.stack([x[:, 0], x[:, 1]], dim=-1))
    batch = prob.generate_batch(4, empty=True)
    batch.access_values().copy_(torch.tensor([[0.0, 1.0], [1.0, 0.0], [0.5, 0.5], [1.0, 1.0]]))
    prob.evaluate(batch)
    ranks, crowd = batch.compute_pareto_ranks()
    # first three are mutually non-dominating; last is dominated
    assert ranks.tolist()[:3] == [0, 0, 0]
    assert ranks.tolist()[3] >= 1
    fronts = batch.arg_pareto_sort()
    assert set(fronts[0].tolist()) == {0, 1, 2}
    order = batch.argsort()
    assert order.tolist()[-1] == 3


def test_best_worst_tracking():
    prob = make_problem(store_solution_stats=True)
    for _ in range(3):
        batch = prob.generate_batch(16)
        prob.evaluate(batch)
    assert prob.best is not None
    assert prob.worst is not None
    assert float(prob.best.evaluation) <= float(prob.worst.evaluation)


def test_hooks_fire():
    prob = make_problem()
    calls = []
    prob.before_eval_hook.append(lambda b: calls.append(("before", len(b))))
    prob.after_eval_hook.append(lambda b: {"note": len(b)})
    batch = prob.generate_batch(6)
    prob.evaluate(batch)
    assert calls == [("before", 6)]
    assert prob.status["note"] == 6


def test_problem_pickling():
    import pickle

    prob = make_problem()
    batch = prob.generate_batch(4)
    prob.evaluate(batch)
    blob = pickle.dumps(prob)
    prob2 = pickle.loads(blob)
    assert prob2.solution_length == 8
    b2 = prob2.generate_batch(4)
    prob2.evaluate(b2)


def test_object_dtype_problem():
    class PermProblem(Problem):
        def __init__(self):
            super().__init__(objective_sense="max", dtype=object, eval_dtype=torch.float32)

        def _fill(self, values):
            for i in range(len(values)):
                values[i] = list(np.random.permutation(5))

        def _evaluate(self, solution):
            solution.set_evaluation(float(sum(solution.values)))

    prob = PermProblem()
    batch = prob.generate_batch(3)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    assert float(batch.unsafe_evals[0, 0]) == pytest.approx(10.0)


def test_eval_data():
    @vectorized
    def f(x):
        return (x**2).sum(-1), x[:, :2]

    prob = Problem("min", f, solution_length=4, initial_bounds=(-1, 1), eval_data_length=2)
    batch = prob.generate_batch(5)
    prob.evaluate(batch)
    assert batch.eval_shape == (5, 3)
    assert torch.allclose(batch.unsafe_evals[:, 1:], batch.unsafe_values[:, :2])


def test_solution_batch_to_device_noop_on_same():
    prob = make_problem()
    batch = prob.generate_batch(4)
    same = batch.to("cpu")
    assert same is batch


def test_problem_bound_evaluator():
    prob = make_problem()
    ev = prob.make_callable_evaluator()
    x = torch.randn(6, 8)
    out = ev(x)
    assert torch.allclose(out, (x**2).sum(-1), atol=1e-5)
    x3 = torch.randn(2, 5, 8)
    out3 = ev(x3)
    assert out3.shape == (2, 5)
    assert torch.allclose(out3, (x3**2).sum(-1), atol=1e-5)


def test_crowding_distances_match_bruteforce():
    from evotorch_amd.core import _crowding_distances

    torch.manual_seed(0)
    for trial in range(3):
        n, m = 60, 2 + trial % 2
        utils = torch.randn(n, m)
        # synthetic front assignment with varied sizes
        ranks = torch.randint(0, 7, (n,))

        # brute force reference (the per-front formulation)
        crowd_ref = torch.zeros(n)
        for front in torch.unique(ranks):
            idx = torch.nonzero(ranks == front, as_tuple=True)[0]
            if len(idx) <= 2:
                crowd_ref[idx] = float("inf")
                continue
            sub = utils[idx]
            for j in range(m):
                order = sub[:, j].argsort()
                sv = sub[order, j]
                span = sv[-1] - sv[0]
                if float(span) == 0.0:
                    continue
                contrib = torch.zeros(len(idx))
                contrib[order[0]] = float("inf")
                contrib[order[-1]] = float("inf")
                contrib[order[1:-1]] = (sv[2:] - sv[:-2]) / span
                crowd_ref[idx] += contrib

        crowd = _crowding_distances(utils, ranks)
        finite = torch.isfinite(crowd_ref)
        assert torch.equal(torch.isfinite(crowd), finite)
        assert torch.allclose(crowd[finite], crowd_ref[finite], atol=1e-5)


def test_solution_batch_pieces_alias_parent():
    """split() pieces are views: evaluating pieces evaluates the parent
    (reference core.py:4603 SolutionBatchPieces)."""
    prob = make_problem(seed=91)
    batch = prob.generate_batch(10)
    pieces = batch.split(3)
    assert len(pieces) == 3
    assert sum(len(p) for p in pieces) == 10
    for piece in pieces:
        prob.evaluate(piece)
    assert batch.evals_are_ready
    # max_size variant
    pieces2 = batch.split(max_size=4)
    assert all(len(p) <= 4 for p in pieces2)
    assert sum(len(p) for p in pieces2) == 10
    # indexing
    assert len(pieces2[0]) == len(pieces2[0])


def test_solution_comparison_methods():
    prob = make_problem(seed=92)
    batch = prob.generate_batch(4)
    prob.evaluate(batch)
    order = batch.argsort()
    best, worst = batch[int(order[0])], batch[int(order[-1])]
    assert prob.is_better(best, worst)
    assert prob.is_worse(worst, best)
    assert prob.compare_solutions(best, worst) > 0
    assert prob.normalize_obj_index(None) == 0
    assert prob.normalize_obj_index(-1) == 0
    assert prob.is_on_cpu


def test_batch_utility_and_utils():
    """SolutionBatch.utility (single-objective) and .utils (per-objective
    stack) — reference core.py:4207."""
    prob = make_problem(seed=93)
    batch = prob.generate_batch(8)
    prob.evaluate(batch)
    u = batch.utility(ranking_method="centered")
    assert u.shape == (8,)
    # best solution gets the highest utility
    best = int(batch.argbest())
    assert int(u.argmax()) == best

    @vectorized
    def two(x):
        return torch.stack([x.sum(-1), -x.sum(-1)], dim=-1)

    mo = Problem(["min", "max"], two, solution_length=3, initial_bounds=(-1, 1), seed=94)
    b2 = mo.generate_batch(6)
    mo.evaluate(b2)
    us = b2.utils(ranking_method="centered")
    assert us.shape == (6, 2)


def test_ray_era_kwargs_accepted_with_warning():
    """Reference code passing num_actors etc. migrates without a TypeError
    (the knobs are ignored; parallelism comes from the torchrun topology)."""
    import warnings

    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        prob = make_problem_with_kwargs = Problem(
            "min", sphere, solution_length=4, initial_bounds=(-1, 1),
            num_actors="max", num_gpus_per_actor=0.5, subbatch_size=50)
    assert any("Ray-era" in str(x.message) for x in w)
    batch = prob.generate_batch(4)
    prob.evaluate(batch)
    assert batch.evals_are_ready


def test_eval_data_roundtrip():
    """eval_data columns ride along with fitnesses (reference core.py:
    eval_data_length); both batch- and solution-level setters agree."""

    @vectorized
    def with_data(x):
        fit = (x**2).sum(-1)
        return fit, x[:, :2] * 10

    prob = Problem("min", with_data, solution_length=4, initial_bounds=(-1, 1),
                   eval_data_length=2, seed=44)
    batch = prob.generate_batch(5)
    prob.evaluate(batch)
    evals = batch.access_evals()
    assert evals.shape == (5, 3)  # 1 objective + 2 data columns
    assert torch.allclose(evals[:, 1:], batch.unsafe_values[:, :2] * 10)

    sol = batch[0]
    assert sol.evals.shape == (3,)
    sol.set_evals(torch.tensor([7.0]), torch.tensor([1.0, 2.0]))
    assert batch.unsafe_evals[0].tolist() == [7.0, 1.0, 2.0]


def test_batch_set_values_subset():
    prob = make_problem(seed=45)
    batch = prob.generate_batch(6)
    prob.evaluate(batch)
    new_rows = torch.zeros(2, batch.unsafe_values.shape[1])
    batch.set_values(new_rows, solutions=torch.tensor([1, 3]))
    assert bool((batch.unsafe_values[1] == 0).all())
    assert bool((batch.unsafe_values[3] == 0).all())
    # touched rows forget their evals, untouched rows keep them
    assert torch.isnan(batch.unsafe_evals[1, 0])
    assert not torch.isnan(batch.unsafe_evals[0, 0])


def test_clear_error_messages():
    """Common misuses raise targeted errors, not obscure tracebacks."""
    with pytest.raises(ValueError):
        Problem("sideways", sphere, solution_length=4, initial_bounds=(-1, 1))  # bad sense
    with pytest.raises(ValueError):
        prob_bad = Problem("min", sphere, solution_length=4, initial_bounds=(3, -3))
        prob_bad.generate_batch(2)
    prob = make_problem()
    with pytest.raises(IndexError):
        prob.normalize_obj_index(5)
    batch = prob.generate_batch(4)
    with pytest.raises(IndexError):
        batch[10]
    with pytest.raises(ValueError):
        from evotorch_amd.distributions import SeparableGaussian

        SeparableGaussian({"mu": torch.zeros(3)})  # sigma missing
