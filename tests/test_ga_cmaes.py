"""GA stack + CMA-ES + MAPElites + Restart tests (mirrors reference
tests/test_ga.py and the CMAES smoke coverage in test_examples.py)."""

import math

import pytest
import torch

from evotorch_amd import Problem, SolutionBatch
from evotorch_amd.algorithms import CMAES, Cosyne, GeneticAlgorithm, IPOP, MAPElites, Restart, SteadyStateGA
from evotorch_amd.decorators import vectorized
from evotorch_amd.operators import (
    CosynePermutation,
    GaussianMutation,
    MultiPointCrossOver,
    OnePointCrossOver,
    PolynomialMutation,
    SimulatedBinaryCrossOver,
    TwoPointCrossOver,
)


@vectorized
def sphere(x):
    return (x**2).sum(-1)


@vectorized
def rastrigin(x):
    return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)


def make_problem(length=10, seed=0, **kw):
    defaults = dict(objective_sense="min", objective_func=sphere, solution_length=length, initial_bounds=(-3, 3), seed=seed)
    defaults.update(kw)
    return Problem(**defaults)


# -- operators ---------------------------------------------------------------


def test_gaussian_mutation_changes_values_respects_bounds():
    prob = make_problem(bounds=(-1.0, 1.0))
    batch = prob.generate_batch(20)
    prob.evaluate(batch)
    op = GaussianMutation(prob, stdev=0.5)
    out = op(batch)
    assert not torch.allclose(out.unsafe_values, batch.unsafe_values)
    assert float(out.unsafe_values.min()) >= -1.0
    assert float(out.unsafe_values.max()) <= 1.0


def test_crossover_children_mix_parent_genes():
    prob = make_problem(length=8)
    batch = prob.generate_batch(16, empty=True)
    batch.access_values().copy_(torch.arange(16).unsqueeze(-1).expand(16, 8).float())
    batch.set_evals(-torch.arange(16, dtype=torch.float32))  # row 0 best ("min")
    for op_cls, kw in [
        (OnePointCrossOver, {}),
        (TwoPointCrossOver, {}),
        (MultiPointCrossOver, {"num_points": 3}),
    ]:
        op = op_cls(prob, tournament_size=4, **kw)
        children = op(batch)
        assert len(children) == 16
        vals = children.unsafe_values
        # each gene of each child comes from some parent row (constant rows)
        assert torch.all((vals >= 0) & (vals <= 15))
        # each child's genes come from at most 2 distinct parents
        for r in range(len(vals)):
            assert len(set(vals[r].tolist())) <= 2


def test_sbx_preserves_mean():
    prob = make_problem(length=6)
    batch = prob.generate_batch(40)
    prob.evaluate(batch)
    op = SimulatedBinaryCrossOver(prob, tournament_size=2, eta=10.0)
    children = op(batch)
    assert len(children) == 40


def test_polynomial_mutation_requires_bounds():
    prob = make_problem()
    with pytest.raises(ValueError):
        PolynomialMutation(prob)
    prob_b = make_problem(bounds=(-2.0, 2.0))
    op = PolynomialMutation(prob_b, eta=20.0)
    batch = prob_b.generate_batch(10)
    prob_b.evaluate(batch)
    out = op(batch)
    assert float(out.unsafe_values.abs().max()) <= 2.0


def test_cosyne_permutation_preserves_column_multisets():
    prob = make_problem(length=5)
    batch = prob.generate_batch(12)
    prob.evaluate(batch)
    op = CosynePermutation(prob, permute_all=True)
    out = op(batch)
    a = batch.unsafe_values
    b = out.unsafe_values
    for col in range(5):
        assert torch.allclose(a[:, col].sort().values, b[:, col].sort().values)


# -- GA ----------------------------------------------------------------------


def test_ga_converges_on_sphere():
    prob = make_problem(seed=42)
    ga = GeneticAlgorithm(
        prob,
        popsize=60,
        operators=[
            OnePointCrossOver(prob, tournament_size=4),
            GaussianMutation(prob, stdev=0.2),
        ],
    )
    ga.step()
    first = ga.status["pop_best_eval"]
    ga.run(40)
    assert ga.status["pop_best_eval"] < first * 0.3
    assert len(ga.population) == 60


def test_steady_state_ga_use():
    prob = make_problem(seed=1)
    ga = SteadyStateGA(prob, popsize=30)
    ga.use(OnePointCrossOver(prob, tournament_size=2))
    ga.use(GaussianMutation(prob, stdev=0.1))
    ga.run(5)
    assert ga.status["iter"] == 5


def test_ga_multiobjective_nsga2():
    @vectorized
    def two_obj(x):
        f1 = (x**2).sum(-1)
        f2 = ((x - 2.0) ** 2).sum(-1)
        return torch.stack([f1, f2], dim=-1)

    prob = Problem(["min", "min"], two_obj, solution_length=5, initial_bounds=(-4, 4), seed=3)
    ga = GeneticAlgorithm(
        prob,
        popsize=40,
        operators=[SimulatedBinaryCrossOver(prob, tournament_size=3, eta=8.0), GaussianMutation(prob, stdev=0.2)],
    )
    ga.run(15)
    pop = ga.population
    ranks, _ = pop.compute_pareto_ranks()
    # after optimization most of the population should be on the first front
    assert float((ranks == 0).float().mean()) > 0.5


def test_cosyne_runs_and_improves():
    prob = make_problem(seed=7)
    searcher = Cosyne(prob, popsize=50, tournament_size=4, mutation_stdev=0.2, num_elites=2)
    searcher.step()
    first = searcher.status["pop_best_eval"]
    searcher.run(30)
    assert searcher.status["pop_best_eval"] <= first


# -- CMA-ES ------------------------------------------------------------------


def test_cmaes_converges_on_sphere():
    prob = make_problem(length=12, seed=5)
    searcher = CMAES(prob, stdev_init=2.0)
    searcher.run(120)
    assert searcher.status["pop_best_eval"] < 1e-2


def test_cmaes_separable_converges():
    prob = make_problem(length=12, seed=6)
    searcher = CMAES(prob, stdev_init=2.0, separable=True)
    searcher.run(150)
    assert searcher.status["pop_best_eval"] < 0.5


def test_cmaes_rotated_ellipsoid_needs_cov():
    # full-covariance CMA should handle a correlated quadratic
    torch.manual_seed(0)
    n = 8
    Q, _ = torch.linalg.qr(torch.randn(n, n))
    scales = torch.logspace(0, 2, n)
    Amat = Q @ torch.diag(scales) @ Q.T

    @vectorized
    def rotated(x):
        y = x @ Amat
        return (y**2).sum(-1)

    prob = Problem("min", rotated, solution_length=n, initial_bounds=(-3, 3), seed=8)
    searcher = CMAES(prob, stdev_init=1.0)
    searcher.run(250)
    assert searcher.status["pop_best_eval"] < 1.0


def test_cmaes_status_keys():
    prob = make_problem()
    searcher = CMAES(prob, stdev_init=1.0)
    searcher.step()
    for key in ("center", "stdev", "sigma", "mean_eval", "pop_best"):
        assert key in searcher.status


# -- MAPElites ---------------------------------------------------------------


def test_mapelites_fills_grid():
    @vectorized
    def f(x):
        fitness = -(x**2).sum(-1)
        feature = x[:, :1]  # first coordinate is the descriptor
        return fitness, feature

    prob = Problem("max", f, solution_length=3, initial_bounds=(-2, 2), eval_data_length=1, seed=11)
    grid = MAPElites.make_feature_grid([-2.0], [2.0], 10)
    assert grid.shape == (10, 1, 2)
    me = MAPElites(prob, operators=[GaussianMutation(prob, stdev=0.3)], feature_grid=grid)
    me.run(20)
    assert me.filled is not None
    assert int(me.filled.sum()) >= 5  # most cells discovered
    # each filled cell's occupant has its feature inside the cell box
    evals = me.population.access_evals()
    for c in torch.nonzero(me.filled).reshape(-1).tolist():
        feat = float(evals[c, 1])
        lo, hi = float(grid[c, 0, 0]), float(grid[c, 0, 1])
        assert lo <= feat <= hi


# -- Restart -----------------------------------------------------------------


def test_restart_and_ipop():
    from evotorch_amd.algorithms import CMAES

    prob = make_problem(length=5, seed=13)
    restarter = IPOP(
        prob,
        lambda p, **kw: CMAES(p, stdev_init=1.0, **kw),
        algorithm_args={"popsize": 8},
        max_inner_steps=5,
    )
    restarter.run(12)
    assert restarter.num_restarts >= 2
    assert restarter._algorithm_args["popsize"] >= 16


def test_object_dtype_ga_with_cut_and_splice():
    """Variable-length sequence evolution (object dtype) end to end —
    mirrors the reference's CutAndSplice usage (operators/sequence.py:25)."""
    import numpy as np

    from evotorch_amd.operators import CutAndSplice

    target = [1.0, 2.0, 3.0, 4.0, 5.0]

    class SeqProblem(Problem):
        def __init__(self):
            super().__init__(objective_sense="min", dtype=object, eval_dtype=torch.float32, seed=3)
            self._rng = np.random.default_rng(0)

        def _fill(self, values):
            for i in range(len(values)):
                n = int(self._rng.integers(1, 8))
                values[i] = [float(x) for x in self._rng.uniform(0, 6, n)]

        def _evaluate(self, solution):
            seq = list(solution.values)
            # distance to target: elementwise + length penalty
            cost = abs(len(seq) - len(target)) * 5.0
            for a, b in zip(seq, target):
                cost += abs(float(a) - b)
            solution.set_evaluation(cost)

    prob = SeqProblem()
    ga = GeneticAlgorithm(prob, popsize=40, operators=[CutAndSplice(prob, tournament_size=4)])
    ga.step()
    first = ga.status["pop_best_eval"]
    ga.run(25)
    assert ga.status["pop_best_eval"] <= first
    best = ga.population.take_best()
    assert isinstance(list(best.values), list)


@pytest.mark.parametrize("kwargs", [
    {"active": False},
    {"csa_squared": True},
    {"c_sigma_ratio": 0.5, "damp_sigma_ratio": 2.0},
    {"stdev_min": 1e-8, "stdev_max": 5.0},
    {"limit_C_decomposition": False},
])
def test_cmaes_variants_converge(kwargs):
    """Every CMA-ES configuration knob still yields a working optimizer."""
    prob = Problem("min", sphere, solution_length=8, initial_bounds=(-3, 3), seed=31)
    searcher = CMAES(prob, stdev_init=2.0, **kwargs)
    searcher.step()
    first = float(searcher.status["mean_eval"])
    searcher.run(60)
    assert float(searcher.status["mean_eval"]) < first * 0.2, kwargs


def test_cmaes_rotation_invariance_on_ellipsoid():
    """Full-covariance CMA-ES solves a ROTATED ill-conditioned ellipsoid
    about as fast as the axis-aligned one (the defining property of
    covariance adaptation; a separable method cannot do this)."""
    d = 12
    cond = torch.logspace(0, 3, d)  # condition number 1e3
    q, _ = torch.linalg.qr(torch.randn(d, d, generator=torch.Generator().manual_seed(3)))

    def make(rotated):
        @vectorized
        def ell(x):
            y = x @ q.T if rotated else x
            return (cond * y**2).sum(-1)

        return Problem("min", ell, solution_length=d, initial_bounds=(-3, 3), seed=5)

    finals = {}
    for rotated in (False, True):
        s = CMAES(make(rotated), stdev_init=2.0, popsize=24)
        s.run(400)
        finals[rotated] = float(s.status["pop_best_eval"])
    assert finals[True] < 1e-3, finals
    assert finals[False] < 1e-3, finals


def test_bipop_restart_regimes():
    """Native BIPOP: restarts alternate large (doubling popsize) and small
    (randomized sub-default-to-half-large popsize) regimes by spent
    evaluation budget, and the search still descends."""
    import torch

    from evotorch_amd import Problem
    from evotorch_amd.algorithms import BIPOP, CMAES
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=6, initial_bounds=(-3, 3), seed=2)
    meta = BIPOP(
        prob,
        lambda p, **kw: CMAES(p, stdev_init=1.0, **kw),
        algorithm_args={"popsize": 8},
        max_inner_steps=5,  # force frequent restarts
        seed=7,
    )
    regimes = []
    popsizes = []
    for _ in range(40):
        meta.step()
        regimes.append(meta.status["regime"])
        popsizes.append(meta.status["current_popsize"])
    assert meta.num_restarts >= 6
    assert "small" in regimes and "large" in regimes
    large_sizes = sorted({p for r, p in zip(regimes, popsizes) if r == "large"})
    assert len(large_sizes) >= 2 and large_sizes[1] == large_sizes[0] * 2  # doubling
    small_sizes = {p for r, p in zip(regimes, popsizes) if r == "small"}
    assert all(4 <= p <= max(large_sizes) for p in small_sizes)
    assert meta.status["mean_eval"] < 54.0  # descending on sphere (E||x||² = 6·9)
