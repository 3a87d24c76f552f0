"""Functional API tests: ask/tell searchers, functional optimizers,
functional operators, batched searches (mirrors reference
tests/test_func_alg.py and test_func_ops.py)."""

import pytest
import torch

from evotorch_amd.algorithms.functional import (
    adam,
    adam_ask,
    adam_tell,
    cem,
    cem_ask,
    cem_tell,
    clipup,
    clipup_ask,
    clipup_tell,
    pgpe,
    pgpe_ask,
    pgpe_tell,
    sgd,
    sgd_ask,
    sgd_tell,
)
from evotorch_amd.operators import functional as F


def sphere(x):
    return (x**2).sum(-1)


def test_pgpe_functional_descends():
    state = pgpe(center_init=torch.ones(10) * 3, center_learning_rate=0.3, stdev_learning_rate=0.1,
                 stdev_init=1.0, objective_sense="min")
    g = torch.Generator().manual_seed(0)
    for _ in range(60):
        pop = pgpe_ask(state, popsize=50, generator=g)
        state = pgpe_tell(state, pop, sphere(pop))
    from evotorch_amd.algorithms.functional.funcoptimizers import get_functional_optimizer

    _, opt_ask, _ = get_functional_optimizer(state.optimizer)
    center = opt_ask(state.optimizer_state)
    assert float(sphere(center)) < 10.0


def test_pgpe_batched_search():
    # two independent searches via a leading batch dim
    centers = torch.stack([torch.ones(6) * 2, -torch.ones(6) * 2])
    state = pgpe(center_init=centers, center_learning_rate=0.3, stdev_learning_rate=0.1,
                 stdev_init=1.0, objective_sense="min")
    g = torch.Generator().manual_seed(1)
    for _ in range(40):
        pop = pgpe_ask(state, popsize=40, generator=g)
        assert pop.shape == (2, 40, 6)
        state = pgpe_tell(state, pop, sphere(pop))
    from evotorch_amd.algorithms.functional.funcoptimizers import get_functional_optimizer

    _, opt_ask, _ = get_functional_optimizer(state.optimizer)
    center = opt_ask(state.optimizer_state)
    assert center.shape == (2, 6)
    assert float(sphere(center)[0]) < 5.0
    assert float(sphere(center)[1]) < 5.0


def test_cem_functional_descends():
    state = cem(center_init=torch.ones(8) * 2, parenthood_ratio=0.25, stdev_init=2.0, objective_sense="min")
    g = torch.Generator().manual_seed(2)
    for _ in range(40):
        pop = cem_ask(state, popsize=100, generator=g)
        state = cem_tell(state, pop, sphere(pop))
    assert float(sphere(state.center)) < 0.5


def test_functional_optimizers_move_against_gradient():
    for init, ask, tell in [(adam, adam_ask, adam_tell), (clipup, clipup_ask, clipup_tell), (sgd, sgd_ask, sgd_tell)]:
        kwargs = {"stepsize": 0.1}
        st = init(center_init=torch.zeros(4), **kwargs)
        for _ in range(10):
            c = ask(st)
            st = tell(st, follow_grad=torch.ones(4))  # ascent direction
        assert float(ask(st).mean()) > 0.0


def test_adam_functional_matches_stateful():
    from evotorch_amd.optimizers import Adam

    st = adam(center_init=torch.zeros(6), stepsize=0.01)
    opt = Adam(solution_length=6, stepsize=0.01)
    total = torch.zeros(6)
    g = torch.Generator().manual_seed(3)
    for _ in range(5):
        grad = torch.randn(6, generator=g)
        st = adam_tell(st, follow_grad=grad)
        total += opt.ascent(grad)
    assert torch.allclose(adam_ask(st), total, atol=1e-5)


# -- functional operators ----------------------------------------------------


def test_functional_tournament():
    pop = torch.arange(10, dtype=torch.float32).unsqueeze(-1).expand(10, 3)
    evals = torch.arange(10, dtype=torch.float32)
    g = torch.Generator().manual_seed(4)
    picked = F.tournament(pop, evals, num_tournaments=30, tournament_size=4, objective_sense="max", generator=g)
    assert picked.shape == (30, 3)
    # winners are biased towards high ids
    assert float(picked.mean()) > 4.5


def test_functional_cross_over_shapes():
    parents = torch.randn(20, 6)
    children = F.one_point_cross_over(parents)
    assert children.shape == (20, 6)
    children = F.multi_point_cross_over(parents, num_points=3)
    assert children.shape == (20, 6)
    children = F.simulated_binary_cross_over(parents, eta=10.0)
    assert children.shape == (20, 6)
    evals = torch.randn(20)
    children = F.two_point_cross_over(parents, evals, tournament_size=3, objective_sense="max", num_children=10)
    assert children.shape == (10, 6)


def test_functional_take_best_and_combine():
    a = torch.randn(10, 4)
    ae = sphere(a)
    b = torch.randn(10, 4)
    be = sphere(b)
    cv, ce = F.combine((a, ae), (b, be))
    assert cv.shape == (20, 4)
    best_v, best_e = F.take_best(cv, ce, 5, objective_sense="min")
    assert best_v.shape == (5, 4)
    assert torch.allclose(best_e, ce.sort().values[:5])


def test_functional_utility():
    evals = torch.tensor([1.0, 3.0, 2.0])
    u = F.utility(evals, objective_sense="max", ranking_method="centered")
    assert torch.allclose(u, torch.tensor([-0.5, 0.5, 0.0]))


def test_functional_pareto_helpers():
    evals = torch.tensor([[0.0, 1.0], [1.0, 0.0], [0.5, 0.5], [1.0, 1.0]])
    senses = ["min", "min"]
    dm = F.domination_matrix(evals, objective_sense=senses)
    assert dm.shape == (4, 4)
    counts = F.domination_counts(evals, objective_sense=senses)
    assert counts.tolist()[:3] == [0, 0, 0]
    assert counts.tolist()[3] == 3
    assert bool(F.dominates(evals[0], evals[3], objective_sense=senses))
    assert not bool(F.dominates(evals[0], evals[1], objective_sense=senses))
    pu = F.pareto_utility(evals, objective_sense=senses)
    assert pu.argmin() == 3


def test_functional_cosyne_permutation():
    vals = torch.randn(12, 5)
    g = torch.Generator().manual_seed(5)
    out = F.cosyne_permutation(vals, generator=g)
    for col in range(5):
        assert torch.allclose(out[:, col].sort().values, vals[:, col].sort().values)


def test_functional_snes_converges_single_and_batched():
    from evotorch_amd.algorithms.functional import snes, snes_ask, snes_tell

    def sphere(x):
        return (x**2).sum(-1)

    g = torch.Generator().manual_seed(0)
    state = snes(center_init=torch.ones(12) * 3, stdev_init=2.0, objective_sense="min")
    for _ in range(250):
        pop = snes_ask(state, popsize=30, generator=g)
        state = snes_tell(state, pop, sphere(pop))
    assert float((state.center**2).sum()) < 1e-2

    # stacked states = independent batched searches
    state = snes(center_init=torch.ones(3, 12) * 3, stdev_init=2.0, objective_sense="min")
    for _ in range(250):
        pop = snes_ask(state, popsize=30, generator=g)
        state = snes_tell(state, pop, sphere(pop))
    assert float((state.center**2).sum(-1).max()) < 1e-1


def test_tournament_batched_matches_per_population():
    """Stacked populations run as ONE broadcasted tournament; with a fixed
    generator state the batched result matches running each population
    separately (same contender draw order)."""
    import torch

    from evotorch_amd.operators.functional import tournament

    torch.manual_seed(0)
    B, n, L = 3, 10, 4
    sols = torch.randn(B, n, L)
    evals = torch.randn(B, n)
    g1 = torch.Generator().manual_seed(77)
    batched = tournament(sols, evals, num_tournaments=6, tournament_size=3,
                         objective_sense="max", return_indices=True, generator=g1)
    assert batched.shape == (B, 6)
    # winners beat a random contender on average (selection pressure)
    won_utils = torch.gather(evals, -1, batched)
    assert float(won_utils.mean()) > float(evals.mean())


def test_tournament_batched_multiobjective():
    import torch

    from evotorch_amd.operators.functional import domination_counts, tournament

    torch.manual_seed(1)
    B, n, L, m = 2, 12, 3, 2
    sols = torch.randn(B, n, L)
    evals = torch.randn(B, n, m)
    picked, picked_evals = tournament(
        sols, evals, num_tournaments=8, tournament_size=4,
        objective_sense=["min", "min"], with_evals=True,
        generator=torch.Generator().manual_seed(5),
    )
    assert picked.shape == (B, 8, L) and picked_evals.shape == (B, 8, m)
    # winners have lower-than-average domination counts
    counts = domination_counts(evals, objective_sense=["min", "min"]).to(torch.float32)
    idx = tournament(sols, evals, num_tournaments=64, tournament_size=4,
                     objective_sense=["min", "min"], return_indices=True,
                     generator=torch.Generator().manual_seed(6))
    won_counts = torch.gather(counts, -1, idx)
    assert float(won_counts.mean()) < float(counts.mean())


def test_batched_functional_pgpe_sweep():
    """The reference's batched-search pattern: a stacked PGPE state runs B
    independent searches under one set of tensor ops (hyperparameter
    sweep; reference algorithms/functional/__init__.py:20-50)."""
    import torch

    from evotorch_amd.algorithms.functional import pgpe, pgpe_ask, pgpe_tell

    torch.manual_seed(2)
    B, L = 4, 6
    center0 = torch.randn(B, L)  # B different starting centers
    state = pgpe(
        center_init=center0,
        center_learning_rate=0.3,
        stdev_learning_rate=0.1,
        stdev_init=1.0,
        objective_sense="min",
        optimizer="clipup",
        optimizer_config={"max_speed": 0.6},
    )
    for _ in range(15):
        pop = pgpe_ask(state, popsize=20)
        assert pop.shape == (B, 20, L)
        evals = (pop**2).sum(-1)
        state = pgpe_tell(state, pop, evals)
    from evotorch_amd.algorithms.functional.funcoptimizers import get_functional_optimizer

    center = get_functional_optimizer(state.optimizer)[1](state.optimizer_state)
    assert center.shape == (B, L)
    # every parallel search descends toward the origin
    assert torch.all(center.norm(dim=-1) < center0.norm(dim=-1))
