"""GPU edge-shape battery: odd/minimal sizes through every kernel-backed
path (guards and tail handling, not throughput)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs ROCm GPU")


@requires_gpu
def test_pgpe_tiny_popsize_gpu():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized

    @vectorized
    def f(x):
        return (x**2).sum(-1)

    prob = Problem("min", f, solution_length=3, initial_bounds=(-1, 1), seed=1, device="cuda:0")
    s = PGPE(prob, popsize=2, center_learning_rate=0.1, stdev_learning_rate=0.1, stdev_init=0.5, distributed=True)
    s.run(3)
    assert s.status["iter"] == 3


@requires_gpu
def test_cma_small_and_odd_dims_gpu():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import CMAES
    from evotorch_amd.decorators import vectorized

    @vectorized
    def f(x):
        return (x**2).sum(-1)

    for d in (1, 2, 65):
        prob = Problem("min", f, solution_length=d, initial_bounds=(-1, 1), seed=d, device="cuda:0")
        s = CMAES(prob, stdev_init=1.0, popsize=8)
        s.run(5)
        assert s.status["iter"] == 5


@requires_gpu
def test_nsga_tiny_and_three_objectives_gpu():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import GeneticAlgorithm
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.operators import PolynomialMutation, SimulatedBinaryCrossOver

    @vectorized
    def f3(x):
        return torch.stack([(x**2).sum(-1), ((x - 1) ** 2).sum(-1), ((x + 1) ** 2).sum(-1)], dim=-1)

    prob = Problem(["min", "min", "min"], f3, solution_length=5, initial_bounds=(0, 1), bounds=(0.0, 1.0),
                   seed=2, device="cuda:0")
    ga = GeneticAlgorithm(prob, popsize=7, operators=[
        SimulatedBinaryCrossOver(prob, tournament_size=2, eta=10),
        PolynomialMutation(prob, eta=20),
    ])
    ga.run(4)
    ranks, crowd = ga.population.compute_pareto_ranks()
    assert int(ranks.min()) == 0


@requires_gpu
def test_rollout_single_member_and_tails_gpu():
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    for pop in (2, 18):  # 2 < one block; 18 = one full v7 block + tail of 2
        prob = SyntheticRolloutProblem(device="cuda:0", seed=pop, episode_length=7)
        s = PGPE(prob, popsize=pop, center_learning_rate=0.05, stdev_learning_rate=0.1,
                 radius_init=1.0, distributed=True)
        s.run(2)
        assert s.status["iter"] == 2


@requires_gpu
def test_fused_rank_extremes_gpu():
    from evotorch_amd import ops
    from evotorch_amd.utils import ranking

    for n in (1, 2, 3, 8192):
        x = torch.randn(n, device="cuda:0")
        for method in ("centered", "linear", "nes"):
            w = ranking.rank(x, method, higher_is_better=True)
            assert w.shape == (n,)
            assert torch.isfinite(w).all()


@requires_gpu
def test_mapelites_tiny_grid_gpu():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import MAPElites, make_feature_grid
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.operators import GaussianMutation

    @vectorized
    def f(x):
        base = (x**2).sum(-1)
        return torch.stack([base, x[:, 0]], dim=-1)

    prob = Problem("min", f, solution_length=4, initial_bounds=(-1, 1), seed=5, device="cuda:0",
                   eval_data_length=1)
    grid = make_feature_grid(lower_bounds=[-1.0], upper_bounds=[1.0], num_bins=3, device="cuda:0")
    s = MAPElites(prob, feature_grid=grid, re_evaluate=False, operators=[GaussianMutation(prob, stdev=0.2)])
    s.run(4)
    assert s.status["iter"] == 4


@requires_gpu
@pytest.mark.parametrize("dt", [torch.bfloat16, torch.float16, torch.float64])
def test_nondefault_dtypes_gpu(dt):
    """Non-fp32 problem dtypes through the kernel-backed PGPE loop."""
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized

    @vectorized
    def f(x):
        return (x.float() ** 2).sum(-1)

    prob = Problem("min", f, solution_length=33, initial_bounds=(-1, 1), seed=3, device="cuda:0", dtype=dt)
    s = PGPE(prob, popsize=16, center_learning_rate=0.1, stdev_learning_rate=0.1, stdev_init=0.5, distributed=True)
    s.run(3)
    assert s.status["iter"] == 3
    assert torch.isfinite(torch.as_tensor(float(s.status["mean_eval"])))


@requires_gpu
def test_m7_mlp_rollout_bitwise_deterministic():
    """The m7 MLP kernel has no atomics and fixed-order reductions: two
    identical launches must agree BITWISE (same contract as v7)."""
    from evotorch_amd import ops
    from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec

    mod = ops.hip_required()
    spec = SyntheticEnvSpec(episode_length=100, device="cuda", policy_hidden=64)
    torch.manual_seed(9)
    params = 0.1 * torch.randn(37, spec.solution_length, device="cuda")
    mean = torch.zeros(spec.obs_dim, device="cuda")
    std = torch.ones(spec.obs_dim, device="cuda")
    blob = spec.env_blob(mean, std, device="cuda")

    def run():
        stats = torch.zeros(2 * spec.obs_dim, device="cuda")
        fit = mod.rollout_linear(params, blob, stats, spec.obs_dim, spec.act_dim, spec.rank,
                                 spec.episode_length, spec.alive_bonus, spec.act_cost, 55, 0,
                                 spec.policy_hidden)
        return fit, stats

    f1, s1 = run()
    f2, s2 = run()
    assert torch.equal(f1, f2)
    assert torch.equal(s1, s2)
