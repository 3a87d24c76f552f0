"""End-to-end GPU runs of the wider algorithm portfolio (beyond the
kernel-numerics suite)."""

import pytest
import torch

pytestmark = pytest.mark.gpu
requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs ROCm GPU")


@requires_gpu
def test_cmaes_gpu_converges():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import CMAES
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=256, initial_bounds=(-3, 3), device="cuda:0", seed=2)
    searcher = CMAES(prob, stdev_init=2.0, popsize=64)
    searcher.step()
    first = searcher.status["pop_best_eval"]
    searcher.run(60)
    assert searcher.status["pop_best_eval"] < first * 0.2


@requires_gpu
def test_vecenvne_gpu_rollout():
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE

    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=20, obs_dim=32, act_dim=8, rank=8, device="cuda:0")

    prob = VecEnvNE(env_factory, "Linear(obs_length, 16) >> Tanh() >> Linear(16, act_length)",
                    device="cuda:0", seed=3)
    searcher = PGPE(prob, popsize=64, center_learning_rate=0.05, stdev_learning_rate=0.1,
                    stdev_init=0.1, distributed=True)
    searcher.run(3)
    assert searcher.status["iter"] == 3
    assert prob.obs_norm.has_data


@requires_gpu
def test_supervisedne_gpu():
    import torch.nn as nn

    from evotorch_amd.algorithms import CEM
    from evotorch_amd.neuroevolution import SupervisedNE

    torch.manual_seed(0)
    x = torch.randn(512, 8)
    y = x @ torch.randn(8, 1)
    ds = torch.utils.data.TensorDataset(x, y)
    prob = SupervisedNE(ds, nn.Linear(8, 1), nn.MSELoss(), minibatch_size=128,
                        common_minibatch=True, device="cuda:0", seed=4)
    searcher = CEM(prob, popsize=64, parenthood_ratio=0.25, stdev_init=0.5)
    searcher.step()
    first = searcher.status["mean_eval"]
    searcher.run(20)
    assert searcher.status["mean_eval"] < first


@requires_gpu
def test_ga_singleobj_gpu():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import GeneticAlgorithm
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.operators import GaussianMutation, OnePointCrossOver

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=64, initial_bounds=(-3, 3), device="cuda:0", seed=5)
    ga = GeneticAlgorithm(prob, popsize=256, operators=[
        OnePointCrossOver(prob, tournament_size=4),
        GaussianMutation(prob, stdev=0.2),
    ])
    ga.step()
    first = ga.status["pop_best_eval"]
    ga.run(20)
    assert ga.status["pop_best_eval"] < first


@requires_gpu
def test_graphed_snes_converges_and_resamples():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import SNES, GraphedSearch
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=100, initial_bounds=(-5, 5), device="cuda:0", seed=1)
    searcher = SNES(prob, stdev_init=5.0, popsize=500)
    graphed = GraphedSearch(searcher)
    graphed.capture()
    first = graphed.mean_eval
    pop_before = searcher.population.unsafe_values.clone()
    graphed.run(100)
    pop_after = searcher.population.unsafe_values
    # replays resample (graph-safe seed bump) and the search descends
    assert not torch.allclose(pop_before, pop_after)
    assert graphed.mean_eval < first * 0.5, (first, graphed.mean_eval)
    assert searcher.status["iter"] >= 100


@requires_gpu
def test_graphed_pgpe_clipup():
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE, GraphedSearch
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=64, initial_bounds=(-3, 3), device="cuda:0", seed=2)
    searcher = PGPE(prob, popsize=100, center_learning_rate=0.2, stdev_learning_rate=0.1,
                    stdev_init=2.0, optimizer="clipup")
    graphed = GraphedSearch(searcher)
    graphed.capture()
    first = graphed.mean_eval
    graphed.run(300)
    assert graphed.mean_eval < first * 0.5, (first, graphed.mean_eval)


@requires_gpu
def test_structures_on_gpu():
    from evotorch_amd.utils import CDict, CList, CMemory

    m = CMemory(4, num_keys=8, batch_size=16, fill_with=0.0, device="cuda:0")
    keys = torch.randint(0, 8, (16,), device="cuda:0")
    m.set_(keys, torch.randn(16, 4, device="cuda:0"))
    got = m.get(keys)
    assert got.shape == (16, 4)

    lst = CList(max_length=8, batch_size=32, device="cuda:0")
    lst.append_(torch.arange(32.0, device="cuda:0"))
    lst.append_(torch.arange(32.0, device="cuda:0") * 2, where=torch.arange(32, device="cuda:0") % 2 == 0)
    assert lst.length.sum().item() == 32 + 16

    d = CDict(num_keys=5, batch_size=8, fill_with=0.0, device="cuda:0")
    k = torch.zeros(8, dtype=torch.int64, device="cuda:0")
    d.set_(k, 3.0)
    assert bool(d.contains(k).all())


@requires_gpu
def test_genetic_programming_example_gpu():
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [sys.executable, os.path.join(repo, "examples", "genetic_programming.py"),
         "--generations", "10", "--popsize", "128", "--device", "cuda:0"],
        capture_output=True, text=True, timeout=300,
    )
    assert result.returncode == 0, result.stderr[-2000:]
    assert "best MSE" in result.stdout


@requires_gpu
def test_state_dict_resume_on_gpu():
    """Checkpoint from a GPU searcher restores onto a fresh one (cross-device:
    saved tensors are cpu, load puts them back on cuda)."""
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    def fresh(seed):
        prob = SyntheticRolloutProblem(seed=90, episode_length=20, device="cuda:0")
        return PGPE(prob, popsize=64, center_learning_rate=0.15, stdev_learning_rate=0.1,
                    stdev_init=0.1, distributed=True)

    a = fresh(90)
    a.run(3)
    sd = a.state_dict()
    assert all(not v.is_cuda for v in sd["items"]["distribution"].values() if torch.is_tensor(v))

    b = fresh(90)
    b.load_state_dict(sd)
    assert b.step_count == 3
    center_a = torch.Tensor.as_subclass(a.status["center"], torch.Tensor)
    center_b = torch.Tensor.as_subclass(b.status["center"], torch.Tensor)
    assert center_b.is_cuda and torch.allclose(center_a, center_b)
    assert float(b.problem.obs_norm.count) == float(a.problem.obs_norm.count)
    b.run(2)
    assert b.step_count == 5


@requires_gpu
def test_misc_searchers_on_gpu():
    """One compact pass over the searcher families not covered by the
    dedicated GPU tests: XNES, MAPElites, NSGA-II GA, IPOP restarts, and
    the functional API — all on cuda:0."""
    _misc_searchers_body("cuda:0")


def _misc_searchers_body(dev):
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import IPOP, SNES, XNES, GeneticAlgorithm, MAPElites
    from evotorch_amd.algorithms.functional import pgpe, pgpe_ask, pgpe_tell
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.operators import GaussianMutation, PolynomialMutation, SimulatedBinaryCrossOver

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    # XNES (full covariance, matrix_exp on rocBLAS/rocSOLVER)
    prob = Problem("min", sphere, solution_length=16, initial_bounds=(-3, 3), device=dev, seed=1)
    xnes = XNES(prob, stdev_init=2.0)
    xnes.run(20)
    assert float(xnes.status["mean_eval"]) < 16 * 9

    # MAPElites
    @vectorized
    def with_features(x):
        fitness = -(x**2).sum(-1)
        return fitness, x[:, :2]

    prob = Problem("max", with_features, solution_length=8, initial_bounds=(-2, 2),
                   device=dev, seed=2, eval_data_length=2)
    grid = MAPElites.make_feature_grid([-2.0, -2.0], [2.0, 2.0], [6, 6], device=dev)
    me = MAPElites(prob, operators=[GaussianMutation(prob, stdev=0.3)], feature_grid=grid)
    me.run(10)
    assert int(me.filled.sum()) > 10

    # NSGA-II end to end on device (pareto HIP kernels inside argsort on cuda)
    @vectorized
    def two_obj(x):
        f1 = x[:, 0]
        g = 1 + 9 * x[:, 1:].mean(dim=-1)
        return torch.stack([f1, g * (1 - torch.sqrt((f1 / g).clamp(min=0)))], dim=-1)

    prob = Problem(["min", "min"], two_obj, solution_length=8, bounds=(0.0, 1.0),
                   initial_bounds=(0.0, 1.0), device=dev, seed=3)
    ga = GeneticAlgorithm(prob, popsize=256, operators=[
        SimulatedBinaryCrossOver(prob, eta=15, tournament_size=2),
        PolynomialMutation(prob, eta=20, mutation_probability=0.125)])
    ga.run(60)
    ranks, _ = ga.population.compute_pareto_ranks(crowdsort=False)
    assert int((ranks == 0).sum()) > 128

    # Cosyne (column permutation operator on device)
    from evotorch_amd.algorithms import Cosyne

    prob = Problem("min", sphere, solution_length=8, initial_bounds=(-3, 3), device=dev, seed=11)
    cosyne = Cosyne(prob, popsize=40, tournament_size=4, mutation_stdev=0.3)
    cosyne.run(10)
    assert float(cosyne.status["pop_best_eval"]) < float("inf")

    # IPOP restart wrapper around SNES
    prob = Problem("min", sphere, solution_length=8, initial_bounds=(-3, 3), device=dev, seed=4)
    ipop = IPOP(prob, SNES, algorithm_args={"stdev_init": 1.0, "popsize": 20},
                max_inner_steps=5)
    ipop.run(12)
    assert ipop.step_count == 12
    assert ipop.num_restarts >= 2

    # functional API on device
    state = pgpe(center_init=torch.ones(8, device=dev) * 2, radius_init=1.0,
                 center_learning_rate=0.3, stdev_learning_rate=0.1,
                 optimizer="clipup", optimizer_config={"max_speed": 0.3},
                 ranking_method="centered", objective_sense="min")
    for _ in range(30):
        pop = pgpe_ask(state, popsize=64)
        fit = sphere(pop)
        state = pgpe_tell(state, pop, fit)
    from evotorch_amd.algorithms.functional import get_functional_optimizer

    _, opt_ask, _ = get_functional_optimizer(state.optimizer)
    center = opt_ask(state.optimizer_state)
    assert float((center**2).sum()) < 8 * 4


@requires_gpu
def test_streaming_with_rollout_problem():
    """Streaming gradients compose with the fused-rollout problem: chunked
    evaluation feeds obs-norm per chunk and the searcher still learns."""
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    prob = SyntheticRolloutProblem(device="cuda:0", seed=17, episode_length=50)
    searcher = PGPE(prob, popsize=256, radius_init=2.25, center_learning_rate=0.1,
                    stdev_learning_rate=0.1, optimizer="clipup",
                    distributed=True, grad_chunk_rows=32)
    searcher.step()
    first = float(searcher.status["mean_eval"])
    searcher.run(30)
    last = float(searcher.status["mean_eval"])
    assert prob.obs_norm.count > 0
    assert last > first, (first, last)


@requires_gpu
def test_on_aux_device_fitness_runs_on_gpu():
    """A CPU problem whose fitness is @on_aux_device gets cuda batches
    (the reference's aux_device pattern for heavy fitness on accelerators)."""
    from evotorch_amd import Problem
    from evotorch_amd.decorators import on_aux_device, vectorized

    seen = []

    @vectorized
    @on_aux_device
    def f(x):
        seen.append(x.device.type)
        return (x**2).sum(-1)

    prob = Problem("min", f, solution_length=8, initial_bounds=(-1, 1), seed=1)  # cpu problem
    batch = prob.generate_batch(16)
    prob.evaluate(batch)
    assert seen[0] == "cuda"
    assert batch.evals_are_ready
    assert batch.unsafe_evals.device.type == "cpu"  # results land back on the problem device


@requires_gpu
def test_graphed_cem_and_adam():
    """Graph capture extended to CEM (elite top-k is shape-static) and to
    Adam via the device-side step counter."""
    import torch

    from evotorch_amd import Problem
    from evotorch_amd.algorithms import CEM, PGPE, GraphedSearch
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=30, initial_bounds=(-1, 1), seed=1, device="cuda:0")
    cem = CEM(prob, popsize=64, parenthood_ratio=0.25, stdev_init=2.0,
              center_init=torch.full((30,), 3.0))
    g = GraphedSearch(cem)
    g.capture()
    first = g.mean_eval
    g.run(60)
    assert g.mean_eval < first * 0.2

    prob2 = Problem("min", sphere, solution_length=30, initial_bounds=(-1, 1), seed=2, device="cuda:0")
    pgpe = PGPE(prob2, popsize=64, center_learning_rate=0.1, stdev_learning_rate=0.05,
                stdev_init=1.0, center_init=torch.full((30,), 3.0), optimizer="adam",
                ranking_method="centered")
    g2 = GraphedSearch(pgpe)
    g2.capture()
    first2 = g2.mean_eval
    g2.run(100)
    assert g2.mean_eval < first2 * 0.5


@requires_gpu
def test_graphsafe_adam_matches_host_adam():
    """adam_step_graphsafe (device step counter) must track the host-count
    kernel step for step for several iterations."""
    import evotorch_amd._C as C
    import torch

    torch.manual_seed(0)
    n = 257
    m1 = torch.zeros(n, device="cuda"); v1 = torch.zeros(n, device="cuda")
    m2 = torch.zeros(n, device="cuda"); v2 = torch.zeros(n, device="cuda")
    t_buf = torch.zeros(1, dtype=torch.int64, device="cuda")
    out1 = torch.empty(n, device="cuda"); out2 = torch.empty(n, device="cuda")
    for t in range(1, 6):
        g = torch.randn(n, device="cuda")
        C.adam_step(out1, g, m1, v1, t, 1e-2, 0.9, 0.999, 1e-8)
        C.adam_step_graphsafe(out2, g, m2, v2, t_buf, 1e-2, 0.9, 0.999, 1e-8)
        assert torch.allclose(out1, out2, rtol=1e-5, atol=1e-7), t
    assert int(t_buf) == 5


@requires_gpu
def test_vecenvne_hip_graph_matches_eager():
    """use_hip_graph=True replays the whole episode as one hipGraph; the
    trajectory must match the eager loop (same ops, same order — the first
    generation runs eagerly by design while obs-norm has no data)."""
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE

    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=24, obs_dim=32, act_dim=8, rank=8, device="cuda:0")

    def run(use_graph):
        prob = VecEnvNE(env_factory, "Linear(obs_length, 16) >> Tanh() >> Linear(16, act_length)",
                        device="cuda:0", seed=7, use_hip_graph=use_graph)
        searcher = PGPE(prob, popsize=64, center_learning_rate=0.05, stdev_learning_rate=0.1,
                        stdev_init=0.1, distributed=True)
        searcher.run(4)
        return searcher.status["mean_eval"], searcher._distribution.mu.clone(), prob.interaction_count

    eval_e, mu_e, steps_e = run(False)
    eval_g, mu_g, steps_g = run(True)
    assert steps_e == steps_g
    torch.testing.assert_close(mu_g, mu_e, rtol=1e-4, atol=1e-5)
    assert abs(float(eval_g) - float(eval_e)) < 1e-2 * max(1.0, abs(float(eval_e)))


@requires_gpu
def test_vecenvne_hip_graph_recaptures_on_popsize_change():
    from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
    from evotorch_amd.core import SolutionBatch

    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=12, obs_dim=32, act_dim=8, rank=8, device="cuda:0")

    prob = VecEnvNE(env_factory, "Linear(obs_length, act_length)", device="cuda:0", seed=5,
                    use_hip_graph=True, observation_normalization=False)
    for n in (16, 16, 32, 16):
        batch = SolutionBatch(prob, popsize=n, device="cuda:0")
        batch.access_values()[:] = 0.01 * torch.randn(n, prob.solution_length, device="cuda:0")
        prob.evaluate(batch)
        assert batch.evals_are_ready


@requires_gpu
def test_graphed_flagship_rollout():
    """GraphedSearch over the fused-rollout flagship problem: the whole
    PGPE generation (sampling, rollout kernel, ranking, gradients,
    ClipUp, obs-norm merge) replays as ONE hipGraph. Episode seeds come
    from a device splitmix chain (bumped in-graph), so replays run fresh
    episodes and observation statistics keep accumulating."""
    from evotorch_amd.algorithms import PGPE, GraphedSearch
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    prob = SyntheticRolloutProblem(device="cuda:0", seed=11, episode_length=50)
    r = 2.25
    s = PGPE(prob, popsize=512, radius_init=r, center_learning_rate=0.75 * r / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15})
    g = GraphedSearch(s, generations_per_capture=5)
    g.capture()
    seed_before = int(prob._graph_seed_buf.item())
    count_before = prob.obs_norm.count
    first = float(g.mean_eval)
    g.run(100)
    assert int(prob._graph_seed_buf.item()) != seed_before  # in-graph bump ran
    assert prob.obs_norm.count > count_before  # stats merged inside the graph
    assert float(g.mean_eval) > first + 20.0  # it optimizes


@requires_gpu
def test_graphed_mlp_flagship_rollout():
    """GraphedSearch over the MLP-64 rollout problem (m7 kernel) with the
    device episode-seed chain."""
    from evotorch_amd.algorithms import PGPE, GraphedSearch
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    prob = SyntheticRolloutProblem(device="cuda:0", seed=13, episode_length=40, policy_hidden=64)
    r = 2.25
    s = PGPE(prob, popsize=128, radius_init=r, center_learning_rate=0.75 * r / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15})
    g = GraphedSearch(s, generations_per_capture=5)
    g.capture()
    first = float(g.mean_eval)
    g.run(60)
    assert float(g.mean_eval) > first + 10.0
