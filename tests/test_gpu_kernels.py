"""HIP kernel numerics: every gfx950 kernel vs its plain fp32 eager
reference (run on MI355X via `gpurun -- python -m pytest tests -m gpu`)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs ROCm GPU")


@pytest.fixture(scope="module")
def C():
    import evotorch_amd._C as _C

    return _C


@requires_gpu
class TestSampleGaussian:
    def test_matches_philox_reference(self, C):
        from evotorch_amd.neuroevolution.philox_ref import philox_normals_2d

        n, length = 64, 37
        mu = torch.linspace(-1, 1, length, device="cuda")
        sigma = torch.linspace(0.5, 2.0, length, device="cuda")
        out = torch.empty(n, length, device="cuda")
        C.sample_gaussian(out, mu, sigma, False, 12345)
        z = philox_normals_2d(12345, 0, n, length).cuda()
        expected = mu + sigma * z
        assert torch.allclose(out, expected, atol=1e-4, rtol=1e-4)

    def test_symmetric_mirrors(self, C):
        n, length = 128, 100
        mu = torch.randn(length, device="cuda")
        sigma = torch.rand(length, device="cuda") + 0.5
        out = torch.empty(n, length, device="cuda")
        C.sample_gaussian(out, mu, sigma, True, 7)
        mid = (out[: n // 2] + out[n // 2 :]) / 2
        assert torch.allclose(mid, mu.expand(n // 2, length), atol=1e-5)

    def test_statistics(self, C):
        n, length = 100_000, 16
        mu = torch.full((length,), 3.0, device="cuda")
        sigma = torch.full((length,), 2.0, device="cuda")
        out = torch.empty(n, length, device="cuda")
        C.sample_gaussian(out, mu, sigma, False, 99)
        assert torch.allclose(out.mean(0), mu, atol=0.05)
        assert torch.allclose(out.std(0), sigma, atol=0.05)

    def test_deterministic(self, C):
        mu = torch.zeros(50, device="cuda")
        sigma = torch.ones(50, device="cuda")
        a = torch.empty(32, 50, device="cuda")
        b = torch.empty(32, 50, device="cuda")
        C.sample_gaussian(a, mu, sigma, False, 42)
        C.sample_gaussian(b, mu, sigma, False, 42)
        assert torch.equal(a, b)

    def test_bf16(self, C):
        mu = torch.zeros(64, device="cuda", dtype=torch.bfloat16)
        sigma = torch.ones(64, device="cuda", dtype=torch.bfloat16)
        out = torch.empty(256, 64, device="cuda", dtype=torch.bfloat16)
        C.sample_gaussian(out, mu, sigma, True, 5)
        assert float(out.float().mean().abs()) < 0.1


@requires_gpu
class TestEsGradients:
    def _eager(self, samples, mu, sigma, weights, symmetric):
        import os

        os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "1"
        try:
            from evotorch_amd.ops.dispatch import es_gradients

            return es_gradients(samples, mu, sigma, weights, symmetric=symmetric)
        finally:
            os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "0"

    @pytest.mark.parametrize("symmetric", [False, True])
    def test_matches_eager(self, C, symmetric):
        torch.manual_seed(0)
        n, length = 512, 1000
        mu = torch.randn(length, device="cuda")
        sigma = torch.rand(length, device="cuda") + 0.5
        samples = mu + sigma * torch.randn(n, length, device="cuda")
        weights = torch.randn(n, device="cuda")
        mg, sg = C.es_gradients(samples, mu, sigma, weights, symmetric)
        emg, esg = self._eager(samples, mu, sigma, weights, symmetric)
        assert torch.allclose(mg, emg, rtol=1e-3, atol=1e-3)
        assert torch.allclose(sg, esg, rtol=1e-3, atol=1e-3)

    def test_snes_matches_eager(self, C):
        import os

        torch.manual_seed(1)
        n, length = 256, 5000
        mu = torch.randn(length, device="cuda")
        sigma = torch.rand(length, device="cuda") + 0.5
        samples = mu + sigma * torch.randn(n, length, device="cuda")
        weights = torch.randn(n, device="cuda")
        mg, sg = C.snes_gradients(samples, mu, sigma, weights)
        os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "1"
        try:
            from evotorch_amd.ops.dispatch import snes_gradients

            emg, esg = snes_gradients(samples, mu, sigma, weights)
        finally:
            os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "0"
        assert torch.allclose(mg, emg, rtol=1e-3, atol=1e-3)
        assert torch.allclose(sg, esg, rtol=1e-3, atol=1e-3)

    def test_large_L_small_N(self, C):
        # tall-skinny: the row-chunked atomic path
        torch.manual_seed(2)
        n, length = 16, 300_000
        mu = torch.zeros(length, device="cuda")
        sigma = torch.ones(length, device="cuda")
        samples = torch.randn(n, length, device="cuda")
        weights = torch.randn(n, device="cuda")
        mg, sg = C.es_gradients(samples, mu, sigma, weights, False)
        expected_mg = weights @ samples
        assert torch.allclose(mg, expected_mg, rtol=1e-3, atol=1e-3)


@requires_gpu
class TestOptimizerKernels:
    def test_clipup_matches_eager(self, C):
        torch.manual_seed(3)
        length = 6409
        v_gpu = torch.zeros(length, device="cuda")
        v_cpu = torch.zeros(length)
        for i in range(5):
            g = torch.randn(length)
            C.clipup_step(v_gpu, g.cuda(), 0.01, 0.15, 0.9)
            from evotorch_amd.ops.dispatch import clipup_step_

            clipup_step_(v_cpu, g, step_size=0.01, max_speed=0.15, momentum=0.9)
            assert torch.allclose(v_gpu.cpu(), v_cpu, rtol=1e-4, atol=1e-6), f"step {i}"

    def test_adam_matches_eager(self, C):
        torch.manual_seed(4)
        length = 1000
        m_gpu = torch.zeros(length, device="cuda")
        v_gpu = torch.zeros(length, device="cuda")
        m_cpu = torch.zeros(length)
        v_cpu = torch.zeros(length)
        for step in range(1, 4):
            g = torch.randn(length)
            out_gpu = torch.empty(length, device="cuda")
            out_cpu = torch.empty(length)
            C.adam_step(out_gpu, g.cuda(), m_gpu, v_gpu, step, 1e-2, 0.9, 0.999, 1e-8)
            from evotorch_amd.ops.dispatch import fused_adam_step_

            fused_adam_step_(out_cpu, g, m_cpu, v_cpu, step_count=step, stepsize=1e-2)
            assert torch.allclose(out_gpu.cpu(), out_cpu, rtol=1e-4, atol=1e-7)


@requires_gpu
class TestRollout:
    def test_matches_eager_reference(self, C):
        from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec, rollout_eager

        spec = SyntheticEnvSpec(episode_length=10, device="cuda")
        torch.manual_seed(5)
        n = 32
        params = 0.1 * torch.randn(n, spec.solution_length, device="cuda")
        mean = torch.zeros(spec.obs_dim, device="cuda")
        std = torch.ones(spec.obs_dim, device="cuda")
        blob = spec.env_blob(mean, std, device="cuda")
        obs_stats = torch.zeros(2 * spec.obs_dim, device="cuda")
        fit = C.rollout_linear(params, blob, obs_stats, spec.obs_dim, spec.act_dim, spec.rank,
                               spec.episode_length, spec.alive_bonus, spec.act_cost, 77, 0)
        efit, (count, esum, esumsq) = rollout_eager(spec, params, mean, std, init_seed=77)
        assert torch.allclose(fit, efit, rtol=2e-2, atol=2e-2), (fit[:4], efit[:4])
        assert torch.allclose(obs_stats[: spec.obs_dim], esum, rtol=5e-2, atol=5e-1)
        assert torch.allclose(obs_stats[spec.obs_dim :], esumsq, rtol=5e-2, atol=5e-1)

    def test_deterministic(self, C):
        from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec

        spec = SyntheticEnvSpec(episode_length=50, device="cuda")
        params = 0.1 * torch.randn(16, spec.solution_length, device="cuda")
        mean = torch.zeros(spec.obs_dim, device="cuda")
        std = torch.ones(spec.obs_dim, device="cuda")
        blob = spec.env_blob(mean, std, device="cuda")
        fits = []
        for _ in range(2):
            obs_stats = torch.zeros(2 * spec.obs_dim, device="cuda")
            fits.append(C.rollout_linear(params, blob, obs_stats, spec.obs_dim, spec.act_dim, spec.rank,
                                         spec.episode_length, spec.alive_bonus, spec.act_cost, 3, 0))
        assert torch.equal(fits[0], fits[1])

    def test_member_offset_matches_concat(self, C):
        """Sharding invariance: rows [8:16] computed with offset 8 equal the
        same rows of a 16-member run with offset 0."""
        from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec

        spec = SyntheticEnvSpec(episode_length=20, device="cuda")
        params = 0.1 * torch.randn(16, spec.solution_length, device="cuda")
        mean = torch.zeros(spec.obs_dim, device="cuda")
        std = torch.ones(spec.obs_dim, device="cuda")
        blob = spec.env_blob(mean, std, device="cuda")
        s1 = torch.zeros(2 * spec.obs_dim, device="cuda")
        full = C.rollout_linear(params, blob, s1, spec.obs_dim, spec.act_dim, spec.rank,
                                spec.episode_length, spec.alive_bonus, spec.act_cost, 9, 0)
        s2 = torch.zeros(2 * spec.obs_dim, device="cuda")
        shard = C.rollout_linear(params[8:].contiguous(), blob, s2, spec.obs_dim, spec.act_dim, spec.rank,
                                 spec.episode_length, spec.alive_bonus, spec.act_cost, 9, 8)
        assert torch.equal(full[8:], shard)


@requires_gpu
class TestEndToEndGPU:
    def test_pgpe_improves_on_gpu(self):
        from evotorch_amd.algorithms import PGPE
        from evotorch_amd.neuroevolution import SyntheticRolloutProblem

        problem = SyntheticRolloutProblem(device="cuda:0", seed=11, episode_length=50)
        searcher = PGPE(problem, popsize=256, radius_init=2.25,
                        center_learning_rate=0.1125, stdev_learning_rate=0.1,
                        optimizer="clipup", optimizer_config={"max_speed": 0.15},
                        distributed=True)
        searcher.step()
        first = searcher.status["mean_eval"]
        for _ in range(30):
            searcher.step()
        last = searcher.status["mean_eval"]
        assert last > first, f"PGPE did not improve on GPU: {first} -> {last}"

    def test_snes_sphere_gpu(self):
        from evotorch_amd import Problem
        from evotorch_amd.algorithms import SNES
        from evotorch_amd.decorators import vectorized

        @vectorized
        def sphere(x):
            return (x**2).sum(-1)

        prob = Problem("min", sphere, solution_length=100, initial_bounds=(-5, 5),
                       device="cuda:0", seed=1)
        searcher = SNES(prob, stdev_init=5.0, popsize=500)
        searcher.step()
        first = searcher.status["mean_eval"]
        searcher.run(50)
        assert searcher.status["mean_eval"] < first * 0.5


@requires_gpu
class TestParetoKernels:
    def test_matches_eager(self, C):
        import os

        torch.manual_seed(7)
        n, m = 2000, 3
        utils = torch.randn(n, m, device="cuda")
        counts = C.domination_counts(utils)
        ranks = C.pareto_ranks(utils)
        os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "1"
        try:
            from evotorch_amd.core import _compute_pareto_ranks_eager

            eranks, _ = _compute_pareto_ranks_eager(utils, crowdsort=False)
            a = utils.unsqueeze(1)
            b = utils.unsqueeze(0)
            dom = (a >= b).all(dim=-1) & (a > b).any(dim=-1)
            ecounts = dom.sum(dim=0)
        finally:
            os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "0"
        assert torch.equal(counts.to(torch.int64), ecounts.to(torch.int64))
        assert torch.equal(ranks, eranks)

    def test_nsga2_ga_on_gpu(self):
        from evotorch_amd import Problem
        from evotorch_amd.algorithms import GeneticAlgorithm
        from evotorch_amd.decorators import vectorized
        from evotorch_amd.operators import GaussianMutation, SimulatedBinaryCrossOver

        @vectorized
        def two_obj(x):
            f1 = (x**2).sum(-1)
            f2 = ((x - 2.0) ** 2).sum(-1)
            return torch.stack([f1, f2], dim=-1)

        prob = Problem(["min", "min"], two_obj, solution_length=8, initial_bounds=(-4, 4), seed=3, device="cuda:0")
        ga = GeneticAlgorithm(
            prob,
            popsize=512,
            operators=[SimulatedBinaryCrossOver(prob, tournament_size=3, eta=8.0), GaussianMutation(prob, stdev=0.2)],
        )
        ga.run(20)
        ranks, _ = ga.population.compute_pareto_ranks()
        assert float((ranks == 0).float().mean()) > 0.1


@requires_gpu
class TestMlpRollout:
    def test_mlp64_matches_eager(self, C):
        from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec, rollout_eager

        spec = SyntheticEnvSpec(episode_length=10, policy_hidden=64, device="cuda")
        torch.manual_seed(9)
        n = 16
        params = 0.1 * torch.randn(n, spec.solution_length, device="cuda")
        mean = torch.zeros(spec.obs_dim, device="cuda")
        std = torch.ones(spec.obs_dim, device="cuda")
        blob = spec.env_blob(mean, std, device="cuda")
        obs_stats = torch.zeros(2 * spec.obs_dim, device="cuda")
        fit = C.rollout_linear(params, blob, obs_stats, spec.obs_dim, spec.act_dim, spec.rank,
                               spec.episode_length, spec.alive_bonus, spec.act_cost, 55, 0,
                               spec.policy_hidden)
        efit, _ = rollout_eager(spec, params, mean, std, init_seed=55)
        assert torch.allclose(fit, efit, rtol=2e-2, atol=2e-2), (fit[:4], efit[:4])

    def test_mlp_problem_end_to_end(self):
        from evotorch_amd.algorithms import PGPE
        from evotorch_amd.neuroevolution import SyntheticRolloutProblem

        problem = SyntheticRolloutProblem(device="cuda:0", seed=12, episode_length=50, policy_hidden=64)
        searcher = PGPE(problem, popsize=128, radius_init=2.25, center_learning_rate=0.1,
                        stdev_learning_rate=0.1, optimizer="clipup", distributed=True)
        searcher.step()
        first = searcher.status["mean_eval"]
        for _ in range(20):
            searcher.step()
        assert searcher.status["mean_eval"] > first - 5.0


@requires_gpu
def test_counter_addressed_sampling_matches_cpu_reference():
    """GPU counter-addressed sampling (seed + row_offset, stream-per-row
    philox) is bit-equal in fp32 to the numpy philox reference, for any row
    partition — including an odd solution length (L % 4 != 0), where the
    old flat addressing could not align chunk starts."""
    from evotorch_amd.ops import sample_gaussian

    for L in (12, 7):  # multiple-of-4 fast path and odd-length tail path
        N = 8
        torch.manual_seed(0)
        mu = torch.randn(L, device="cuda:0")
        sigma = torch.rand(L, device="cuda:0") + 0.5

        full = torch.empty(N, L, device="cuda:0")
        sample_gaussian(full, mu, sigma, symmetric=True, seed=777)

        cpu_full = torch.empty(N, L)
        sample_gaussian(cpu_full, mu.cpu(), sigma.cpu(), symmetric=True, seed=777)
        assert torch.allclose(full.cpu(), cpu_full, atol=1e-6), (full.cpu() - cpu_full).abs().max()

        # chunked regeneration equals the whole on device
        d = N // 2
        for r0, rows in [(0, 1), (1, 3)]:
            chunk = torch.empty(rows * 2, L, device="cuda:0")
            sample_gaussian(chunk, mu, sigma, symmetric=True, seed=777, row_offset=r0)
            assert torch.equal(chunk[:rows], full[r0 : r0 + rows])
            assert torch.equal(chunk[rows:], full[d + r0 : d + r0 + rows])


@requires_gpu
def test_streamed_equals_materialized_gradients_gpu():
    """On GPU both paths draw the same philox seed from the problem
    generator, so streamed and materialized gradients agree numerically."""
    from evotorch_amd import Problem
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.distributions import SymmetricSeparableGaussian

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    def make(seed):
        return Problem("min", sphere, solution_length=16, initial_bounds=(-1, 1),
                       seed=seed, device="cuda:0")

    def dist():
        return SymmetricSeparableGaussian(
            {"mu": torch.zeros(16, device="cuda:0"), "sigma": torch.ones(16, device="cuda:0"),
             "divide_mu_grad_by": "num_directions", "divide_sigma_grad_by": "num_directions"})

    r_mat = make(21).sample_and_compute_gradients(dist(), 64, ranking_method="centered")
    r_str = make(21).sample_and_compute_gradients(dist(), 64, ranking_method="centered", chunk_rows=8)
    for k in ("mu", "sigma"):
        a = r_mat["gradients"][k]
        b = r_str["gradients"][k]
        assert torch.allclose(a, b, atol=1e-4), (k, (a - b).abs().max())


@requires_gpu
def test_streaming_large_l_smoke():
    """L = 20M separable ES generation with chunked (never-materialized)
    population: peak extra memory is chunk*L, not popsize*L."""
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized

    L = 20_000_000

    @vectorized
    def first_coords(x):
        return (x[:, :64] ** 2).sum(-1)

    prob = Problem("min", first_coords, solution_length=L, initial_bounds=(-0.1, 0.1),
                   seed=3, device="cuda:0")
    searcher = PGPE(prob, popsize=32, center_learning_rate=0.05, stdev_learning_rate=0.05,
                    stdev_init=0.1, distributed=True, grad_chunk_rows=4)
    # isolate from residue of earlier tests (e.g. hipGraph private pools
    # whose owners are awaiting garbage collection)
    import gc

    gc.collect()
    torch.cuda.empty_cache()
    base = torch.cuda.memory_allocated() / 2**30
    torch.cuda.reset_peak_memory_stats()
    searcher.step()
    searcher.step()
    peak = torch.cuda.max_memory_allocated() / 2**30 - base
    # materialized would need 32*20e6*4 = 2.4 GB for the population alone
    # (plus allocator churn); chunked path stays near parameter-vector cost
    assert peak < 2.2, f"peak {peak:.2f} GiB above baseline — streaming not effective"
    assert searcher.step_count == 2


@requires_gpu
def test_kernel_fuzz_random_shapes():
    """Randomized-shape sweep of K1/K3 vs the eager fp32 reference:
    odd lengths, odd popsizes, both dtypes, both layouts."""
    import os

    from evotorch_amd.ops import es_gradients, sample_gaussian, snes_gradients

    torch.manual_seed(7)  # global cpu+cuda seed: draws below must not depend on test order
    g = torch.Generator().manual_seed(7)
    for trial in range(12):
        length = int(torch.randint(1, 700, (1,), generator=g))
        rows = int(torch.randint(1, 97, (1,), generator=g))
        symmetric = bool(torch.rand(1, generator=g) < 0.5)
        dtype = torch.float32 if torch.rand(1, generator=g) < 0.7 else torch.bfloat16
        n = rows * 2 if symmetric else rows
        mu = torch.randn(length, device="cuda:0")
        sigma = torch.rand(length, device="cuda:0") + 0.3

        out = torch.empty(n, length, device="cuda:0", dtype=dtype)
        seed = 1000 + trial
        sample_gaussian(out, mu.to(dtype), sigma.to(dtype), symmetric=symmetric, seed=seed)
        cpu = torch.empty(n, length, dtype=dtype)
        sample_gaussian(cpu, mu.cpu().to(dtype), sigma.cpu().to(dtype), symmetric=symmetric, seed=seed)
        # fp32: the kernel uses fmaf while the cpu reference multiplies then
        # adds; near mu = -sigma*z cancellation the ABSOLUTE error is
        # ~ulp(sigma*z) (~2e-6). bf16: fma-vs-mul-add can land on either
        # side of a rounding boundary, so allow one bf16 ulp (2^-8 relative)
        # on top of a small absolute floor.
        if dtype == torch.float32:
            ok = torch.allclose(out.cpu().float(), cpu.float(), atol=1e-5)
        else:
            ok = torch.allclose(out.cpu().float(), cpu.float(), rtol=1.0 / 128.0, atol=0.05)
        assert ok, (trial, length, rows, symmetric, dtype, (out.cpu().float() - cpu.float()).abs().max())

        if dtype == torch.float32:
            weights = torch.randn(n, device="cuda:0")
            gm, gs = es_gradients(out, mu, sigma, weights, symmetric=symmetric)
            os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"] = "1"
            try:
                gm_ref, gs_ref = es_gradients(out, mu, sigma, weights, symmetric=symmetric)
            finally:
                del os.environ["EVOTORCH_AMD_ALLOW_EAGER_GPU"]
            assert torch.allclose(gm, gm_ref, atol=1e-3, rtol=1e-3), (trial, (gm - gm_ref).abs().max())
            assert torch.allclose(gs, gs_ref, atol=1e-3, rtol=1e-3), (trial, (gs - gs_ref).abs().max())


@requires_gpu
def test_pregen_overlap_bitwise_equals_direct_sampling():
    """The side-stream noise pre-generation + affine pass must produce the
    SAME population as direct counter-addressed sampling (both end in
    fmaf(sigma, z, mu) on the same philox draws)."""
    from evotorch_amd import Problem
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.distributions import SymmetricSeparableGaussian

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=33, initial_bounds=(-1, 1), seed=3, device="cuda:0")
    L = 33
    dist = SymmetricSeparableGaussian(
        {"mu": torch.randn(L, device="cuda"), "sigma": torch.rand(L, device="cuda") + 0.5}
    )
    dirs, n = 10, 20
    prob._schedule_pregen(dirs, 5, seed=991)
    out_pre = torch.empty(n, L, device="cuda")
    prob._sample_with_pregen(dist, out_pre, 991, 5, dirs)
    out_direct = torch.empty(n, L, device="cuda")
    dist.fill_counter_addressed(out_direct, seed=991, row_offset=5)
    torch.cuda.synchronize()
    assert torch.equal(out_pre, out_direct)


@requires_gpu
def test_sharded_world1_pgpe_with_overlap_improves():
    """The SPMD path (counter-addressed sampling + pregen overlap + fused
    all-reduce) at world size 1 on a real GPU: descends and is
    run-to-run deterministic."""
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.parallel.comm import Comm

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    def run():
        comm = Comm(device=torch.device("cuda", 0))
        prob = Problem("min", sphere, solution_length=24, initial_bounds=(-1, 1), seed=7, device="cuda:0")
        prob.use_comm(comm)
        searcher = PGPE(prob, popsize=64, center_learning_rate=0.3, stdev_learning_rate=0.1,
                        stdev_init=1.0, center_init=torch.ones(24), distributed=True)
        evals = []
        for _ in range(8):
            searcher.step()
            evals.append(float(searcher.status["mean_eval"]))
        center = torch.Tensor.as_subclass(searcher.status["center"], torch.Tensor).clone()
        return evals, center

    evals1, c1 = run()
    evals2, c2 = run()
    assert evals1[-1] < evals1[0]
    assert evals1 == evals2, "SPMD GPU trajectory must be deterministic"
    assert torch.equal(c1, c2)


@requires_gpu
def test_cma_update_c_matches_eager():
    """K5 fused covariance update vs the eager torch chain; exact symmetry
    by construction."""
    from evotorch_amd.ops import cma_update_c_

    torch.manual_seed(4)
    d, lam = 300, 48
    B = torch.randn(d, d)
    C0 = (B @ B.T) / d + torch.eye(d)
    y = torch.randn(lam, d)
    w = torch.cat([torch.rand(lam // 2) + 0.1, -0.1 * torch.rand(lam - lam // 2)])
    pc = torch.randn(d)
    for hs in (0.0, 1.0):
        hs_t = torch.tensor(hs)
        C_cpu = C0.clone()
        cma_update_c_(C_cpu, y, w, pc, hs_t, c1=0.02, cmu=0.05, cc=0.1)
        C_gpu = C0.clone().cuda()
        cma_update_c_(C_gpu, y.cuda(), w.cuda(), pc.cuda(), hs_t.cuda(), c1=0.02, cmu=0.05, cc=0.1)
        assert torch.allclose(C_gpu.cpu(), C_cpu, rtol=1e-4, atol=1e-4)
        assert torch.equal(C_gpu, C_gpu.T), "fused update must be exactly symmetric"


@requires_gpu
def test_cma_update_c_odd_dim_tail():
    """d not a multiple of the 64-tile: boundary tiles guarded."""
    from evotorch_amd.ops import cma_update_c_

    torch.manual_seed(9)
    d, lam = 130, 20
    C0 = torch.eye(d) * 2.0
    y = torch.randn(lam, d)
    w = torch.rand(lam) / lam
    pc = torch.randn(d)
    hs = torch.tensor(1.0)
    C_cpu = C0.clone()
    cma_update_c_(C_cpu, y, w, pc, hs, c1=0.1, cmu=0.2, cc=0.3)
    C_gpu = C0.clone().cuda()
    cma_update_c_(C_gpu, y.cuda(), w.cuda(), pc.cuda(), hs.cuda(), c1=0.1, cmu=0.2, cc=0.3)
    assert torch.allclose(C_gpu.cpu(), C_cpu, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_fused_rank_matches_eager():
    """K2 fused bitonic ranking vs the eager reference rankers."""
    import evotorch_amd._C as C
    from evotorch_amd.utils import ranking

    torch.manual_seed(12)
    for n in (2, 5, 100, 1000, 4000, 8192):
        fit = torch.randn(n, device="cuda")
        for mi, method in enumerate(("centered", "linear", "nes")):
            for hib in (True, False):
                got = C.fused_rank(fit, mi, hib)
                ref = ranking.rankers[method](fit.cpu(), higher_is_better=hib)
                assert torch.allclose(got.cpu(), ref.to(torch.float32), atol=1e-5), (n, method, hib)


@requires_gpu
def test_rank_dispatch_uses_fused_kernel():
    from evotorch_amd.utils import ranking

    fit = torch.randn(4000, device="cuda")
    out = ranking.rank(fit, "centered", higher_is_better=True)
    ref = ranking.centered(fit.cpu(), higher_is_better=True)
    assert torch.allclose(out.cpu(), ref.to(torch.float32), atol=1e-5)
    assert float(out.sum().abs()) < 1e-3  # centered utilities sum to ~0


@requires_gpu
def test_mapelites_assign_matches_eager():
    """K9 streamed cell assignment vs the eager (C, N) broadcast."""
    import evotorch_amd._C as C

    torch.manual_seed(3)
    cells, n, f = 64, 500, 3
    centers = torch.rand(cells, f, device="cuda")
    grid = torch.stack([centers - 0.15, centers + 0.15], dim=-1)
    feats = torch.rand(n, f, device="cuda")
    utils = torch.randn(n, device="cuda")
    best, valid = C.mapelites_assign(grid, feats, utils)
    inside = ((feats.unsqueeze(0) >= grid[:, :, 0].unsqueeze(1)) & (feats.unsqueeze(0) <= grid[:, :, 1].unsqueeze(1))).all(-1)
    masked = torch.where(inside, utils.unsqueeze(0), torch.full_like(utils, float("-inf")).unsqueeze(0).expand_as(inside))
    ref_best = masked.argmax(dim=1)
    ref_valid = inside.any(dim=1)
    assert torch.equal(valid, ref_valid)
    assert torch.equal(best[ref_valid], ref_best[ref_valid])


@requires_gpu
def test_potrf_tile_matches_torch():
    """K5b panel Cholesky vs torch.linalg.cholesky: sizes across the
    range, a strided diagonal-block view, and the non-PD info flag."""
    from evotorch_amd.ops import potrf_tile_

    torch.manual_seed(11)
    info = torch.zeros(1, dtype=torch.int32, device="cuda:0")
    for n in (1, 7, 37, 64, 100, 128):
        A = torch.randn(n, n, device="cuda:0")
        C = A @ A.T + n * torch.eye(n, device="cuda:0")
        ref = torch.linalg.cholesky(C)
        out = C.clone()
        info.zero_()
        potrf_tile_(out, info)
        assert int(info.item()) == 0
        torch.testing.assert_close(torch.tril(out), ref, rtol=3e-5, atol=3e-5)
    # strided view: a diagonal block inside a larger matrix
    M = torch.randn(200, 200, device="cuda:0")
    M = M @ M.T + 200 * torch.eye(200, device="cuda:0")
    blk = M[64:192, 64:192]
    ref = torch.linalg.cholesky(blk.clone())
    info.zero_()
    potrf_tile_(blk, info)
    assert int(info.item()) == 0
    torch.testing.assert_close(torch.tril(blk), ref, rtol=3e-5, atol=3e-5)
    # non-positive pivot reports its column in info
    B = torch.eye(16, device="cuda:0")
    B[5, 5] = -1.0
    info.zero_()
    potrf_tile_(B, info)
    assert int(info.item()) == 6


@requires_gpu
def test_blocked_cholesky_gpu_matches_torch():
    from evotorch_amd.algorithms.cmaes import CMAES

    torch.manual_seed(12)
    n = 1500  # not a multiple of the 512 block or the 128 panel
    A = torch.randn(n, n, device="cuda:0")
    C = A @ A.T + n * torch.eye(n, device="cuda:0")
    L1 = CMAES._blocked_cholesky(C)
    L2 = torch.linalg.cholesky(C)
    torch.testing.assert_close(L1, L2, rtol=1e-3, atol=1e-3)
    with pytest.raises(RuntimeError):
        CMAES._blocked_cholesky(-torch.eye(700, device="cuda:0"))
