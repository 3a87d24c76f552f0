"""Synthetic rollout problem (CPU eager path) + RunningNorm tests."""

import pytest
import torch

from evotorch_amd.neuroevolution import RunningNorm, SyntheticEnvSpec, SyntheticRolloutProblem, rollout_eager


def test_running_norm_matches_batch_stats():
    rn = RunningNorm(shape=4)
    data = torch.randn(100, 4) * 3 + 1
    rn.update(data[:60])
    rn.update(data[60:])
    assert rn.count == 100
    assert torch.allclose(rn.mean, data.mean(0), atol=1e-5)
    expected_var = (data**2).mean(0) - data.mean(0) ** 2
    assert torch.allclose(rn.stdev, expected_var.clamp(min=1e-2).sqrt(), atol=1e-4)


def test_running_norm_merge_equivalence():
    a = RunningNorm(shape=3)
    b = RunningNorm(shape=3)
    merged = RunningNorm(shape=3)
    x1 = torch.randn(10, 3)
    x2 = torch.randn(20, 3)
    a.update(x1)
    b.update(x2)
    merged.update(a)
    merged.update(b)
    direct = RunningNorm(shape=3)
    direct.update(torch.cat([x1, x2]))
    assert torch.allclose(merged.mean, direct.mean, atol=1e-6)
    assert torch.allclose(merged.stdev, direct.stdev, atol=1e-6)


def test_running_norm_triple_roundtrip():
    rn = RunningNorm(shape=2)
    rn.update(torch.randn(5, 2))
    c, s, ss = rn.stats_triple()
    rn2 = RunningNorm(shape=2)
    rn2.update((float(c), s, ss))
    assert torch.allclose(rn.mean, rn2.mean)


def test_philox_ref_shapes():
    from evotorch_amd.neuroevolution.philox_ref import philox_normal_rows, philox_normals

    z = philox_normals(42, 0, 1000)
    assert z.shape == (1000,)
    assert abs(float(z.mean())) < 0.15
    assert abs(float(z.std()) - 1.0) < 0.1
    rows = philox_normal_rows(42, 3, 4, 10)
    assert rows.shape == (4, 10)
    # stream separation: different members differ
    assert not torch.allclose(rows[0], rows[1])
    # offset consistency: member (offset 3 + 1) == member (offset 4 + 0)
    rows2 = philox_normal_rows(42, 4, 1, 10)
    assert torch.equal(rows[1], rows2[0])


def test_rollout_eager_runs_and_is_deterministic():
    spec = SyntheticEnvSpec(episode_length=10)
    params = 0.05 * torch.randn(8, spec.solution_length)
    mean = torch.zeros(spec.obs_dim)
    std = torch.ones(spec.obs_dim)
    f1, (c1, s1, ss1) = rollout_eager(spec, params, mean, std, init_seed=5)
    f2, _ = rollout_eager(spec, params, mean, std, init_seed=5)
    assert torch.equal(f1, f2)
    assert f1.shape == (8,)
    assert c1 == 80.0
    f3, _ = rollout_eager(spec, params, mean, std, init_seed=6)
    assert not torch.equal(f1, f3)


def test_synthetic_problem_end_to_end_cpu():
    prob = SyntheticRolloutProblem(seed=3, episode_length=10)
    batch = prob.generate_batch(6)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    assert prob.last_eval_interaction_count == 60
    assert prob.obs_norm.count == 60.0  # stats merged after eval
    status = prob.status
    assert status["total_interaction_count"] == 60
    assert status["total_episode_count"] == 6


def test_synthetic_problem_to_policy():
    prob = SyntheticRolloutProblem(seed=3, episode_length=5)
    batch = prob.generate_batch(2)
    prob.evaluate(batch)
    x = batch[0].values
    policy = prob.to_policy(torch.Tensor.as_subclass(x, torch.Tensor))
    obs = torch.randn(prob.spec.obs_dim)
    act = policy(obs)
    assert act.shape == (prob.spec.act_dim,)
    assert float(act.abs().max()) <= 1.0


def test_pgpe_on_synthetic_cpu_improves():
    from evotorch_amd.algorithms import PGPE

    prob = SyntheticRolloutProblem(seed=7, episode_length=10)
    searcher = PGPE(prob, popsize=32, radius_init=2.25, center_learning_rate=0.1,
                    stdev_learning_rate=0.1, optimizer="clipup", distributed=True)
    searcher.step()
    first = searcher.status["mean_eval"]
    for _ in range(15):
        searcher.step()
    assert searcher.status["mean_eval"] > first - 1.0  # no collapse; usually improves


def test_mlp_policy_eager_and_to_policy():
    prob = SyntheticRolloutProblem(seed=4, episode_length=5, policy_hidden=64)
    assert prob.solution_length == 64 * 376 + 64 + 17 * 64 + 17
    batch = prob.generate_batch(4)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    policy = prob.to_policy(torch.Tensor.as_subclass(batch[0].values, torch.Tensor))
    act = policy(torch.randn(prob.spec.obs_dim))
    assert act.shape == (prob.spec.act_dim,)
    assert float(act.abs().max()) <= 1.0


def test_pickling_logger_with_policy_export(tmp_path):
    """PicklingLogger(make_policy_from=...) embeds a ready-to-run policy in
    the checkpoint (reference logging.py:128-180)."""
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.logging import PicklingLogger
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    prob = SyntheticRolloutProblem(seed=9, episode_length=5)
    searcher = PGPE(prob, popsize=8, center_learning_rate=0.1,
                    stdev_learning_rate=0.1, stdev_init=0.1)
    logger = PicklingLogger(searcher, interval=2, directory=str(tmp_path),
                            make_policy_from="center", verbose=False)
    searcher.run(2)
    payload = logger.unpickle_last_file()
    policy = payload["policy"]
    obs = torch.randn(prob._spec.obs_dim)
    act = policy(obs)
    assert act.shape == (prob._spec.act_dim,)
    assert bool((act.abs() <= 1.0 + 1e-6).all())
