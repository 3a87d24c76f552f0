"""Executable versions of the code blocks in docs/quickstart.md,
docs/problems_guide.md, docs/neuroevolution_guide.md and docs/extending.md.

Each test mirrors a guide's snippets (condensed to test-sized runtimes) so
the documented API surface stays accurate: if a guide example would break,
the matching test here breaks first.
"""

import os
import tempfile

import torch

from evotorch_amd import Problem
from evotorch_amd.decorators import vectorized


@vectorized
def _sphere(x):
    return (x ** 2).sum(-1)


def _sphere_problem(d=10):
    return Problem("min", _sphere, solution_length=d, initial_bounds=(-1, 1))


def test_quickstart_blocks():
    from evotorch_amd.algorithms import SNES, PGPE
    from evotorch_amd.logging import StdOutLogger, PicklingLogger
    from evotorch_amd.neuroevolution import GymNE

    @vectorized
    def rastrigin(x):
        return 10.0 * x.shape[-1] + (x ** 2 - 10.0 * torch.cos(2 * torch.pi * x)).sum(-1)

    problem = Problem(
        "min", rastrigin, solution_length=20, initial_bounds=(-5.12, 5.12), dtype=torch.float32
    )
    searcher = SNES(problem, popsize=50, stdev_init=5.0)
    StdOutLogger(searcher, interval=1000)  # subscribed but silent at this run length
    searcher.run(5)
    assert "best_eval" in searcher.status

    gym_prob = GymNE(
        "CartPole-v1", "Linear(obs_length, act_length)",
        observation_normalization=True, num_episodes=1,
    )
    pgpe = PGPE(
        gym_prob, popsize=8,
        center_learning_rate=0.05, stdev_learning_rate=0.1,
        radius_init=0.3, optimizer="clipup",
    )
    pgpe.run(1)
    policy = gym_prob.to_policy(pgpe.status["center"])
    assert policy(torch.randn(4)).shape == (2,)

    with tempfile.TemporaryDirectory() as d:
        PicklingLogger(pgpe, interval=1, directory=d, verbose=False)
        pgpe.run(1)
        assert len(os.listdir(d)) == 1
    pgpe.load_state_dict(pgpe.state_dict())


def test_problems_guide_blocks():
    from evotorch_amd.core import Problem as CoreProblem, SolutionBatch
    from evotorch_amd.tools import violation

    class Sphere(CoreProblem):
        def __init__(self, d=30):
            super().__init__("min", solution_length=d, initial_bounds=(-1, 1))

        def _evaluate_batch(self, batch: SolutionBatch):
            batch.set_evals((batch.values ** 2).sum(-1))

    p = Sphere()
    batch = p.generate_batch(40)
    p.evaluate(batch)
    assert len(batch.take_best(10)) == 10
    assert batch.utility().shape == (40,)

    seen = []
    p.before_eval_hook.append(lambda b: seen.append(len(b)))
    p.evaluate(p.generate_batch(4))
    assert seen == [4]

    assert float(violation(torch.tensor(3.0), "<=", 1.0)) == 2.0

    ev = _sphere_problem().make_callable_evaluator()
    assert ev(torch.zeros(5, 10)).shape == (5,)


def test_neuroevolution_guide_blocks():
    from evotorch_amd.models import MultiLayered, ensure_stateful, str_to_net
    from evotorch_amd.neuroevolution import (
        NEProblem, SupervisedNE, SyntheticTorchEnv, VecEnvNE, load_policy, save_policy,
    )
    from evotorch_amd.algorithms import PGPE

    net = str_to_net("Linear(obs_length, 64) >> Tanh() >> Linear(64, act_length)", obs_length=6, act_length=2)
    assert isinstance(net, MultiLayered)
    rnn = ensure_stateful(str_to_net("RNN(obs_length, 8) >> Linear(8, act_length)", obs_length=6, act_length=2))
    rnn.reset()
    assert rnn(torch.randn(6)).shape == (2,)

    ne = NEProblem("max", lambda: torch.nn.Linear(4, 2), network_eval_func=lambda n: n(torch.ones(4)).sum())
    assert isinstance(ne.parameterize_net(ne.generate_batch(1).values[0]), torch.nn.Linear)

    X = torch.randn(64, 8)
    ds = torch.utils.data.TensorDataset(X, X.sum(-1, keepdim=True))
    sup = SupervisedNE(ds, "Linear(8, 1)", loss_func=torch.nn.functional.mse_loss, minibatch_size=32)
    sup.evaluate(sup.generate_batch(4))

    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=10, obs_dim=6, act_dim=2, rank=4)

    vec = VecEnvNE(env_factory, "Linear(obs_length, act_length)", observation_normalization=True)
    vb = vec.generate_batch(8)
    vec.evaluate(vb)
    assert vec.status["total_interaction_count"] == 80

    searcher = PGPE(
        vec, popsize=8, center_learning_rate=0.05, stdev_learning_rate=0.1,
        stdev_init=0.1, num_interactions=100,
    )
    searcher.step()
    # adaptive popsize keeps evaluating until >=100 interactions happened
    assert vec.status["total_interaction_count"] >= 100 + 80

    with tempfile.TemporaryDirectory() as d:
        pol = vec.to_policy(vb[0].values.as_subclass(torch.Tensor))
        path = save_policy(pol, os.path.join(d, "pol"), metadata={"gens": 5})
        fresh = vec.to_policy(vb[1].values.as_subclass(torch.Tensor))
        load_policy(path, fresh)
        a = pol(torch.zeros(6))
        b = fresh(torch.zeros(6))
        assert torch.allclose(a, b)


def test_extending_blocks():
    from evotorch_amd.algorithms import GeneticAlgorithm
    from evotorch_amd.algorithms.searchalgorithm import SearchAlgorithm, SinglePopulationAlgorithmMixin
    from evotorch_amd.logging import Logger
    from evotorch_amd.operators import CopyingOperator, GaussianMutation

    problem = _sphere_problem()

    class RandomSearch(SearchAlgorithm, SinglePopulationAlgorithmMixin):
        def __init__(self, problem, *, popsize):
            SearchAlgorithm.__init__(self, problem)
            self._popsize = popsize
            self._population = problem.generate_batch(popsize)
            SinglePopulationAlgorithmMixin.__init__(self)

        @property
        def population(self):
            return self._population

        def _step(self):
            self._population = self.problem.generate_batch(self._popsize)
            self.problem.evaluate(self._population)

    class BestTracker(Logger):
        def __init__(self, searcher, **kwargs):
            super().__init__(searcher, **kwargs)
            self.history = []

        def _log(self, status):
            self.history.append(status["mean_eval"])

    rs = RandomSearch(problem, popsize=16)
    tracker = BestTracker(rs)
    rs.run(3)
    assert len(tracker.history) == 3

    class ResetWorstGene(CopyingOperator):
        def _do(self, batch):
            result = batch.take(torch.arange(len(batch)))
            data = result.access_values()
            idx = data.abs().argmax(dim=-1, keepdim=True)
            data.scatter_(-1, idx, 0.0)
            data.copy_(self._respect_bounds(data))
            return result

    op = ResetWorstGene(problem)
    kids = op(problem.generate_batch(6))
    assert (kids.values == 0).any(dim=-1).all()

    ga = GeneticAlgorithm(
        problem, popsize=16, operators=[ResetWorstGene(problem), GaussianMutation(problem, stdev=0.1)]
    )
    ga.run(2)
    assert "pop_best_eval" in ga.status
