"""End-to-end searcher smoke tests over sphere/rastrigin (mirrors the
reference's tests/test_examples.py strategy: few generations, status-key
assertions, convergence direction)."""

import math

import pytest
import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import CEM, PGPE, SNES, XNES
from evotorch_amd.decorators import vectorized
from evotorch_amd.logging import PandasLogger, PicklingLogger, StdOutLogger


@vectorized
def sphere(x):
    return (x**2).sum(-1)


@vectorized
def rastrigin(x):
    return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)


def make_problem(length=12, seed=10):
    return Problem("min", sphere, solution_length=length, initial_bounds=(-3.0, 3.0), seed=seed)


SEARCHERS = {
    "SNES": lambda p: SNES(p, stdev_init=2.0),
    "XNES": lambda p: XNES(p, stdev_init=2.0),
    "PGPE": lambda p: PGPE(p, popsize=50, center_learning_rate=0.2, stdev_learning_rate=0.1, stdev_init=2.0),
    "PGPE_nonsym": lambda p: PGPE(p, popsize=50, center_learning_rate=0.2, stdev_learning_rate=0.1, stdev_init=2.0, symmetric=False),
    "PGPE_adam": lambda p: PGPE(p, popsize=50, center_learning_rate=0.05, stdev_learning_rate=0.1, stdev_init=2.0, optimizer="adam"),
    "CEM": lambda p: CEM(p, popsize=100, parenthood_ratio=0.25, stdev_init=2.0),
}


@pytest.mark.parametrize("name", sorted(SEARCHERS))
def test_searcher_basic_run(name):
    prob = make_problem()
    searcher = SEARCHERS[name](prob)
    searcher.run(3)
    status = searcher.status
    assert status["iter"] == 3
    for key in ("center", "stdev", "mean_eval", "pop_best", "pop_best_eval", "best_eval"):
        assert key in status, f"missing status key {key}"
    assert searcher.step_count == 3


@pytest.mark.parametrize("name", ["SNES", "PGPE", "CEM"])
def test_searcher_converges_on_sphere(name):
    prob = make_problem(seed=123)
    searcher = SEARCHERS[name](prob)
    searcher.step()
    first = searcher.status["mean_eval"]
    searcher.run(60)
    last = searcher.status["mean_eval"]
    assert last < first * 0.5, f"{name} did not descend: {first} -> {last}"


def test_pgpe_stdev_max_change_respected():
    prob = make_problem()
    searcher = PGPE(prob, popsize=20, center_learning_rate=0.5, stdev_learning_rate=5.0, stdev_init=1.0, stdev_max_change=0.2)
    sigma_before = searcher.status["stdev"].clone()
    searcher.step()
    searcher.step()
    sigma_after = torch.Tensor.as_subclass(searcher.status["stdev"], torch.Tensor)
    ratio = sigma_after / torch.Tensor.as_subclass(sigma_before, torch.Tensor)
    assert torch.all(ratio <= 1.2 * 1.2 + 1e-5)
    assert torch.all(ratio >= 0.8 * 0.8 - 1e-5)


def test_multiobjective_rejected_by_gaussian():
    prob = Problem(["min", "max"], vectorized=True, solution_length=4, initial_bounds=(-1, 1),
                   objective_func=lambda x: torch.stack([x.sum(-1), x.prod(-1)], dim=-1))
    with pytest.raises(ValueError):
        SNES(prob, stdev_init=1.0)  # multi-objective: obj_index must be picked... single-obj check


def test_bounded_problem_rejected():
    prob = Problem("min", sphere, solution_length=4, bounds=(-1.0, 1.0))
    with pytest.raises(ValueError):
        SNES(prob, stdev_init=1.0)


def test_stdout_and_pandas_logger(capsys):
    prob = make_problem()
    searcher = SNES(prob, stdev_init=1.0)
    StdOutLogger(searcher, interval=2)
    pl = PandasLogger(searcher)
    searcher.run(4)
    out = capsys.readouterr().out
    assert "mean_eval" in out
    df = pl.to_dataframe()
    assert len(df) == 4
    assert "mean_eval" in df.columns


def test_pickling_logger(tmp_path):
    prob = make_problem()
    searcher = SNES(prob, stdev_init=1.0)
    logger = PicklingLogger(searcher, interval=2, directory=str(tmp_path), verbose=False)
    searcher.run(4)
    assert logger.last_file_name is not None
    payload = logger.unpickle_last_file()
    assert "center" in payload
    assert payload["center"].shape == (12,)


def test_hooks_on_searcher():
    prob = make_problem()
    searcher = SNES(prob, stdev_init=1.0)
    events = []
    searcher.before_step_hook.append(lambda: events.append("before"))
    searcher.after_step_hook.append(lambda: {"custom": 1})
    searcher.run(2)
    assert events == ["before", "before"]
    assert searcher.status["custom"] == 1


def test_num_interactions_adaptive_popsize():
    class CountingProblem(Problem):
        def __init__(self):
            super().__init__("min", solution_length=4, initial_bounds=(-1, 1), vectorized=True, objective_func=sphere)
            self.last_eval_interaction_count = 0

        def evaluate(self, batch):
            self.last_eval_interaction_count = len(batch) * 3  # 3 "interactions" per solution
            super().evaluate(batch)

    prob = CountingProblem()
    searcher = PGPE(prob, popsize=10, center_learning_rate=0.1, stdev_learning_rate=0.1, stdev_init=1.0,
                    num_interactions=100, popsize_max=100)
    searcher.step()
    # 10 solutions/batch * 3 = 30 interactions; need >= 100 → 4 batches = 40 solutions
    assert len(searcher.population) == 40


def test_state_dict_resume_pgpe():
    import pickle

    def fresh():
        prob = make_problem(seed=77)
        return PGPE(prob, popsize=40, center_learning_rate=0.2, stdev_learning_rate=0.1,
                    stdev_init=2.0, optimizer="clipup")

    a = fresh()
    a.run(10)
    blob = pickle.dumps(a.state_dict())

    b = fresh()
    b.load_state_dict(pickle.loads(blob))
    assert b.step_count == 10
    assert torch.allclose(
        torch.Tensor.as_subclass(b.status["center"], torch.Tensor),
        torch.Tensor.as_subclass(a.status["center"], torch.Tensor),
    )
    b.run(5)  # resumes cleanly
    assert b.step_count == 15


def test_state_dict_resume_cmaes():
    from evotorch_amd.algorithms import CMAES

    prob = make_problem(seed=78)
    a = CMAES(prob, stdev_init=1.0)
    a.run(10)
    sd = a.state_dict()
    prob2 = make_problem(seed=78)
    b = CMAES(prob2, stdev_init=1.0)
    b.load_state_dict(sd)
    assert float(b.status["sigma"]) == pytest.approx(float(a.status["sigma"]))


def test_state_dict_resume_ga():
    from evotorch_amd.algorithms import GeneticAlgorithm
    from evotorch_amd.operators import GaussianMutation, OnePointCrossOver

    prob = make_problem(seed=79)
    ga = GeneticAlgorithm(prob, popsize=20, operators=[
        OnePointCrossOver(prob, tournament_size=2), GaussianMutation(prob, stdev=0.2)])
    ga.run(5)
    sd = ga.state_dict()
    prob2 = make_problem(seed=79)
    ga2 = GeneticAlgorithm(prob2, popsize=20, operators=[
        OnePointCrossOver(prob2, tournament_size=2), GaussianMutation(prob2, stdev=0.2)])
    ga2.load_state_dict(sd)
    assert torch.allclose(ga2.population.unsafe_values, ga.population.unsafe_values)
    ga2.run(3)


def test_state_dict_includes_obs_norm():
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    prob = SyntheticRolloutProblem(seed=80, episode_length=5)
    searcher = PGPE(prob, popsize=16, center_learning_rate=0.1, stdev_learning_rate=0.1,
                    stdev_init=0.1, distributed=True)
    searcher.run(2)
    sd = searcher.state_dict()
    assert sd["obs_norm"]["count"] > 0
    prob2 = SyntheticRolloutProblem(seed=81, episode_length=5)
    s2 = PGPE(prob2, popsize=16, center_learning_rate=0.1, stdev_learning_rate=0.1,
              stdev_init=0.1, distributed=True)
    s2.load_state_dict(sd)
    assert prob2.obs_norm.count == prob.obs_norm.count


def test_pgpe_with_torch_optimizer():
    """Any torch.optim optimizer plugs in through TorchOptimizer
    (reference optimizers.py TorchOptimizer wrapper)."""
    from functools import partial

    from evotorch_amd.optimizers import TorchOptimizer

    prob = make_problem(seed=101)
    searcher = PGPE(
        prob, popsize=50, center_learning_rate=0.2, stdev_learning_rate=0.1, stdev_init=2.0,
        optimizer=partial(TorchOptimizer, torch.optim.RMSprop),
    )
    searcher.step()
    first = searcher.status["mean_eval"]
    searcher.run(40)
    assert searcher.status["mean_eval"] < first * 0.5


def test_num_interactions_adaptive_popsize_distributed():
    """distributed=True honors num_interactions: extra sub-batches are drawn
    until the interaction budget is met (reference core.py:3239-3274)."""

    class CountingProblem(Problem):
        def __init__(self):
            super().__init__("min", solution_length=4, initial_bounds=(-1, 1), vectorized=True, objective_func=sphere)
            self.last_eval_interaction_count = 0

        def evaluate(self, batch):
            self.last_eval_interaction_count = len(batch) * 3
            super().evaluate(batch)

    prob = CountingProblem()
    searcher = PGPE(prob, popsize=10, center_learning_rate=0.1, stdev_learning_rate=0.1, stdev_init=1.0,
                    num_interactions=100, popsize_max=100, distributed=True)
    searcher.step()
    assert searcher.status["num_solutions"] == 40  # 4 sub-batches of 10


def test_xnes_rotation_invariance_on_ellipsoid():
    """XNES (full covariance, exponential map) adapts to a rotated
    ill-conditioned ellipsoid — separable SNES cannot."""
    d = 8
    cond = torch.logspace(0, 2.5, d)
    q, _ = torch.linalg.qr(torch.randn(d, d, generator=torch.Generator().manual_seed(4)))

    @vectorized
    def rotated_ellipsoid(x):
        y = x @ q.T
        return (cond * y**2).sum(-1)

    prob = Problem("min", rotated_ellipsoid, solution_length=d, initial_bounds=(-3, 3), seed=9)
    searcher = XNES(prob, stdev_init=2.0, popsize=32)
    searcher.run(500)
    assert float(searcher.status["pop_best_eval"]) < 1e-2


def test_random_config_chaos():
    """Randomized searcher configurations (seeded) all construct and step
    without crashing — guards kwarg plumbing across the whole family."""
    import random

    from evotorch_amd.algorithms import CEM, CMAES, XNES

    random.seed(0)
    for trial in range(25):
        length = random.choice([2, 3, 7, 16, 33])
        prob = Problem("min" if random.random() < 0.7 else "max", sphere,
                       solution_length=length, initial_bounds=(-2, 2), seed=trial)
        kind = random.choice(["pgpe", "snes", "cem", "xnes", "cmaes"])
        if kind == "pgpe":
            searcher = PGPE(prob, popsize=random.choice([4, 10, 25]) * 2,
                            center_learning_rate=random.uniform(0.01, 0.5),
                            stdev_learning_rate=random.uniform(0.01, 0.3),
                            stdev_init=random.uniform(0.1, 3.0),
                            symmetric=random.random() < 0.8,
                            optimizer=random.choice(["clipup", "adam", "sgd", None]),
                            ranking_method=random.choice(["centered", "linear", "nes", "raw", None]),
                            distributed=random.random() < 0.4)
        elif kind == "snes":
            searcher = SNES(prob, stdev_init=random.uniform(0.1, 3.0),
                            popsize=random.choice([None, 8, 30]),
                            distributed=random.random() < 0.4)
        elif kind == "cem":
            searcher = CEM(prob, popsize=random.choice([20, 50]),
                           parenthood_ratio=random.uniform(0.1, 0.5),
                           stdev_init=random.uniform(0.5, 3.0),
                           stdev_max_change=random.choice([None, 0.2, 1.0]))
        elif kind == "xnes":
            searcher = XNES(prob, stdev_init=random.uniform(0.5, 2.0), popsize=random.choice([None, 12]))
        else:
            searcher = CMAES(prob, stdev_init=random.uniform(0.5, 2.0),
                             popsize=random.choice([None, 8, 24]),
                             separable=random.random() < 0.3,
                             active=random.random() < 0.7)
        searcher.run(3)
        assert searcher.step_count == 3
        float(searcher.status["mean_eval"])
