"""Real-environment neuroevolution without any installed simulator:
PGPE over a linear policy balancing the vendored CartPole
(evotorch_amd.neuroevolution.gym_compat — textbook dynamics, gymnasium
API). With real gymnasium installed, the same code runs against it
unchanged.

Run:  python examples/cartpole_gymne.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd.algorithms import PGPE
from evotorch_amd.logging import StdOutLogger
from evotorch_amd.neuroevolution import GymNE


def main(generations: int = 20):
    torch.manual_seed(0)
    problem = GymNE(
        env="CartPole-v1",
        network="Linear(obs_length, act_length)",
        episode_length=200,
        observation_normalization=True,
        seed=42,
    )
    searcher = PGPE(
        problem,
        popsize=24,
        center_learning_rate=0.4,
        stdev_learning_rate=0.1,
        stdev_init=0.5,
    )
    StdOutLogger(searcher, interval=5)
    searcher.run(generations)
    policy = problem.to_policy(searcher.status["center"])
    score = problem.run(torch.Tensor.as_subclass(searcher.status["center"], torch.Tensor))
    print(f"final mean_eval={searcher.status['mean_eval']:.1f}, center policy episode return={score:.1f}")
    return searcher.status["mean_eval"]


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--generations", type=int, default=20)
    main(ap.parse_args().generations)
