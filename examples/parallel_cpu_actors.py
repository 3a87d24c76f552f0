"""One-script CPU parallelism with `num_actors` (the reference's Ray
actor use case, here on a standard-library multiprocessing pool):
Acrobot swing-up rollouts fan out over worker processes, each holding
its own copy of the environment; observation-normalization statistics
and interaction counters merge back every generation.

Run:  python examples/parallel_cpu_actors.py --num-actors 4
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--generations", type=int, default=10)
    ap.add_argument("--num-actors", type=int, default=2)
    ap.add_argument("--popsize", type=int, default=16)
    args = ap.parse_args()

    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.logging import StdOutLogger
    from evotorch_amd.neuroevolution import GymNE

    torch.manual_seed(0)
    problem = GymNE(
        env="Acrobot-v1",
        network="Linear(obs_length, act_length)",
        episode_length=100,
        observation_normalization=True,
        seed=7,
        num_actors=args.num_actors,
    )
    searcher = PGPE(
        problem,
        popsize=args.popsize,
        center_learning_rate=0.3,
        stdev_learning_rate=0.1,
        radius_init=0.7,
    )
    StdOutLogger(searcher, interval=max(1, args.generations // 5))
    try:
        searcher.run(args.generations)
        print(f"best_eval={float(searcher.status['pop_best_eval']):.1f} "
              f"interactions={problem._total_interactions}")
    finally:
        problem.kill_actors()


if __name__ == "__main__":
    main()
