"""PGPE + ClipUp over the synthetic Humanoid-shaped environment — the
flagship configuration (mirrors the reference's brax humanoid PGPE
notebook, offline). On a GPU the whole generation runs as one fused HIP
kernel; checkpoints (center policy + obs-norm data) are pickled every 50
generations."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse

import torch

from evotorch_amd.algorithms import PGPE
from evotorch_amd.logging import PicklingLogger, StdOutLogger
from evotorch_amd.neuroevolution import SyntheticRolloutProblem


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    p.add_argument("--generations", type=int, default=200)
    p.add_argument("--popsize", type=int, default=4000)
    p.add_argument("--policy", choices=["linear", "mlp64"], default="linear",
                   help="mlp64 = the reference paper's MLP-64-tanh brax architecture (m7 kernel)")
    args = p.parse_args()

    problem = SyntheticRolloutProblem(device=args.device, seed=1, episode_length=200,
                                      policy_hidden=64 if args.policy == "mlp64" else 0)
    radius = 2.25
    max_speed = radius / 15.0
    searcher = PGPE(
        problem,
        popsize=args.popsize,
        radius_init=radius,
        center_learning_rate=0.75 * max_speed,
        stdev_learning_rate=0.1,
        optimizer="clipup",
        optimizer_config={"max_speed": max_speed},
        ranking_method="centered",
        distributed=True,
    )
    StdOutLogger(searcher, interval=10)
    PicklingLogger(searcher, interval=50, directory="checkpoints", verbose=False)
    searcher.run(args.generations)
    policy = problem.to_policy(torch.Tensor.as_subclass(searcher.status["center"], torch.Tensor))
    print(policy)


if __name__ == "__main__":
    main()
