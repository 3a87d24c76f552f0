"""Checkpoint/resume: run PGPE for 30 generations, save a checkpoint,
restore it into a freshly constructed searcher, and continue — the resumed
run picks up exactly where the original left off.

Run: python examples/checkpoint_resume.py
"""

import math
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import PGPE
from evotorch_amd.decorators import vectorized


@vectorized
def rastrigin(x):
    return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)


def make_searcher():
    problem = Problem("min", rastrigin, solution_length=30, initial_bounds=(-5.12, 5.12), seed=42)
    return PGPE(problem, popsize=200, center_learning_rate=0.5, stdev_learning_rate=0.1,
                stdev_init=5.0, optimizer="clipup")


def main():
    searcher = make_searcher()
    searcher.run(30)
    print(f"after 30 gens: mean_eval={float(searcher.status['mean_eval']):.3f}")

    path = os.path.join(tempfile.gettempdir(), "pgpe_ckpt.pt")
    torch.save(searcher.state_dict(), path)
    print(f"checkpoint saved to {path}")

    resumed = make_searcher()
    resumed.load_state_dict(torch.load(path, weights_only=False))
    assert resumed.step_count == 30
    resumed.run(70)
    print(f"after resume to gen 100: mean_eval={float(resumed.status['mean_eval']):.3f}")
    assert float(resumed.status["mean_eval"]) < float(searcher.status["mean_eval"])
    print("resume OK")


if __name__ == "__main__":
    main()
