"""SNES Rastrigin d=100 popsize=1000: eager vs
hipGraph-captured generations/sec on one MI355X."""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import SNES, GraphedSearch
from evotorch_amd.decorators import vectorized


@vectorized
def rastrigin(x):
    return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)


def make(seed):
    prob = Problem("min", rastrigin, solution_length=100, initial_bounds=(-5.12, 5.12),
                   device="cuda:0", seed=seed)
    return SNES(prob, popsize=1000, stdev_init=10.0)


def main():
    gens = 2000
    # eager
    s = make(1)
    for _ in range(10):
        s.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(gens):
        s.step()
    torch.cuda.synchronize()
    eager_dt = time.perf_counter() - t0
    eager_mean = s.status["mean_eval"]
    # graphed (10 generations per captured graph)
    s2 = make(1)
    graphed = GraphedSearch(s2, generations_per_capture=10)
    graphed.capture()
    graphed.run(10)
    t0 = time.perf_counter()
    graphed.run(gens)
    graphed_dt = time.perf_counter() - t0
    print(f"eager:   {gens/eager_dt:9.1f} gens/sec  ({eager_dt/gens*1e3:.3f} ms/gen)  mean_eval={eager_mean:.1f}")
    print(f"graphed: {gens/graphed_dt:9.1f} gens/sec  ({graphed_dt/gens*1e3:.3f} ms/gen)  mean_eval={graphed.mean_eval:.1f}")
    print(f"speedup: {eager_dt/graphed_dt:.1f}x")


if __name__ == "__main__":
    main()
