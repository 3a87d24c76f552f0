"""SNES on Rastrigin d=100 — the reference README's first example
(BASELINE.md row 1). Runs on CPU or GPU (--device cuda:0)."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse
import math
import time

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import SNES
from evotorch_amd.decorators import vectorized
from evotorch_amd.logging import PandasLogger, StdOutLogger


@vectorized
def rastrigin(x: torch.Tensor) -> torch.Tensor:
    return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--generations", type=int, default=2000)
    args = p.parse_args()

    problem = Problem("min", rastrigin, solution_length=100, initial_bounds=(-5.12, 5.12),
                      store_solution_stats=True,
                      device=args.device, seed=1)
    searcher = SNES(problem, popsize=1000, stdev_init=10.0)
    StdOutLogger(searcher, interval=max(1, args.generations // 10))
    pandas_logger = PandasLogger(searcher)
    t0 = time.perf_counter()
    searcher.run(args.generations)
    dt = time.perf_counter() - t0
    print(f"{args.generations} generations in {dt:.2f}s = {args.generations/dt:.1f} gens/sec")
    print(pandas_logger.to_dataframe()[["mean_eval", "best_eval"]].tail())


if __name__ == "__main__":
    main()
