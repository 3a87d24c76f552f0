"""CMA-ES d=4096 full-covariance benchmark ("CMA-ES
d=4096 full-covariance rank-mu update (MFMA) on 1 MI355X"): generations/sec
on a synthetic quadratic, with the rank-mu update and sampling running as
rocBLAS GEMMs and the decomposition on rocSOLVER."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--dim", type=int, default=4096)
    p.add_argument("--popsize", type=int, default=64)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--every-gen-decomposition", action="store_true", help="Cholesky every generation (default: reference-style amortized interval)")
    args = p.parse_args()

    from evotorch_amd import Problem
    from evotorch_amd.algorithms import CMAES
    from evotorch_amd.decorators import vectorized

    device = "cuda:0" if torch.cuda.is_available() else "cpu"

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=args.dim, initial_bounds=(-3, 3), device=device, seed=1)
    searcher = CMAES(prob, stdev_init=1.0, popsize=args.popsize, limit_C_decomposition=not args.every_gen_decomposition)
    for _ in range(args.warmup):
        searcher.step()
    if device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        searcher.step()
    if device != "cpu":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "CMA-ES generations/sec",
        "dim": args.dim,
        "popsize": args.popsize,
        "gens_per_sec": args.steps / dt,
        "ms_per_gen": dt / args.steps * 1000,
        "best_eval": searcher.status["pop_best_eval"],
    }))


if __name__ == "__main__":
    main()
