"""Round-2 profiling driver: CMA-ES d=4096 and NSGA-II popsize-32k
per-kernel attribution."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import sys
import time

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import CMAES, GeneticAlgorithm
from evotorch_amd.decorators import vectorized
from evotorch_amd.operators import PolynomialMutation, SimulatedBinaryCrossOver


def time_block(label, fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{label}: {dt*1000:.3f} ms/iter")
    return dt


def main():
    which = sys.argv[1] if len(sys.argv) > 1 else "both"
    if which in ("cma", "both"):
        @vectorized
        def sphere(x):
            return (x**2).sum(-1)

        prob = Problem("min", sphere, solution_length=4096, initial_bounds=(-1, 1), seed=1, device="cuda:0")
        searcher = CMAES(prob, stdev_init=1.0, popsize=64)
        time_block("cmaes.step (full gen)", searcher.step, iters=50, warmup=10)
        # phase attribution
        z, y, x = searcher.sample_distribution()
        hs = torch.ones((), device="cuda")
        time_block("  sample_distribution", lambda: searcher.sample_distribution(), iters=50)
        time_block("  update_C", lambda: searcher.update_C(z, y, hs), iters=50)
        time_block("  cholesky(C)", lambda: searcher._cholesky(searcher._C), iters=10)
        w = searcher._weights[: searcher._mu]
        time_block("  recomb (y_w,z_w)", lambda: (w @ y[: searcher._mu], w @ z[: searcher._mu]), iters=50)

    if which in ("nsga", "both"):
        @vectorized
        def multi(x):
            f1 = (x**2).sum(-1)
            f2 = ((x - 2.0) ** 2).sum(-1)
            return torch.stack([f1, f2], dim=-1)

        for N in (8192, 32768):
            prob = Problem(["min", "min"], multi, solution_length=64, initial_bounds=(0, 1), bounds=(0.0, 1.0), seed=2, device="cuda:0")
            ga = GeneticAlgorithm(
                prob, popsize=N,
                operators=[
                    SimulatedBinaryCrossOver(prob, tournament_size=4, cross_over_rate=1.0, eta=8),
                    PolynomialMutation(prob, eta=20, mutation_probability=0.2),
                ],
            )
            ga.step()
            time_block(f"nsga2.step N={N}", ga.step, iters=10, warmup=2)
            # attribution
            batch = ga.population
            from evotorch_amd.core import _compute_pareto_ranks
            utils = batch.utils()
            time_block(f"  pareto_ranks+crowd N={N}", lambda: _compute_pareto_ranks(utils), iters=10)
            ext = batch.concat(batch)
            time_block(f"  take_best N={N}", lambda: ext.take_best(N), iters=10)
            ops = ga._operators
            time_block(f"  sbx N={N}", lambda: ops[0](batch), iters=10)
            time_block(f"  polymut N={N}", lambda: ops[1](batch), iters=10)
            time_block(f"  evaluate N={N}", lambda: prob.evaluate(batch), iters=10)


main()
