"""Upper-scale probes: CMA-ES at d=8192 (rocSOLVER Cholesky amortization)
and NSGA-II at popsize 65536 (K7 compacted peel)."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import CMAES, GeneticAlgorithm
from evotorch_amd.decorators import vectorized
from evotorch_amd.operators import PolynomialMutation, SimulatedBinaryCrossOver


@vectorized
def sphere(x):
    return (x**2).sum(-1)


def main():
    # CMA-ES d=8192 full covariance
    prob = Problem("min", sphere, solution_length=8192, initial_bounds=(-3, 3), device="cuda:0", seed=1)
    s = CMAES(prob, stdev_init=2.0, popsize=64)
    for _ in range(3):
        s.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        s.step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 50
    print(f"CMA-ES d=8192 pop=64 full-cov: {dt*1000:.1f} ms/gen ({1/dt:.1f} gens/s), "
          f"decompose interval {s._decompose_interval}")

    # NSGA-II popsize 65536
    @vectorized
    def two_obj(x):
        f1 = x[:, 0]
        g = 1 + 9 * x[:, 1:].mean(dim=-1)
        return torch.stack([f1, g * (1 - torch.sqrt((f1 / g).clamp(min=0)))], dim=-1)

    prob = Problem(["min", "min"], two_obj, solution_length=12, bounds=(0.0, 1.0),
                   initial_bounds=(0.0, 1.0), device="cuda:0", seed=2)
    ga = GeneticAlgorithm(prob, popsize=65536, operators=[
        SimulatedBinaryCrossOver(prob, eta=15, tournament_size=2),
        PolynomialMutation(prob, eta=20, mutation_probability=1.0 / 12)])
    ga.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        ga.step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    print(f"NSGA-II popsize 65536: {dt*1000:.1f} ms/gen ({1/dt:.1f} gens/s)")


if __name__ == "__main__":
    main()
