"""Convergence validation battery (GPU): runs each major searcher to
convergence on its reference-style workload and prints a summary table.
The numbers land in profiles/VALIDATION.md as empirical evidence that the
framework optimizes, not just runs."""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import CEM, CMAES, PGPE, SNES, XNES, Cosyne, GeneticAlgorithm, GraphedSearch, MAPElites
from evotorch_amd.decorators import vectorized
from evotorch_amd.neuroevolution import SyntheticRolloutProblem
from evotorch_amd.operators import GaussianMutation, PolynomialMutation, SimulatedBinaryCrossOver

DEVICE = "cuda:0" if torch.cuda.is_available() else "cpu"


@vectorized
def rastrigin(x):
    return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)


@vectorized
def sphere(x):
    return (x**2).sum(-1)


def report(name, gens, dt, metric):
    print(f"{name:<44} {gens:>6} gens {dt:8.2f}s  {gens/dt:9.1f} gens/s  final={metric:.4f}")


def main():
    torch.manual_seed(0)

    # 1. SNES Rastrigin d=100 popsize 1000, full 2000-gen reference run (graphed)
    prob = Problem("min", rastrigin, solution_length=100, initial_bounds=(-5.12, 5.12), device=DEVICE, seed=1)
    s = SNES(prob, popsize=1000, stdev_init=10.0)
    g = GraphedSearch(s, generations_per_capture=10)
    g.capture()
    t0 = time.perf_counter()
    g.run(2000)
    report("SNES Rastrigin d=100 pop=1000 (graphed)", 2000, time.perf_counter() - t0, g.mean_eval)

    # 2. CMA-ES d=512 sphere to low residual
    prob = Problem("min", sphere, solution_length=512, initial_bounds=(-3, 3), device=DEVICE, seed=2)
    s = CMAES(prob, stdev_init=2.0, popsize=64)
    t0 = time.perf_counter()
    s.run(800)
    report("CMA-ES sphere d=512 pop=64", 800, time.perf_counter() - t0, s.status["pop_best_eval"])

    # 3. PGPE + ClipUp synthetic humanoid (fused rollout), 300 gens
    prob = SyntheticRolloutProblem(device=DEVICE, seed=3, episode_length=200)
    radius = 2.25
    s = PGPE(prob, popsize=4000, radius_init=radius, center_learning_rate=0.75 * radius / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": radius / 15},
             ranking_method="centered", distributed=True)
    s.step()
    first = float(s.status["mean_eval"])
    t0 = time.perf_counter()
    for _ in range(300):
        s.step()
    dt = time.perf_counter() - t0
    report("PGPE synthetic-humanoid linear pop=4000", 300, dt, float(s.status["mean_eval"]))
    print(f'    (reward {first:.1f} -> {float(s.status["mean_eval"]):.1f})')

    # 4. PGPE MLP-64 policy, 150 gens
    prob = SyntheticRolloutProblem(device=DEVICE, seed=4, episode_length=200, policy_hidden=64)
    s = PGPE(prob, popsize=4000, radius_init=radius, center_learning_rate=0.75 * radius / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": radius / 15},
             ranking_method="centered", distributed=True)
    s.step()
    first = float(s.status["mean_eval"])
    t0 = time.perf_counter()
    for _ in range(150):
        s.step()
    dt = time.perf_counter() - t0
    report("PGPE synthetic-humanoid MLP-64 pop=4000", 150, dt, float(s.status["mean_eval"]))
    print(f'    (reward {first:.1f} -> {float(s.status["mean_eval"]):.1f})')

    # 5. NSGA-II on ZDT1-style biobjective, front coverage
    @vectorized
    def zdt1(x):
        f1 = x[:, 0]
        gg = 1 + 9 * x[:, 1:].mean(-1)
        f2 = gg * (1 - torch.sqrt(torch.clamp(f1 / gg, min=0)))
        return torch.stack([f1, f2], dim=-1)

    prob = Problem(["min", "min"], zdt1, solution_length=12, initial_bounds=(0.0, 1.0), bounds=(0.0, 1.0),
                   device=DEVICE, seed=5)
    ga = GeneticAlgorithm(prob, popsize=512, operators=[
        SimulatedBinaryCrossOver(prob, tournament_size=2, eta=15.0),
        PolynomialMutation(prob, eta=20.0),
    ])
    t0 = time.perf_counter()
    ga.run(100)
    dt = time.perf_counter() - t0
    ranks, _ = ga.population.compute_pareto_ranks()
    frac = float((ranks == 0).float().mean())
    report("NSGA-II ZDT1 d=12 pop=512 (HIP pareto sort)", 100, dt, frac)
    print(f"    (fraction of population on the non-dominated front: {frac:.2f})")

    # 6. XNES full covariance d=256 (rocBLAS exp-map updates)
    prob = Problem("min", sphere, solution_length=256, initial_bounds=(-3, 3), device=DEVICE, seed=6)
    s = XNES(prob, stdev_init=1.0, popsize=64)
    t0 = time.perf_counter()
    s.run(400)
    report("XNES sphere d=256 pop=64", 400, time.perf_counter() - t0, s.status["pop_best_eval"])

    # 7. CEM d=1000
    prob = Problem("min", sphere, solution_length=1000, initial_bounds=(-3, 3), device=DEVICE, seed=7)
    s = CEM(prob, stdev_init=2.0, popsize=500, parenthood_ratio=0.2)
    t0 = time.perf_counter()
    s.run(300)
    report("CEM sphere d=1000 pop=500", 300, time.perf_counter() - t0, s.status["pop_best_eval"])

    # 8. Cosyne Rastrigin d=30
    prob = Problem("min", rastrigin, solution_length=30, initial_bounds=(-5.12, 5.12), device=DEVICE, seed=8)
    s = Cosyne(prob, popsize=128, tournament_size=4, mutation_stdev=0.3)
    t0 = time.perf_counter()
    s.run(300)
    report("Cosyne Rastrigin d=30 pop=128", 300, time.perf_counter() - t0, s.status["pop_best_eval"])

    # 9. MAPElites on Rastrigin with 2 descriptor dims (K9 assignment kernel)
    from evotorch_amd.algorithms import make_feature_grid

    @vectorized
    def rastrigin_with_feats(x):
        f = 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)
        return torch.stack([f, x[:, 0], x[:, 1]], dim=-1)

    prob = Problem("min", rastrigin_with_feats, solution_length=16, initial_bounds=(-5.12, 5.12),
                   device=DEVICE, seed=9, eval_data_length=2)
    grid = make_feature_grid(lower_bounds=[-5.12, -5.12], upper_bounds=[5.12, 5.12], num_bins=20, device=DEVICE)
    s = MAPElites(prob, feature_grid=grid, re_evaluate=False,
                  operators=[GaussianMutation(prob, stdev=0.3)])
    t0 = time.perf_counter()
    s.run(200)
    dt = time.perf_counter() - t0
    filled = float(s.filled.float().mean())
    report("MAPElites Rastrigin 20x20 grid", 200, dt, filled)
    print(f"    (fraction of grid cells filled: {filled:.2f})")


if __name__ == "__main__":
    main()
