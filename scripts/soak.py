"""Long-soak stability run: 2000-gen PGPE MLP-64 + 50k-gen graphed SNES."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import PGPE, SNES, GraphedSearch
from evotorch_amd.decorators import vectorized
from evotorch_amd.neuroevolution import SyntheticRolloutProblem


def main():
    # v7 flagship (linear, T=1000) through the SPMD path (comm world-1 ==
    # the code the driver's scaling run executes), incl. pregen overlap
    from evotorch_amd.parallel.comm import Comm

    comm = Comm(device=torch.device("cuda", 0))
    prob_lin = SyntheticRolloutProblem(device="cuda:0", seed=3, episode_length=1000)
    prob_lin.use_comm(comm)
    r = 2.25
    s_lin = PGPE(prob_lin, popsize=4000, radius_init=r, center_learning_rate=0.75 * r / 15,
                 stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15},
                 distributed=True)
    t0 = time.perf_counter()
    for g in range(1000):
        s_lin.step()
        if (g + 1) % 250 == 0:
            me = float(s_lin.status["mean_eval"])
            assert math.isfinite(me), f"non-finite mean_eval at gen {g+1}"
            print(f"v7 linear T=1000 gen {g+1}: mean={me:.1f}", flush=True)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"PGPE v7 linear 1000-gen soak (SPMD world-1 + overlap): {dt:.1f}s, "
          f"{1000*4000/dt:,.0f} sol/s sustained, final {float(s_lin.status['mean_eval']):.1f}")

    prob = SyntheticRolloutProblem(device="cuda:0", seed=7, episode_length=200, policy_hidden=64)
    r = 2.25
    s = PGPE(prob, popsize=4000, radius_init=r, center_learning_rate=0.75 * r / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15},
             distributed=True)
    t0 = time.perf_counter()
    for g in range(2000):
        s.step()
        if (g + 1) % 500 == 0:
            print(f"mlp64 gen {g+1}: mean={float(s.status['mean_eval']):.1f}", flush=True)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"PGPE mlp64 2000-gen soak: {dt:.1f}s, {2000*4000/dt:,.0f} sol/s sustained, final {float(s.status['mean_eval']):.1f}")
    center = torch.Tensor.as_subclass(s.status["center"], torch.Tensor)
    assert torch.isfinite(center).all(), "non-finite center!"

    @vectorized
    def rastrigin(x):
        return 10 * x.shape[-1] + (x**2 - 10 * torch.cos(2 * math.pi * x)).sum(-1)

    prob2 = Problem("min", rastrigin, solution_length=100, initial_bounds=(-5.12, 5.12), device="cuda:0", seed=2)
    s2 = SNES(prob2, popsize=1000, stdev_init=10.0)
    g2 = GraphedSearch(s2, generations_per_capture=10)
    g2.capture()
    t0 = time.perf_counter()
    g2.run(50000)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"graphed SNES 50k-gen soak: {dt:.1f}s, {50000/dt:,.0f} gens/s, final {g2.mean_eval:.4f}")
    assert g2.mean_eval < 1.0
    print("SOAK OK")


if __name__ == "__main__":
    main()
