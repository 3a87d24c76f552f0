"""Rollout-kernel diagnostic sweep: time vs N (occupancy) and vs T
(per-step cost), printed as one table. Run on the GPU box."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import evotorch_amd._C as C
from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec


def time_kernel(spec, params, steps, iters=5):
    O = spec.obs_dim
    mean = torch.zeros(O, device="cuda")
    std = torch.ones(O, device="cuda")
    blob = spec.env_blob(mean, std, device="cuda")
    stats = torch.zeros(2 * O, device="cuda")
    # warmup
    C.rollout_linear(params, blob, stats, O, spec.act_dim, spec.rank, steps,
                     spec.alive_bonus, spec.act_cost, 1, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        C.rollout_linear(params, blob, stats, O, spec.act_dim, spec.rank, steps,
                         spec.alive_bonus, spec.act_cost, 1, 0)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    spec = SyntheticEnvSpec(device="cuda")
    print(f"L={spec.solution_length} O={spec.obs_dim} A={spec.act_dim} R={spec.rank}")
    print("--- N sweep (T=200) ---")
    for n in [256, 512, 1024, 2048, 4000, 8000]:
        params = 0.1 * torch.randn(n, spec.solution_length, device="cuda")
        dt = time_kernel(spec, params, 200)
        print(f"N={n:6d}  {dt*1e3:8.3f} ms   {n*200/dt/1e6:8.2f} M member-steps/s")
    print("--- T sweep (N=4000) ---")
    params = 0.1 * torch.randn(4000, spec.solution_length, device="cuda")
    for t in [1, 10, 50, 200, 400]:
        dt = time_kernel(spec, params, t)
        print(f"T={t:4d}  {dt*1e3:8.3f} ms   per-step: {dt/t*1e6:8.2f} us")


if __name__ == "__main__":
    main()
