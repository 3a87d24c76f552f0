"""Large-scale distributed ES benchmark ("PGPE
1M-param MLP policy popsize=100k, population sharded across 8 MI355X via
RCCL"): measures the pure ES machinery — philox sampling (K1), fitness
evaluation, global centered ranking (all-gather), fused gradient
reduction (K3) and ClipUp update (K4) — at production scale.

Run:  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
          --master-addr 127.0.0.1 scripts/bench_large_es.py
The fitness function is a quadratic over a fixed random projection
(2 rocBLAS GEMV-shaped passes over the population), standing in for a
1M-parameter policy evaluation without a simulator."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--length", type=int, default=1_000_000)
    p.add_argument("--popsize-per-gpu", type=int, default=12_500)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=2)
    args = p.parse_args()

    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.parallel import init_comm

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    have_gpu = torch.cuda.is_available()
    # one Comm for every world size: world-1 runs the same SPMD path
    # (counter-addressed sampling + side-stream noise pre-generation)
    comm = init_comm()
    device = comm.device

    L = args.length
    g = torch.Generator(device="cpu").manual_seed(7)
    target = torch.randn(L, generator=g).to(device)

    scratch = {}

    @vectorized
    def quadratic(x):
        # reuse one difference buffer: fresh 50 GB temporaries every
        # generation would thrash the caching allocator
        d = scratch.get("d")
        if d is None or d.shape != x.shape:
            d = torch.empty_like(x)
            scratch["d"] = d
        torch.sub(x, target, out=d)
        d.mul_(d)
        return d.sum(-1)

    prob = Problem("min", quadratic, solution_length=L, initial_bounds=(-1, 1), device=device, seed=1 + rank)
    prob.use_comm(comm)
    searcher = PGPE(
        prob,
        popsize=args.popsize_per_gpu * world,
        center_learning_rate=0.3,
        stdev_learning_rate=0.1,
        stdev_init=1.0,
        optimizer="clipup",
        ranking_method="centered",
        distributed=True,
    )

    def sync():
        comm.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        searcher.step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        searcher.step()
    sync()
    dt = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([dt], dtype=torch.float64, device=device if have_gpu else "cpu")
        comm.all_reduce_(t, op="max")
        dt = float(t)
    if rank == 0:
        total_pop = args.popsize_per_gpu * world
        print(json.dumps({
            "metric": "large-ES solutions/sec (1M params)",
            "solution_length": L,
            "global_popsize": total_pop,
            "n_gpus": world,
            "solutions_per_sec": total_pop * args.steps / dt,
            "ms_per_gen": dt / args.steps * 1000,
            "mean_eval": float(searcher.status["mean_eval"]),
        }))


if __name__ == "__main__":
    main()
