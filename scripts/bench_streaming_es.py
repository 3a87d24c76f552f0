"""Streaming large-L ES benchmark: PGPE over a quadratic at L where the
materialized population would not be practical. Population memory is
O(chunk_rows x L); noise is regenerated in pass 2 from the counter-addressed
philox stream. Usage: python scripts/bench_streaming_es.py [L] [popsize] [chunk]."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import PGPE
from evotorch_amd.decorators import vectorized


def main():
    L = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000_000
    popsize = int(sys.argv[2]) if len(sys.argv) > 2 else 64
    chunk = int(sys.argv[3]) if len(sys.argv) > 3 else 8
    device = "cuda:0" if torch.cuda.is_available() else "cpu"

    target = None

    @vectorized
    def quad(x):
        nonlocal target
        if target is None or target.shape[-1] != x.shape[-1]:
            g = torch.Generator(device=x.device).manual_seed(7)
            target = torch.empty(x.shape[-1], device=x.device).uniform_(-0.05, 0.05, generator=g)
        d = x - target
        sq = d * d
        n, length = sq.shape
        block = 65536
        if length % block == 0:
            # two-stage reduction: n*(L/block) partial rows give the GPU
            # reducer enough parallelism (a single (n, 1e8) rowwise sum
            # drops to ~300 GB/s; this stays bandwidth-bound)
            return sq.view(n, length // block, block).sum(dim=2).sum(dim=1)
        return sq.sum(dim=-1)

    prob = Problem("min", quad, solution_length=L, initial_bounds=(-0.1, 0.1), seed=1, device=device)
    searcher = PGPE(prob, popsize=popsize, center_learning_rate=0.02, stdev_learning_rate=0.05,
                    stdev_init=0.1, distributed=True, grad_chunk_rows=chunk)

    searcher.step()  # warmup (allocations, first philox)
    if device != "cpu":
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats()
    steps = 5
    t0 = time.perf_counter()
    for _ in range(steps):
        searcher.step()
    if device != "cpu":
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    peak = torch.cuda.max_memory_allocated() / 2**30 if device != "cpu" else 0.0
    pop_gib = popsize * L * 4 / 2**30
    print(f"L={L:,} popsize={popsize} chunk={chunk}: {dt*1000:.1f} ms/gen, "
          f"peak mem {peak:.2f} GiB (materialized population alone would be {pop_gib:.1f} GiB), "
          f"mean_eval={float(searcher.status['mean_eval']):.5f}")


if __name__ == "__main__":
    main()
