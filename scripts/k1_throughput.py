"""Measure K1 (philox sampling) and K3 (gradient reduction) standalone
throughput at streaming-scale shapes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd.ops import es_gradients, sample_gaussian


def bench(fn, iters=20):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    for rows, length in [(16, 100_000_000), (64, 10_000_000), (4000, 6409), (12500, 1_000_000)]:
        mu = torch.zeros(length, device="cuda:0")
        sigma = torch.ones(length, device="cuda:0")
        out = torch.empty(rows, length, device="cuda:0")
        dt = bench(lambda: sample_gaussian(out, mu, sigma, symmetric=True, seed=1))
        gb = out.numel() * 4 / 2**30
        print(f"K1 sample {rows}x{length:,}: {dt*1000:7.2f} ms  {gb/dt:7.1f} GiB/s write  {out.numel()/dt/1e9:6.2f} Gnormals/s")
        w = torch.randn(rows, device="cuda:0")
        dt = bench(lambda: es_gradients(out, mu, sigma, w, symmetric=True))
        print(f"K3 grads  {rows}x{length:,}: {dt*1000:7.2f} ms  {gb/dt:7.1f} GiB/s read")
        del out, mu, sigma


if __name__ == "__main__":
    main()
