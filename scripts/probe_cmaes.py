"""Section timing for CMAES._step at d=4096 (host+device per phase)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem, SolutionBatch
from evotorch_amd.algorithms import CMAES
from evotorch_amd.decorators import vectorized


@vectorized
def sphere(x):
    return (x**2).sum(-1)


def main():
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    prob = Problem("min", sphere, solution_length=4096, initial_bounds=(-3, 3), device=device, seed=1)
    s = CMAES(prob, stdev_init=1.0, popsize=64)

    def sync():
        if device != "cpu":
            torch.cuda.synchronize()

    def timed(label, fn):
        sync()
        t0 = time.perf_counter()
        out = fn()
        sync()
        print(f"  {label:<26} {(time.perf_counter()-t0)*1e3:8.2f} ms")
        return out

    for _ in range(2):
        s.step()  # warmup
    for it in range(3):
        print(f"gen {it}:")
        z, y, x = timed("sample", s._sample)
        batch = timed("make batch", lambda: SolutionBatch(prob, popsize=s._popsize, device=s._m.device, empty=True))
        timed("copy values", lambda: batch.access_values().copy_(x))
        timed("evaluate", lambda: prob.evaluate(batch))
        order = timed("argsort", lambda: batch.argsort(obj_index=0))
        def rest():
            s._population = batch
            zz = z[order]
            yy = y[order]
            w = s._weights
            w_pos = w[: s._mu]
            y_w = w_pos @ yy[: s._mu]
            z_w = w_pos @ zz[: s._mu]
            s._m = s._m + s._c_m * s._sigma * y_w
            return yy, zz, y_w, z_w
        yy, zz, y_w, z_w = timed("mean update", rest)
        def cov():
            w = s._weights
            w_adj = torch.where(w < 0, w * 4096 / (zz**2).sum(-1).clamp(min=1e-12), w)
            rank_mu = (yy * w_adj.unsqueeze(-1)).T @ yy
            rank_one = torch.outer(s._p_c, s._p_c)
            C = 0.98 * s._C + 0.01 * rank_one + 0.01 * rank_mu
            return 0.5 * (C + C.T)
        timed("cov update", cov)
        timed("cholesky", lambda: torch.linalg.cholesky(s._C))
        timed("full step()", s.step)


if __name__ == "__main__":
    main()
