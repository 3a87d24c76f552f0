"""Phase-level timing probe for the large-ES path (host+device, per
generation): batch alloc, K1 sample, eval, ranking, K3 grads, K4 update."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem, SolutionBatch
from evotorch_amd.decorators import vectorized
from evotorch_amd.distributions import SymmetricSeparableGaussian
from evotorch_amd.optimizers import ClipUp
from evotorch_amd.utils.ranking import rank


def timed(label, fn, sync=True):
    t0 = time.perf_counter()
    out = fn()
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    print(f"  {label:<28} {(time.perf_counter()-t0)*1e3:9.2f} ms")
    return out


def main():
    L = 1_000_000
    N = 12_500
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    target = torch.randn(L, device=device)

    @vectorized
    def quadratic(x):
        d = x - target
        return (d * d).sum(-1)

    prob = Problem("min", quadratic, solution_length=L, initial_bounds=(-1, 1), device=device, seed=1)
    dist = SymmetricSeparableGaussian(
        {"mu": torch.zeros(L, device=device), "sigma": torch.ones(L, device=device),
         "divide_mu_grad_by": "num_directions", "divide_sigma_grad_by": "num_directions"}
    )
    opt = ClipUp(solution_length=L, device=device, stepsize=0.1, max_speed=0.2)

    for it in range(3):
        print(f"gen {it}:")
        batch = timed("generate_batch(empty)", lambda: prob.generate_batch(N, empty=True))
        timed("K1 sample", lambda: dist.sample(out=batch.access_values(), generator=prob))
        timed("evaluate (quadratic)", lambda: prob.evaluate(batch))
        fits = batch.access_evals()[:, 0]
        w = timed("rank (centered)", lambda: rank(fits, "centered", higher_is_better=False))
        grads = timed("K3 gradients", lambda: dist._compute_gradients(batch.unsafe_values, w.to(torch.float32), "centered"))
        timed("K4 clipup+update", lambda: dist.update_parameters(grads, optimizers={"mu": opt}, learning_rates={"sigma": 0.1}))
        del batch


if __name__ == "__main__":
    main()
