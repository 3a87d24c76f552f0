"""Batched hyperparameter sweep with the functional API: 4 independent
PGPE searches with different stdev inits run in ONE batched tensor
program (the reference's Functional-API notebook pattern)."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import torch

from evotorch_amd.algorithms.functional import pgpe, pgpe_ask, pgpe_tell


def sphere(x):
    return (x**2).sum(-1)


def main():
    B, L = 4, 20
    centers = torch.ones(B, L) * 3.0
    state = pgpe(
        center_init=centers,
        center_learning_rate=0.3,
        stdev_learning_rate=0.1,
        stdev_init=torch.tensor([0.5, 1.0, 2.0, 4.0]).unsqueeze(-1).expand(B, L).clone(),
        objective_sense="min",
    )
    g = torch.Generator().manual_seed(0)
    for gen in range(80):
        pop = pgpe_ask(state, popsize=100, generator=g)   # (B, 100, L)
        state = pgpe_tell(state, pop, sphere(pop))
    from evotorch_amd.algorithms.functional.funcoptimizers import get_functional_optimizer

    _, opt_ask, _ = get_functional_optimizer(state.optimizer)
    final = sphere(opt_ask(state.optimizer_state))
    for b in range(B):
        print(f"search {b}: final sphere value {float(final[b]):.4f}")


if __name__ == "__main__":
    main()
