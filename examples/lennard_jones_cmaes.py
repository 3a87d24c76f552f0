"""Minimizing Lennard-Jones atom-cluster potentials with CMA-ES (mirrors
the reference's Minimizing_Lennard-Jones_Atom_Cluster_Potentials.ipynb:
solutions are flat (n_atoms x 3) coordinate vectors; fitness is the LJ
pair potential; optimum for small clusters is known from the Cambridge
cluster database, e.g. E(6 atoms) = -12.712062).

Run: python examples/lennard_jones_cmaes.py [--atoms 6] [--generations 600]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import CMAES
from evotorch_amd.decorators import vectorized


def make_potential(n_atoms: int):
    @vectorized
    def lj_potential(x: torch.Tensor) -> torch.Tensor:
        pos = x.reshape(x.shape[0], n_atoms, 3)
        diff = pos.unsqueeze(2) - pos.unsqueeze(1)            # (N, a, a, 3)
        r2 = (diff * diff).sum(-1)                            # (N, a, a)
        iu = torch.triu_indices(n_atoms, n_atoms, offset=1)
        r2 = r2[:, iu[0], iu[1]].clamp(min=1e-12)             # (N, pairs)
        inv6 = r2.pow(-3)
        return (4.0 * (inv6 * inv6 - inv6)).sum(-1)

    return lj_potential


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--atoms", type=int, default=6)
    ap.add_argument("--generations", type=int, default=600)
    ap.add_argument("--device", default="cpu")
    args = ap.parse_args()

    problem = Problem(
        "min", make_potential(args.atoms), solution_length=args.atoms * 3,
        initial_bounds=(-1.0, 1.0), device=args.device, seed=0,
        store_solution_stats=True,  # keep best/best_eval on GPU devices too
    )
    searcher = CMAES(problem, stdev_init=0.3, popsize=64)
    searcher.run(args.generations)
    best = float(searcher.status["best_eval"])
    print(f"{args.atoms}-atom cluster: best LJ energy {best:.6f}")
    if args.atoms == 6:
        # the 6-atom landscape has a strong non-global funnel at -12.302931;
        # the global octahedron (-12.712062, Cambridge cluster database)
        # needs either many restarts or an informed init — evaluating the
        # analytic octahedron through this potential reproduces it exactly
        print("known global optimum for 6 atoms: -12.712062 (octahedron)")


if __name__ == "__main__":
    main()
