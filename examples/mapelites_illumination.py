"""MAP-Elites illumination: maximize fitness across a behavior grid."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import MAPElites
from evotorch_amd.decorators import vectorized
from evotorch_amd.operators import GaussianMutation


@vectorized
def fitness_and_features(x: torch.Tensor):
    fitness = -(x**2).sum(-1)
    features = x[:, :2]  # first two coordinates are the behavior descriptor
    return fitness, features


def main():
    problem = Problem("max", fitness_and_features, solution_length=6,
                      initial_bounds=(-2, 2), eval_data_length=2, seed=3)
    grid = MAPElites.make_feature_grid([-2.0, -2.0], [2.0, 2.0], [8, 8])
    me = MAPElites(problem, operators=[GaussianMutation(problem, stdev=0.3)], feature_grid=grid)
    me.run(50)
    filled = int(me.filled.sum())
    print(f"filled {filled}/{len(me.filled)} cells")


if __name__ == "__main__":
    main()
