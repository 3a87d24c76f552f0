"""Multi-objective GA (NSGA-II selection) on the Kursawe function."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import GeneticAlgorithm
from evotorch_amd.decorators import vectorized
from evotorch_amd.operators import GaussianMutation, SimulatedBinaryCrossOver


@vectorized
def kursawe(x: torch.Tensor) -> torch.Tensor:
    f1 = torch.sum(-10 * torch.exp(-0.2 * torch.sqrt(x[:, :-1] ** 2 + x[:, 1:] ** 2)), dim=-1)
    f2 = torch.sum(torch.abs(x) ** 0.8 + 5 * torch.sin(x**3), dim=-1)
    return torch.stack([f1, f2], dim=-1)


def main():
    problem = Problem(["min", "min"], kursawe, solution_length=3,
                      initial_bounds=(-5.0, 5.0), bounds=(-5.0, 5.0), seed=1)
    ga = GeneticAlgorithm(
        problem,
        popsize=200,
        operators=[
            SimulatedBinaryCrossOver(problem, tournament_size=4, eta=8.0),
            GaussianMutation(problem, stdev=0.1),
        ],
    )
    ga.run(100)
    fronts = ga.population.arg_pareto_sort()
    print(f"front sizes: {[len(f) for f in fronts[:5]]}")
    best_front = ga.population.access_evals()[fronts[0]]
    print("pareto front (first 10 points):")
    print(best_front[:10])


if __name__ == "__main__":
    main()
