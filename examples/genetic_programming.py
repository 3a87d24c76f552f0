"""Batched genetic programming on the CMemory/CList structures — the
reference's Genetic_Programming notebook pattern: a whole POPULATION of
stack programs executes in lockstep, one structure op per program step,
with `where`-masked semantics selecting each program's opcode. On a GPU
every step is a handful of batched gather/scatter kernels regardless of
population size.

Task: symbolic regression of y = x^2 + x over x in [-1, 1], with linear
stack programs over {PUSH_X, PUSH_1, ADD, MUL, SUB}.
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import GeneticAlgorithm
from evotorch_amd.decorators import vectorized
from evotorch_amd.operators import MultiPointCrossOver
from evotorch_amd.operators.base import CopyingOperator
from evotorch_amd.utils import CList

N_OPS = 5
PUSH_X, PUSH_1, ADD, MUL, SUB = range(N_OPS)
PROGRAM_LEN = 12
STACK_DEPTH = PROGRAM_LEN + 1


def run_programs(opcodes: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """Execute a batch of programs at a batch of inputs.

    opcodes: (B, P) integer-valued tensor; x: (K,) inputs.
    Returns predictions (B, K). All B programs run in lockstep: program
    step i performs every structure op masked by `where=(opcode == OP)`.
    """
    B, P = opcodes.shape
    K = x.shape[0]
    device = opcodes.device
    preds = torch.empty(B, K, device=device)
    for k in range(K):
        stack = CList(max_length=STACK_DEPTH, batch_size=B, device=device)
        xk = torch.full((B,), float(x[k]), device=device)
        ones = torch.ones(B, device=device)
        for i in range(P):
            op = opcodes[:, i].round().long().clamp(0, N_OPS - 1)
            # pushes (masked per program)
            stack.push_(xk, where=(op == PUSH_X))
            stack.push_(ones, where=(op == PUSH_1))
            # binary ops: pop two, push result — only where the op matches
            # AND the stack has >= 2 elements (else the op is a no-op)
            binary = (op == ADD) | (op == MUL) | (op == SUB)
            able = binary & (stack.length >= 2)
            a = stack.pop_(where=able)
            b = stack.pop_(where=able)
            result = torch.where(op == ADD, a + b, torch.where(op == MUL, a * b, b - a))
            stack.push_(result, where=able)
        top = stack.get(torch.clamp(stack.length - 1, min=0))
        empty = stack.length == 0
        preds[:, k] = torch.where(empty, torch.full_like(top, 1e6), top)
    return preds


class OpcodeMutation(CopyingOperator):
    """Randomly rewrite each gene with a fresh opcode with probability p."""

    def __init__(self, problem, *, probability: float = 0.1):
        super().__init__(problem)
        self._p = float(probability)

    def _do(self, batch):
        result = batch.take(torch.arange(len(batch)))
        vals = result.access_values()
        g = self._problem.generator
        mask = torch.rand(vals.shape, device=vals.device, generator=g) < self._p
        fresh = torch.randint(0, N_OPS, vals.shape, device=vals.device, generator=g).to(vals.dtype)
        vals.copy_(torch.where(mask, fresh, vals.round().clamp(0, N_OPS - 1)))
        return result


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--generations", type=int, default=60)
    p.add_argument("--popsize", type=int, default=512)
    args = p.parse_args()
    device = args.device

    x = torch.linspace(-1, 1, 16, device=device)
    target = x**2 + x

    @vectorized
    def fitness(programs: torch.Tensor) -> torch.Tensor:
        preds = run_programs(programs, x)
        return ((preds - target) ** 2).mean(-1)

    problem = Problem("min", fitness, solution_length=PROGRAM_LEN,
                      initial_bounds=(0, N_OPS - 1), device=device, seed=1)
    ga = GeneticAlgorithm(
        problem,
        popsize=args.popsize,
        operators=[
            MultiPointCrossOver(problem, tournament_size=4, num_points=2),
            OpcodeMutation(problem, probability=0.15),
        ],
    )
    ga.run(args.generations)
    best = ga.population.take_best()
    program = torch.Tensor.as_subclass(best.values, torch.Tensor).round().long().clamp(0, N_OPS - 1)
    names = ["PUSH_X", "PUSH_1", "ADD", "MUL", "SUB"]
    print("best MSE:", float(best.evaluation))
    print("program:", " ".join(names[int(o)] for o in program))


if __name__ == "__main__":
    main()
