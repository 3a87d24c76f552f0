"""Population-based searchers: GeneticAlgorithm (elitist GA / NSGA-II when
multi-objective), SteadyStateGA, Cosyne.

Reference parity: /root/reference/src/evotorch/algorithms/ga.py
(ExtendedPopulationMixin :62, GeneticAlgorithm :266, SteadyStateGA :691,
Cosyne :893). The NSGA-II selection path uses the batched domination-count
front peeling of evotorch_amd.core (K7 in SURVEY.md §2.9).
"""

from typing import Iterable, Optional

import torch

from ..core import Problem, SolutionBatch
from ..operators.base import CopyingOperator, CrossOver, Operator
from ..operators.real import CosynePermutation, GaussianMutation, OnePointCrossOver
from .searchalgorithm import SearchAlgorithm, SinglePopulationAlgorithmMixin

__all__ = ["ExtendedPopulationMixin", "GeneticAlgorithm", "SteadyStateGA", "Cosyne"]


class ExtendedPopulationMixin:
    """Builds the extended population: runs the operator pipeline on the
    current population and concatenates parents + children (reference
    ga.py:62)."""

    def _init_extended(self, *, operators: Iterable, re_evaluate: bool, re_evaluate_parents_first: Optional[bool]):
        self._operators = list(operators)
        self._re_evaluate = bool(re_evaluate)
        if re_evaluate_parents_first is None:
            re_evaluate_parents_first = False
        self._re_evaluate_parents_first = bool(re_evaluate_parents_first)

    def _apply_pipeline(self, batch: SolutionBatch) -> SolutionBatch:
        """Chain the operators: the first takes the population; each later
        operator transforms the previous output (CopyingOperator) or
        mutates it in place (Operator)."""
        current: Optional[SolutionBatch] = None
        for op in self._operators:
            source = batch if current is None else current
            if isinstance(op, (CrossOver, CopyingOperator)):
                current = op(source)
            elif isinstance(op, Operator):
                if current is None:
                    current = source.take(torch.arange(len(source)))
                op(current)
            elif callable(op):
                out = op(source)
                current = out if isinstance(out, SolutionBatch) else current
            else:
                raise TypeError(f"Unsupported operator {op!r}")
        if current is None:
            raise ValueError("Operator pipeline produced no children")
        return current

    def _make_extended_population(self, population: SolutionBatch) -> SolutionBatch:
        problem: Problem = self.problem
        children = self._apply_pipeline(population)
        if self._re_evaluate:
            if self._re_evaluate_parents_first:
                problem.evaluate(population)
                problem.evaluate(children)
                return population.concat(children)
            extended = population.concat(children)
            extended.forget_evals()
            problem.evaluate(extended)
            return extended
        problem.evaluate(children)
        return population.concat(children)


class GeneticAlgorithm(SearchAlgorithm, SinglePopulationAlgorithmMixin, ExtendedPopulationMixin):
    """Elitist GA: extended population = parents + children, then
    `take_best(popsize)` — which is pareto-rank + crowding-distance based
    for multi-objective problems (NSGA-II; reference ga.py:266)."""

    def __init__(
        self,
        problem: Problem,
        *,
        operators: Iterable,
        popsize: int,
        elitist: bool = True,
        re_evaluate: bool = True,
        re_evaluate_parents_first: Optional[bool] = None,
        _allow_empty_operator_list: bool = False,
    ):
        SearchAlgorithm.__init__(self, problem)
        if not _allow_empty_operator_list and len(list(operators)) == 0:
            raise ValueError("GeneticAlgorithm requires at least one operator")
        self._init_extended(operators=operators, re_evaluate=re_evaluate, re_evaluate_parents_first=re_evaluate_parents_first)
        self._popsize = int(popsize)
        self._elitist = bool(elitist)
        self._population: Optional[SolutionBatch] = None
        SinglePopulationAlgorithmMixin.__init__(self)

    @property
    def population(self) -> Optional[SolutionBatch]:
        return self._population

    @property
    def popsize(self) -> int:
        return self._popsize

    def _state_items(self) -> dict:
        if self._population is None:
            return {}
        return {"values": self._population.unsafe_values, "evals": self._population.unsafe_evals}

    def _load_state_items(self, items: dict):
        if "values" not in items:
            self._population = None
            return
        problem = self.problem
        values = torch.as_tensor(items["values"])
        batch = SolutionBatch(problem, popsize=values.shape[0], empty=True)
        batch.access_values().copy_(values.to(batch.device, batch.dtype))
        batch.unsafe_evals.copy_(torch.as_tensor(items["evals"]).to(batch.device))
        self._population = batch

    def _step(self):
        problem = self.problem
        if self._population is None:
            self._population = problem.generate_batch(self._popsize)
            problem.evaluate(self._population)
            return
        extended = self._make_extended_population(self._population)
        if self._elitist:
            self._population = extended.take_best(self._popsize)
        else:
            children = extended[len(self._population):]
            if len(children) >= self._popsize:
                self._population = children.take_best(self._popsize)
            else:
                self._population = extended.take_best(self._popsize)


class SteadyStateGA(GeneticAlgorithm):
    """Alias of the elitist GeneticAlgorithm (reference ga.py:691)."""

    def __init__(self, problem: Problem, *, popsize: int, operators: Optional[Iterable] = None, re_evaluate: bool = True, re_evaluate_parents_first: Optional[bool] = None):
        super().__init__(
            problem,
            operators=list(operators) if operators is not None else [],
            popsize=popsize,
            elitist=True,
            re_evaluate=re_evaluate,
            re_evaluate_parents_first=re_evaluate_parents_first,
            _allow_empty_operator_list=True,
        )

    def use(self, operator):
        """Register an operator after construction (reference SteadyStateGA
        API)."""
        self._operators.append(operator)


class Cosyne(SearchAlgorithm, SinglePopulationAlgorithmMixin):
    """CoSyNE (Gomez et al. 2008): elites + tournament crossover +
    Gaussian mutation + column-wise permutation (reference ga.py:893)."""

    def __init__(
        self,
        problem: Problem,
        *,
        popsize: int,
        tournament_size: int,
        mutation_stdev: Optional[float],
        mutation_probability: Optional[float] = None,
        permute_all: bool = False,
        num_elites: Optional[int] = None,
        elitism_ratio: Optional[float] = None,
        eta: Optional[float] = None,
        num_children: Optional[int] = None,
    ):
        problem.ensure_single_objective()
        problem.ensure_numeric()
        SearchAlgorithm.__init__(self, problem)
        self._popsize = int(popsize)
        self._tournament_size = int(tournament_size)
        if num_elites is not None and elitism_ratio is not None:
            raise ValueError("Provide at most one of num_elites, elitism_ratio")
        if num_elites is not None:
            self._num_elites = int(num_elites)
        elif elitism_ratio is not None:
            self._num_elites = int(self._popsize * float(elitism_ratio))
        else:
            self._num_elites = 0
        self._num_children = int(num_children) if num_children is not None else (self._popsize - self._num_elites)
        if self._num_children % 2 != 0:
            self._num_children += 1

        if eta is not None:
            from ..operators.real import SimulatedBinaryCrossOver

            self._cross = SimulatedBinaryCrossOver(problem, tournament_size=self._tournament_size, eta=float(eta), num_children=self._num_children)
        else:
            self._cross = OnePointCrossOver(problem, tournament_size=self._tournament_size, num_children=self._num_children)
        self._mutate = GaussianMutation(problem, stdev=float(mutation_stdev), mutation_probability=mutation_probability) if mutation_stdev is not None else None
        self._permute = CosynePermutation(problem, permute_all=permute_all)
        self._population: Optional[SolutionBatch] = None
        SinglePopulationAlgorithmMixin.__init__(self)

    @property
    def population(self) -> Optional[SolutionBatch]:
        return self._population

    def _step(self):
        problem = self.problem
        if self._population is None:
            self._population = problem.generate_batch(self._popsize)
            problem.evaluate(self._population)
            return
        pop = self._population
        elites = pop.take_best(self._num_elites) if self._num_elites > 0 else None
        children = self._cross(pop)
        if self._mutate is not None:
            children = self._mutate(children)
        permuted = self._permute(pop)
        parts = [children, permuted]
        if elites is not None:
            parts.insert(0, elites)
        extended = SolutionBatch.cat(parts)
        extended.forget_evals()
        problem.evaluate(extended)
        self._population = extended.take_best(self._popsize)
