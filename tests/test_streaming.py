"""Streaming (two-pass, memory-O(chunk)) ES gradient mode: the N x L
population is never materialized; pass 2 regenerates noise from the
counter-addressed philox stream (evotorch_amd/core.py
_sample_and_compute_gradients_streamed). The reference has no equivalent
(its populations are always materialized — SURVEY.md §5.7)."""

import pytest
import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import PGPE, SNES
from evotorch_amd.decorators import vectorized
from evotorch_amd.distributions import (
    ExpSeparableGaussian,
    SeparableGaussian,
    SymmetricSeparableGaussian,
)


@vectorized
def sphere(x):
    return (x**2).sum(-1)


def make_problem(length=10, seed=5):
    return Problem("min", sphere, solution_length=length, initial_bounds=(-1, 1), seed=seed)


def sym_dist(length):
    return SymmetricSeparableGaussian(
        {"mu": torch.zeros(length), "sigma": torch.ones(length),
         "divide_mu_grad_by": "num_directions", "divide_sigma_grad_by": "num_directions"}
    )


@pytest.mark.parametrize("chunk", [4, 8, 20])
def test_streamed_gradients_chunk_invariant_symmetric(chunk):
    """Any chunking gives the same gradient (noise is counter-addressed)."""
    r_ref = make_problem().sample_and_compute_gradients(sym_dist(10), 40, ranking_method="centered", chunk_rows=20)
    r = make_problem().sample_and_compute_gradients(sym_dist(10), 40, ranking_method="centered", chunk_rows=chunk)
    for k in ("mu", "sigma"):
        assert torch.allclose(r["gradients"][k], r_ref["gradients"][k], atol=1e-5)
    assert float(r["mean_eval"]) == float(r_ref["mean_eval"])


def test_streamed_gradients_chunk_invariant_snes():
    dist = ExpSeparableGaussian({"mu": torch.zeros(10), "sigma": torch.ones(10)})
    r1 = make_problem(seed=7).sample_and_compute_gradients(dist, 24, ranking_method="nes", chunk_rows=4)
    r2 = make_problem(seed=7).sample_and_compute_gradients(dist, 24, ranking_method="nes", chunk_rows=24)
    for k in ("mu", "sigma"):
        assert torch.allclose(r1["gradients"][k], r2["gradients"][k], atol=1e-5)


def test_streamed_gradients_plain_separable():
    dist = SeparableGaussian({"mu": torch.zeros(10), "sigma": torch.ones(10),
                              "divide_mu_grad_by": "num_solutions", "divide_sigma_grad_by": "num_solutions"})
    r1 = make_problem(seed=9).sample_and_compute_gradients(dist, 24, ranking_method="raw", chunk_rows=4)
    r2 = make_problem(seed=9).sample_and_compute_gradients(dist, 24, ranking_method="raw", chunk_rows=24)
    for k in ("mu", "sigma"):
        assert torch.allclose(r1["gradients"][k], r2["gradients"][k], atol=1e-5)


def test_streamed_rejects_unsupported_configs():
    # elite streaming is single-rank only; num_interactions unsupported; chunk_rows >= 1
    dist = SymmetricSeparableGaussian({"mu": torch.zeros(4), "sigma": torch.ones(4)})
    with pytest.raises(ValueError):
        make_problem(length=4).sample_and_compute_gradients(dist, 8, chunk_rows=0)
    with pytest.raises(ValueError):
        make_problem(length=4).sample_and_compute_gradients(dist, 8, chunk_rows=4, num_interactions=100)


def test_pgpe_with_streaming_converges():
    prob = make_problem(length=12, seed=11)
    searcher = PGPE(prob, popsize=48, center_learning_rate=0.3, stdev_learning_rate=0.1,
                    stdev_init=2.0, distributed=True, grad_chunk_rows=8)
    searcher.step()
    first = float(searcher.status["mean_eval"])
    searcher.run(40)
    last = float(searcher.status["mean_eval"])
    assert last < first * 0.5, (first, last)


def test_snes_with_streaming_converges():
    prob = make_problem(length=12, seed=12)
    searcher = SNES(prob, popsize=40, stdev_init=2.0, distributed=True, grad_chunk_rows=8)
    searcher.step()
    first = float(searcher.status["mean_eval"])
    searcher.run(60)
    last = float(searcher.status["mean_eval"])
    assert last < first * 0.5, (first, last)


def test_streaming_requires_distributed_mode():
    with pytest.raises(ValueError):
        PGPE(make_problem(), popsize=20, center_learning_rate=0.1, stdev_learning_rate=0.1,
             stdev_init=1.0, grad_chunk_rows=4)


def test_odd_length_chunk_alignment():
    """L % 4 != 0: chunk_rows is auto-aligned to a multiple of 4 so every
    chunk starts on a philox counter boundary."""
    r1 = make_problem(length=7, seed=13).sample_and_compute_gradients(sym_dist(7), 24, ranking_method="centered", chunk_rows=3)
    r2 = make_problem(length=7, seed=13).sample_and_compute_gradients(sym_dist(7), 24, ranking_method="centered", chunk_rows=12)
    for k in ("mu", "sigma"):
        assert torch.allclose(r1["gradients"][k], r2["gradients"][k], atol=1e-5)


import hypothesis.strategies as st
from hypothesis import given, settings


@settings(max_examples=30, deadline=None)
@given(
    length=st.integers(min_value=2, max_value=23),
    directions=st.integers(min_value=2, max_value=20),
    chunk=st.integers(min_value=1, max_value=24),
    ranking=st.sampled_from(["centered", "nes", "raw", "linear"]),
    divide=st.sampled_from(["num_directions", "num_solutions", "total_weight", None]),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_streamed_gradient_fuzz(length, directions, chunk, ranking, divide, seed):
    """Property: for ANY (length, popsize, chunking, ranking, divisor), the
    streamed gradient equals the single-chunk gradient of the same seed."""
    popsize = directions * 2
    params = {"mu": torch.zeros(length), "sigma": torch.ones(length)}
    if divide is not None:
        params["divide_mu_grad_by"] = divide
        params["divide_sigma_grad_by"] = divide
    prob_a = make_problem(length=length, seed=seed)
    r_a = prob_a.sample_and_compute_gradients(
        SymmetricSeparableGaussian(dict(params)), popsize, ranking_method=ranking, chunk_rows=chunk)
    prob_b = make_problem(length=length, seed=seed)
    r_b = prob_b.sample_and_compute_gradients(
        SymmetricSeparableGaussian(dict(params)), popsize, ranking_method=ranking, chunk_rows=directions)
    for k in ("mu", "sigma"):
        a, b = r_a["gradients"][k], r_b["gradients"][k]
        assert torch.allclose(a, b, atol=1e-4, rtol=1e-4), (k, (a - b).abs().max(), length, directions, chunk, ranking, divide)


def test_streamed_cem_elite_gradients():
    """CEM's elite mean/std gradients stream exactly (masked sum/sumsq
    accumulation) and the searcher converges."""
    from evotorch_amd.algorithms import CEM

    def elite_dist():
        return SeparableGaussian({"mu": torch.zeros(10), "sigma": torch.ones(10), "parenthood_ratio": 0.25})

    r1 = make_problem(length=10, seed=21).sample_and_compute_gradients(elite_dist(), 40, ranking_method="raw", chunk_rows=7)
    r2 = make_problem(length=10, seed=21).sample_and_compute_gradients(elite_dist(), 40, ranking_method="raw", chunk_rows=40)
    for k in ("mu", "sigma"):
        assert torch.allclose(r1["gradients"][k], r2["gradients"][k], atol=1e-5)

    prob = make_problem(length=10, seed=22)
    searcher = CEM(prob, popsize=60, parenthood_ratio=0.25, stdev_init=2.0,
                   distributed=True, grad_chunk_rows=16)
    searcher.step()
    first = float(searcher.status["mean_eval"])
    searcher.run(50)
    assert float(searcher.status["mean_eval"]) < first * 0.3


def test_streaming_with_per_solution_problem():
    """Streaming composes with non-vectorized (per-solution _evaluate)
    problems — pass 1 uses the normal evaluation pipeline."""

    class PerSolutionProblem(Problem):
        def __init__(self):
            super().__init__("min", solution_length=6, initial_bounds=(-1, 1), seed=5)

        def _evaluate(self, solution):
            values = torch.Tensor.as_subclass(solution.values, torch.Tensor)
            solution.set_evaluation(float((values**2).sum()))

    prob = PerSolutionProblem()
    searcher = PGPE(prob, popsize=24, center_learning_rate=0.3, stdev_learning_rate=0.1,
                    stdev_init=1.5, distributed=True, grad_chunk_rows=4)
    searcher.step()
    first = float(searcher.status["mean_eval"])
    searcher.run(30)
    assert float(searcher.status["mean_eval"]) < first * 0.5
