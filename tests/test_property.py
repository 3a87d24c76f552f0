"""Property-based tests (hypothesis) for core invariants."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from evotorch_amd import Problem, SolutionBatch
from evotorch_amd.core import _compute_pareto_ranks_eager, _crowding_distances
from evotorch_amd.decorators import vectorized
from evotorch_amd.utils import CList, rank


@vectorized
def sphere(x):
    return (x**2).sum(-1)


@settings(max_examples=25, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=40),
    method=st.sampled_from(["centered", "linear", "nes", "normalized", "raw"]),
    higher=st.booleans(),
)
def test_rank_respects_order(n, method, higher):
    """Utilities must be monotone in fitness: a better solution never gets
    a lower utility."""
    f = torch.randn(n)
    u = rank(f, method, higher_is_better=higher)
    order = f.argsort(descending=higher)  # best first
    sorted_u = u[order]
    diffs = sorted_u[:-1] - sorted_u[1:]
    assert bool((diffs >= -1e-5).all()), (method, higher, sorted_u)


@settings(max_examples=20, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=30),
    start=st.integers(min_value=0, max_value=29),
    length=st.integers(min_value=1, max_value=30),
)
def test_batch_slice_aliases_parent(n, start, length):
    n = max(n, start + 1)
    stop = min(start + length, n)
    prob = Problem("min", sphere, solution_length=4, initial_bounds=(-1, 1), seed=0)
    batch = prob.generate_batch(n)
    sub = batch[start:stop]
    assert len(sub) == stop - start
    if len(sub) > 0:
        sub.access_values().fill_(7.0)
        assert bool((batch.unsafe_values[start:stop] == 7.0).all())


@settings(max_examples=20, deadline=None)
@given(n=st.integers(min_value=1, max_value=50), m=st.integers(min_value=1, max_value=4))
def test_pareto_ranks_consistency(n, m):
    """Front-0 members are exactly the non-dominated ones, and every
    front-k>0 member is dominated by someone in a lower front."""
    utils = torch.randn(n, m)
    ranks, _ = _compute_pareto_ranks_eager(utils, crowdsort=False)
    a = utils.unsqueeze(1)
    b = utils.unsqueeze(0)
    dom = (a >= b).all(-1) & (a > b).any(-1)  # dom[i, j]: i dominates j
    nondominated = dom.sum(0) == 0
    assert torch.equal(ranks == 0, nondominated)
    for j in range(n):
        r = int(ranks[j])
        if r > 0:
            dominators = torch.nonzero(dom[:, j], as_tuple=True)[0]
            assert int(ranks[dominators].min()) == r - 1 or bool((ranks[dominators] < r).any())


@settings(max_examples=15, deadline=None)
@given(
    ops=st.lists(st.sampled_from(["push", "pop", "pushleft", "popleft"]), min_size=1, max_size=30),
    batch=st.integers(min_value=1, max_value=4),
)
def test_clist_matches_python_deque(ops, batch):
    """CList's batched circular deque must behave like a python deque."""
    from collections import deque

    max_len = 8
    lst = CList(max_length=max_len, batch_size=batch)
    refs = [deque() for _ in range(batch)]
    counter = 0.0
    for op in ops:
        if op in ("push", "pushleft"):
            counter += 1.0
            vals = torch.arange(batch, dtype=torch.float32) + counter * 10
            if op == "push":
                lst.append_(vals)
                for b, d in enumerate(refs):
                    if len(d) < max_len:
                        d.append(float(vals[b]))
            else:
                lst.appendleft_(vals)
                for b, d in enumerate(refs):
                    if len(d) < max_len:
                        d.appendleft(float(vals[b]))
        else:
            popped = lst.pop_() if op == "pop" else lst.popleft_()
            for b, d in enumerate(refs):
                if len(d) > 0:
                    expected = d.pop() if op == "pop" else d.popleft()
                    assert float(popped[b]) == expected
    assert lst.length.tolist() == [len(d) for d in refs]
    for b, d in enumerate(refs):
        for i, expected in enumerate(d):
            assert float(lst.get(torch.tensor([i] * batch))[b]) == expected


@settings(max_examples=15, deadline=None)
@given(
    n=st.integers(min_value=4, max_value=40),
    take=st.integers(min_value=1, max_value=10),
)
def test_take_best_is_truly_best(n, take):
    take = min(take, n)
    prob = Problem("min", sphere, solution_length=3, initial_bounds=(-2, 2), seed=1)
    batch = prob.generate_batch(n)
    prob.evaluate(batch)
    best = batch.take_best(take)
    best_vals = best.unsafe_evals[:, 0]
    all_vals = batch.unsafe_evals[:, 0]
    threshold = all_vals.sort().values[take - 1]
    assert bool((best_vals <= threshold + 1e-6).all())
