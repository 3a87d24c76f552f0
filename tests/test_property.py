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


@settings(max_examples=20, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=20).map(lambda v: v * 2),
    length=st.integers(min_value=2, max_value=30),
    seed=st.integers(min_value=0, max_value=9999),
)
def test_crossover_children_inherit_columns(n, length, seed):
    """One/two-point crossover: every child element comes from one of its
    two parents at the same column."""
    from evotorch_amd.operators.functional import one_point_cross_over, two_point_cross_over

    g = torch.Generator().manual_seed(seed)
    parents = torch.randn(n, length, generator=g)
    for fn in (one_point_cross_over, two_point_cross_over):
        children = fn(parents, generator=g)
        half = n // 2
        p1, p2 = parents[:half], parents[half : 2 * half]
        for row in range(children.shape[0]):
            a = p1[row % half]
            b = p2[row % half]
            c = children[row]
            from_parent = (c == a) | (c == b)
            assert bool(from_parent.all()), (fn.__name__, row)


@settings(max_examples=20, deadline=None)
@given(
    pairs=st.integers(min_value=1, max_value=20),
    length=st.integers(min_value=1, max_value=30),
    eta=st.floats(min_value=1.0, max_value=40.0),
    seed=st.integers(min_value=0, max_value=9999),
)
def test_sbx_preserves_pair_means(pairs, length, eta, seed):
    """Simulated binary crossover: each child pair's mean equals its parent
    pair's mean (definitional SBX property)."""
    from evotorch_amd.operators.functional import simulated_binary_cross_over

    g = torch.Generator().manual_seed(seed)
    parents = torch.randn(pairs * 2, length, generator=g)
    children = simulated_binary_cross_over(parents, eta=eta, generator=g)
    half = pairs
    p_mean = (parents[:half] + parents[half : 2 * half]) / 2
    c_mean = (children[:half] + children[half : 2 * half]) / 2
    assert torch.allclose(p_mean, c_mean, atol=1e-5)


@settings(max_examples=15, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=30),
    length=st.integers(min_value=2, max_value=20),
    seed=st.integers(min_value=0, max_value=9999),
)
def test_cosyne_permutation_is_columnwise_permutation(n, length, seed):
    from evotorch_amd.operators.functional import cosyne_permutation

    g = torch.Generator().manual_seed(seed)
    values = torch.randn(n, length, generator=g)
    permuted = cosyne_permutation(values, permute_all=True, generator=g)
    for col in range(length):
        assert torch.allclose(values[:, col].sort().values, permuted[:, col].sort().values)


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(min_value=0, max_value=9999))
def test_operators_respect_bounds(seed):
    """GaussianMutation/PolynomialMutation/SBX children never leave the
    problem's bounds (Operator._respect_bounds, reference base.py:75)."""
    from evotorch_amd.operators import GaussianMutation, PolynomialMutation, SimulatedBinaryCrossOver

    prob = Problem("min", sphere, solution_length=6, bounds=(-1.0, 1.0),
                   initial_bounds=(-1.0, 1.0), seed=seed)
    batch = prob.generate_batch(20)
    prob.evaluate(batch)
    for op in (GaussianMutation(prob, stdev=5.0), PolynomialMutation(prob, eta=5.0),
               SimulatedBinaryCrossOver(prob, eta=3.0, tournament_size=2)):
        out = op(batch)
        vals = out.unsafe_values if hasattr(out, "unsafe_values") else out
        assert bool((vals >= -1.0 - 1e-6).all()) and bool((vals <= 1.0 + 1e-6).all()), type(op).__name__


@settings(max_examples=25, deadline=None)
@given(
    rows=st.integers(min_value=2, max_value=16),
    length=st.integers(min_value=1, max_value=24),
    split=st.integers(min_value=1, max_value=15),
    base=st.integers(min_value=0, max_value=10_000),
    seed=st.integers(min_value=0, max_value=2**31 - 1),
)
def test_counter_addressed_sampling_partition_invariant(rows, length, split, base, seed):
    """Counter-addressed sampling (K1's stream-per-row mode, the foundation
    of rank sharding and the streaming two-pass gradient): regenerating ANY
    row partition with the matching row_offset reproduces the full
    population's rows exactly — on the CPU philox reference path here,
    bitwise vs the HIP kernel in the GPU suite."""
    from evotorch_amd import ops

    split = min(split, rows - 1)
    mu = torch.randn(length)
    sigma = torch.rand(length) + 0.1

    full = torch.empty(rows, length)
    ops.sample_gaussian(full, mu, sigma, seed=seed, row_offset=base)

    top = torch.empty(split, length)
    bottom = torch.empty(rows - split, length)
    ops.sample_gaussian(top, mu, sigma, seed=seed, row_offset=base)
    ops.sample_gaussian(bottom, mu, sigma, seed=seed, row_offset=base + split)

    assert torch.equal(full[:split], top)
    assert torch.equal(full[split:], bottom)


@settings(max_examples=15, deadline=None)
@given(
    pairs=st.integers(min_value=1, max_value=8),
    length=st.integers(min_value=1, max_value=16),
    seed=st.integers(min_value=0, max_value=2**31 - 1),
)
def test_counter_addressed_symmetric_mirror(pairs, length, seed):
    """Symmetric counter-addressed sampling fills the halves layout with an
    exact mirror: row i + pairs == 2*mu - row i."""
    from evotorch_amd import ops

    mu = torch.randn(length)
    sigma = torch.rand(length) + 0.1
    out = torch.empty(2 * pairs, length)
    ops.sample_gaussian(out, mu, sigma, symmetric=True, seed=seed)
    assert torch.allclose(out[pairs:], 2.0 * mu - out[:pairs], atol=1e-6)


@settings(max_examples=20, deadline=None)
@given(n=st.integers(min_value=2, max_value=64))
def test_centered_ranking_sums_to_zero(n):
    """Centered ranking (PGPE's default fitness shaping) is zero-sum, so the
    mu-gradient is translation-invariant in the fitnesses."""
    from evotorch_amd.utils.ranking import rank

    fit = torch.randn(n)
    w = rank(fit, "centered", higher_is_better=True)
    assert abs(float(w.sum())) < 1e-5
    shifted = rank(fit + 123.456, "centered", higher_is_better=True)
    assert torch.allclose(w, shifted)


@settings(max_examples=20, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=40),
    pieces=st.integers(min_value=1, max_value=6),
    seed=st.integers(min_value=0, max_value=9999),
)
def test_solutionbatch_split_concat_roundtrip(n, pieces, seed):
    from evotorch_amd.core import SolutionBatch

    pieces = min(pieces, n)
    prob = Problem("min", sphere, solution_length=5, initial_bounds=(-1, 1), seed=seed)
    batch = prob.generate_batch(n)
    prob.evaluate(batch)
    rebuilt = SolutionBatch.cat(batch.split(pieces))
    assert torch.equal(rebuilt.values.as_subclass(torch.Tensor), batch.values.as_subclass(torch.Tensor))
    assert torch.equal(rebuilt.evals.as_subclass(torch.Tensor), batch.evals.as_subclass(torch.Tensor))


@settings(max_examples=20, deadline=None)
@given(seed=st.integers(min_value=0, max_value=9999))
def test_modify_tensor_is_idempotent(seed):
    """Applying the stdev-control clamp twice equals applying it once."""
    from evotorch_amd.utils.misc import modify_tensor

    g = torch.Generator().manual_seed(seed)
    original = torch.rand(12, generator=g) + 0.5
    target = original + torch.randn(12, generator=g)
    once = modify_tensor(original, target, lb=0.1, ub=2.0, max_change=0.2)
    twice = modify_tensor(original, once, lb=0.1, ub=2.0, max_change=0.2)
    assert torch.allclose(once, twice)
