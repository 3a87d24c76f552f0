"""Distribution + optimizer semantics tests (eager CPU reference paths;
the HIP kernels are tested against these same formulas in
tests/test_gpu_kernels.py)."""

import math

import pytest
import torch

from evotorch_amd.distributions import (
    ExpGaussian,
    ExpSeparableGaussian,
    SeparableGaussian,
    SymmetricSeparableGaussian,
)
from evotorch_amd.optimizers import SGD, Adam, ClipUp, get_optimizer_class


def test_separable_gaussian_sampling_stats():
    mu = torch.tensor([1.0, -2.0, 0.5])
    sigma = torch.tensor([0.5, 1.0, 2.0])
    dist = SeparableGaussian({"mu": mu, "sigma": sigma})
    g = torch.Generator().manual_seed(3)
    samples = dist.sample(200_000, generator=g)
    assert samples.shape == (200_000, 3)
    assert torch.allclose(samples.mean(0), mu, atol=0.02)
    assert torch.allclose(samples.std(0), sigma, atol=0.02)


def test_symmetric_sampling_mirrors():
    mu = torch.tensor([1.0, 2.0])
    sigma = torch.tensor([1.0, 1.0])
    dist = SymmetricSeparableGaussian({"mu": mu, "sigma": sigma})
    s = dist.sample(10)
    assert torch.allclose((s[:5] + s[5:]) / 2, mu.expand(5, 2), atol=1e-6)
    with pytest.raises(ValueError):
        dist.sample(7)


def test_separable_gradients_match_formula():
    torch.manual_seed(0)
    mu = torch.zeros(4)
    sigma = torch.ones(4) * 2.0
    dist = SeparableGaussian({"mu": mu, "sigma": sigma})
    samples = dist.sample(64, generator=torch.Generator().manual_seed(1))
    fit = samples.sum(-1)
    grads = dist.compute_gradients(samples, fit, objective_sense="max", ranking_method="centered")
    w = torch.as_tensor(
        __import__("evotorch_amd.utils.ranking", fromlist=["rank"]).rank(fit, "centered", higher_is_better=True)
    )
    noises = samples - mu
    assert torch.allclose(grads["mu"], w @ noises, atol=1e-4)
    assert torch.allclose(grads["sigma"], w @ ((noises**2 - sigma**2) / sigma), atol=1e-4)


def test_symmetric_gradients_match_formula():
    mu = torch.zeros(3)
    sigma = torch.ones(3)
    dist = SymmetricSeparableGaussian({"mu": mu, "sigma": sigma, "divide_mu_grad_by": "num_directions", "divide_sigma_grad_by": "num_directions"})
    samples = dist.sample(32, generator=torch.Generator().manual_seed(2))
    fit = (samples**2).sum(-1)
    grads = dist.compute_gradients(samples, fit, objective_sense="min", ranking_method="centered")
    from evotorch_amd.utils.ranking import rank

    w = rank(fit, "centered", higher_is_better=False)
    d = 16
    noises = samples[:d] - mu
    wp, wm = w[:d], w[d:]
    mu_expected = (((wp - wm) / 2) @ noises) / d
    sg_expected = (((wp + wm) / 2) @ ((noises**2 - sigma**2) / sigma)) / d
    assert torch.allclose(grads["mu"], mu_expected, atol=1e-5)
    assert torch.allclose(grads["sigma"], sg_expected, atol=1e-5)


def test_exp_separable_update_is_multiplicative():
    mu = torch.zeros(5)
    sigma = torch.ones(5)
    dist = ExpSeparableGaussian({"mu": mu, "sigma": sigma})
    samples = dist.sample(20, generator=torch.Generator().manual_seed(5))
    fit = -(samples**2).sum(-1)
    grads = dist.compute_gradients(samples, fit, objective_sense="max", ranking_method="nes")
    new_dist = dist.update_parameters(grads, learning_rates={"mu": 1.0, "sigma": 0.1})
    assert torch.all(new_dist.sigma > 0)
    expected_sigma = sigma * torch.exp(0.5 * 0.1 * grads["sigma"])
    assert torch.allclose(new_dist.sigma, expected_sigma, atol=1e-6)


def test_exp_gaussian_roundtrip_and_update():
    mu = torch.zeros(3)
    A = torch.eye(3) * 2.0
    dist = ExpGaussian({"mu": mu, "sigma": A})
    z = torch.randn(10, 3)
    x = dist.to_global_coordinates(z)
    z2 = dist.to_local_coordinates(x)
    assert torch.allclose(z, z2, atol=1e-5)
    samples = dist.sample(50, generator=torch.Generator().manual_seed(0))
    fit = -(samples**2).sum(-1)
    grads = dist.compute_gradients(samples, fit, objective_sense="max", ranking_method="nes")
    assert set(grads.keys()) == {"d", "M"}
    new_dist = dist.update_parameters(grads, learning_rates={"mu": 1.0, "sigma": 0.1})
    # sigma_inv stays consistent with sigma
    prod = new_dist.sigma @ new_dist.sigma_inv
    assert torch.allclose(prod, torch.eye(3), atol=1e-4)


def test_kl_divergence_zero_for_same():
    mu = torch.zeros(4)
    sigma = torch.ones(4)
    d1 = SeparableGaussian({"mu": mu, "sigma": sigma})
    d2 = SeparableGaussian({"mu": mu.clone(), "sigma": sigma.clone()})
    assert d1.relative_entropy(d2) == pytest.approx(0.0, abs=1e-6)
    d3 = SeparableGaussian({"mu": mu + 1.0, "sigma": sigma})
    assert d1.relative_entropy(d3) > 0


# ---------------------------------------------------------------------------
# optimizers
# ---------------------------------------------------------------------------


def test_clipup_max_speed():
    opt = ClipUp(solution_length=10, stepsize=0.1, max_speed=0.15)
    g = torch.randn(10) * 100
    total = torch.zeros(10)
    for _ in range(5):
        step = opt.ascent(g)
        assert float(torch.linalg.vector_norm(step)) <= 0.15 + 1e-5
    # first step: exactly stepsize long (momentum zero)
    opt2 = ClipUp(solution_length=4, stepsize=0.1, max_speed=1.0)
    step = opt2.ascent(torch.tensor([3.0, 0.0, 0.0, 0.0]))
    assert torch.allclose(step, torch.tensor([0.1, 0.0, 0.0, 0.0]), atol=1e-6)


def test_clipup_default_max_speed():
    opt = ClipUp(solution_length=4, stepsize=0.1)
    assert opt.max_speed == pytest.approx(0.2)


def test_adam_matches_torch_adam():
    torch.manual_seed(0)
    length = 16
    opt = Adam(solution_length=length, stepsize=1e-2)
    param = torch.zeros(length, requires_grad=True)
    torch_opt = torch.optim.Adam([param], lr=1e-2)
    ours = torch.zeros(length)
    for i in range(5):
        g = torch.randn(length)
        step = opt.ascent(g)
        ours += step
        param.grad = -g.clone()  # torch minimizes; our ascent maximizes
        torch_opt.step()
    assert torch.allclose(ours, param.detach(), atol=1e-5)


def test_sgd_momentum():
    opt = SGD(solution_length=3, stepsize=0.1, momentum=0.9)
    g = torch.ones(3)
    s1 = opt.ascent(g)
    s2 = opt.ascent(g)
    assert torch.allclose(s1, torch.full((3,), 0.1))
    assert torch.allclose(s2, torch.full((3,), 0.19))


def test_get_optimizer_class():
    assert get_optimizer_class("adam") is Adam
    assert get_optimizer_class("clipup") is ClipUp
    factory = get_optimizer_class("clipup", {"max_speed": 0.5})
    opt = factory(solution_length=4, stepsize=0.1)
    assert opt.max_speed == pytest.approx(0.5)
    with pytest.raises(ValueError):
        get_optimizer_class("nope")


def test_make_functional_sampler_and_grad_estimator():
    from evotorch_amd.distributions import make_functional_sampler, make_functional_grad_estimator

    sample = make_functional_sampler(SeparableGaussian, required_parameters=["mu", "sigma"])
    mu = torch.zeros(6)
    sigma = torch.ones(6)
    pop = sample(40, mu, sigma)
    assert pop.shape == (40, 6)
    # batched parameters -> batched independent samplers
    mus = torch.stack([torch.zeros(6), torch.ones(6) * 5])
    pops = sample(30, mus, sigma)
    assert pops.shape == (2, 30, 6)
    assert float(pops[1].mean()) > 3.0

    grad = make_functional_grad_estimator(
        SeparableGaussian, required_parameters=["mu", "sigma"], objective_sense="min", ranking_method="centered"
    )
    center = torch.ones(6) * 2.0
    pop2 = sample(200, center, sigma)
    fit = (pop2**2).sum(-1)
    mu_g, sigma_g = grad(pop2, fit, center, sigma)
    assert mu_g.shape == (6,)
    assert sigma_g.shape == (6,)
    # following the (ascent) gradient of the min-sense utilities moves the
    # center toward the sphere optimum
    stepped = center + 0.05 * mu_g
    assert float((stepped**2).sum()) < float((center**2).sum())
