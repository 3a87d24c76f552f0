"""Op dispatch: eager (CPU / reference) implementations + HIP kernel routing.

Every hot op of the framework goes through this module. The contract:

* On CPU tensors: run the eager PyTorch implementation below (these are the
  numerics references the HIP kernels are tested against).
* On ROCm (``cuda``) tensors: run the hand-written CDNA4 HIP kernel from the
  in-tree extension ``evotorch_amd._C``. If the extension is missing on a
  machine that has a GPU, we raise loudly instead of silently falling back
  (the eager path would hide a broken native build).

Kernel inventory: K1 sample_gaussian, K2 ranking utilities
(torch.sort based — the sort itself is rocPRIM-backed via torch), K3/K4
es_gradients + fused update, K10 batched policy forward (in
evotorch_amd.models), K11 running-norm update.
"""

import os
from typing import Optional, Tuple

import torch

__all__ = [
    "affine_from_noise",
    "hip_available",
    "hip_required",
    "load_hip",
    "sample_gaussian",
    "es_gradients",
    "snes_gradients",
    "clipup_step_",
    "fused_adam_step_",
    "pareto_ranks",
    "domination_counts",
    "cma_update_c_",
]

_hip_module = None
_hip_load_attempted = False


def load_hip():
    """Load the in-tree HIP extension (returns None on failure, caching the
    outcome)."""
    global _hip_module, _hip_load_attempted
    if _hip_load_attempted:
        return _hip_module
    _hip_load_attempted = True
    try:
        import evotorch_amd._C as _C  # built in-tree by setup_hip.py / __graft_entry__.build()

        _hip_module = _C
    except ImportError:
        _hip_module = None
    return _hip_module


def hip_available() -> bool:
    return load_hip() is not None


def hip_required():
    """The HIP extension, or a loud error on a GPU machine without it."""
    mod = load_hip()
    if mod is None:
        raise RuntimeError(
            "evotorch_amd._C (the gfx950 HIP extension) is not built, but a ROCm GPU "
            "tensor reached the op dispatch layer. Build it in-tree with "
            "`python setup_hip.py build_ext --inplace` (or __graft_entry__.build()). "
            "Refusing to silently fall back to eager PyTorch on GPU."
        )
    return mod


def _allow_eager_on_gpu() -> bool:
    # escape hatch for debugging only
    return os.environ.get("EVOTORCH_AMD_ALLOW_EAGER_GPU", "0") == "1"


_SPLITMIX_INC = 0x9E3779B97F4A7C15
_seed_fallback_counter = [0x1234ABCD]

def _splitmix64(x: int) -> int:
    x = (x + _SPLITMIX_INC) & 0xFFFFFFFFFFFFFFFF
    z = x
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    return (z ^ (z >> 31)) & 0x7FFFFFFFFFFFFFFF


def _seed_from_generator(generator: Optional[torch.Generator], device: torch.device) -> int:
    """Derive a fresh 63-bit seed deterministically from the generator's OWN
    state — no global bookkeeping, no device sync.

    CUDA generators: PyTorch tracks the philox (seed, offset) pair host-side,
    so get_offset()/set_offset() advance the generator without touching the
    GPU; the returned seed is splitmix64(initial_seed xor golden*offset).
    Two generators seeded identically produce identical seed sequences, and
    reseeding (manual_seed resets offset to 0) restarts the sequence.

    CPU generators: a single host randint draw (stateful, cheap).

    (An earlier version kept a module dict keyed by id(generator); after a
    generator was garbage collected its id could be reused by a NEW
    generator, silently resuming the dead one's counter. Generator-intrinsic
    state cannot alias.)"""
    if generator is not None:
        if generator.device.type == "cuda":
            off = int(generator.get_offset())
            generator.set_offset(off + 4)
            mixed = (int(generator.initial_seed()) ^ ((off + 1) * 0x9E3779B97F4A7C15)) & 0xFFFFFFFFFFFFFFFF
            return _splitmix64(mixed)
        return int(torch.randint(0, 2**62, (1,), generator=generator).item())
    _seed_fallback_counter[0] += 1
    return _splitmix64(_seed_fallback_counter[0])


# ============================================================================
# K1 — Gaussian population sampling
# ============================================================================


def sample_gaussian(
    out: torch.Tensor,
    mu: torch.Tensor,
    sigma: torch.Tensor,
    *,
    symmetric: bool = False,
    generator: Optional[torch.Generator] = None,
    seed: Optional[int] = None,
    row_offset: int = 0,
) -> torch.Tensor:
    """Fill `out` (N×L) with x = mu + sigma * z. With symmetric=True, rows
    [0, N/2) hold mu + sigma*z and rows [N/2, N) the mirrored mu - sigma*z
    (halves layout — see evotorch_amd/distributions.py docstring).

    Counter-addressed mode (`seed` given): row r of `out` (a direction row
    for symmetric sampling) draws from philox stream `row_offset + r` with
    the counter walking the row — the SAME values regardless of how the
    virtual population is partitioned into row blocks (sharding across
    ranks, or chunking in the streaming large-L gradient path), and
    identical on CPU (numpy philox reference) and GPU."""
    if out.ndim != 2:
        raise ValueError(f"expected a 2-D population, got shape {tuple(out.shape)}")
    n = out.shape[0]
    if symmetric and n % 2 != 0:
        raise ValueError("symmetric sampling requires even popsize")
    if seed is not None:
        if out.device.type == "cuda" and not _allow_eager_on_gpu():
            mod = hip_required()
            mod.sample_gaussian(out, mu.to(out.dtype), sigma.to(out.dtype), bool(symmetric), int(seed), int(row_offset))
            return out
        # philox-exact eager reference (matches the kernel bit-for-bit in fp32)
        from ..neuroevolution.philox_ref import philox_normals_2d

        rows = n // 2 if symmetric else n
        length = out.shape[1]
        z = philox_normals_2d(int(seed), int(row_offset), rows, length).to(device=out.device)
        mu32 = mu.to(torch.float32)
        sigma32 = sigma.to(torch.float32)
        plus = (mu32 + sigma32 * z).to(out.dtype)
        if symmetric:
            out[:rows] = plus
            out[rows:] = (2.0 * mu32 - plus.to(torch.float32)).to(out.dtype)
        else:
            out.copy_(plus)
        return out
    if out.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        drawn = _seed_from_generator(generator, out.device)
        mod.sample_gaussian(out, mu.to(out.dtype), sigma.to(out.dtype), bool(symmetric), drawn, 0)
        return out
    # eager reference
    if symmetric:
        half = out[: n // 2]
        half.normal_(generator=generator)
        half.mul_(sigma).add_(mu)
        torch.sub(2.0 * mu, half, out=out[n // 2 :])
    else:
        out.normal_(generator=generator)
        out.mul_(sigma).add_(mu)
    return out


def affine_from_noise(
    out: torch.Tensor,
    z: torch.Tensor,
    mu: torch.Tensor,
    sigma: torch.Tensor,
    *,
    symmetric: bool,
) -> torch.Tensor:
    """x = mu + sigma*z from PRE-GENERATED fp32 standard normals (z filled
    counter-addressed on a side stream, hidden behind evaluation and the
    gradient all-reduce). Bitwise-identical to `sample_gaussian` with the
    same seed/row_offset: both compute fmaf(sigma, z, mu) in fp32."""
    if out.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        mod.affine_from_noise(out, z, mu.to(out.dtype), sigma.to(out.dtype), bool(symmetric))
        return out
    n = out.shape[0]
    rows = n // 2 if symmetric else n
    mu32 = mu.to(torch.float32)
    s32 = sigma.to(torch.float32)
    plus = (mu32 + s32 * z.reshape(rows, -1)).to(out.dtype)
    if symmetric:
        out[:rows] = plus
        out[rows:] = (2.0 * mu32 - plus.to(torch.float32)).to(out.dtype)
    else:
        out.copy_(plus)
    return out


# ============================================================================
# K3 — fused ES gradient reductions (N×L → L)
# ============================================================================


def es_gradients(
    samples: torch.Tensor,
    mu: torch.Tensor,
    sigma: torch.Tensor,
    weights: torch.Tensor,
    *,
    symmetric: bool,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Plain/antithetic ES gradients of a separable Gaussian.

    Non-symmetric (reference distributions.py:548-579):
        mu_grad[l]    = Σ_i w_i · (x_il − μ_l)
        sigma_grad[l] = Σ_i w_i · ((x_il − μ_l)² − σ_l²) / σ_l

    Symmetric halves layout (reference distributions.py:708-773 with the
    interleaved layout mapped to halves):
        noise_d       = x_d − μ            (d < N/2; x_{d+N/2} = μ − noise_d)
        mu_grad[l]    = Σ_d (w⁺_d − w⁻_d)/2 · noise_dl
        sigma_grad[l] = Σ_d (w⁺_d + w⁻_d)/2 · (noise_dl² − σ_l²)/σ_l
    """
    if samples.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        return mod.es_gradients(samples, mu.to(samples.dtype), sigma.to(samples.dtype), weights.to(samples.dtype), bool(symmetric))
    if symmetric:
        n = samples.shape[0]
        d = n // 2
        noises = samples[:d].to(torch.float32) - mu.to(torch.float32)
        w_plus = weights[:d].to(torch.float32)
        w_minus = weights[d:].to(torch.float32)
        mu_grad = ((w_plus - w_minus) / 2.0) @ noises
        sigma32 = sigma.to(torch.float32)
        sigma_grad = ((w_plus + w_minus) / 2.0) @ ((noises**2 - sigma32**2) / sigma32)
    else:
        noises = samples.to(torch.float32) - mu.to(torch.float32)
        w = weights.to(torch.float32)
        mu_grad = w @ noises
        sigma32 = sigma.to(torch.float32)
        sigma_grad = w @ ((noises**2 - sigma32**2) / sigma32)
    return mu_grad.to(samples.dtype), sigma_grad.to(samples.dtype)


def snes_gradients(
    samples: torch.Tensor,
    mu: torch.Tensor,
    sigma: torch.Tensor,
    weights: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """SNES natural gradients (reference distributions.py:783-793):
        mu_grad[l]    = Σ_i w_i · (x_il − μ_l)
        sigma_grad[l] = Σ_i w_i · (z_il² − 1),  z = (x − μ)/σ
    """
    if samples.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        return mod.snes_gradients(samples, mu.to(samples.dtype), sigma.to(samples.dtype), weights.to(samples.dtype))
    noises = samples.to(torch.float32) - mu.to(torch.float32)
    raw = noises / sigma.to(torch.float32)
    w = weights.to(torch.float32)
    mu_grad = w @ noises
    sigma_grad = w @ (raw**2 - 1.0)
    return mu_grad.to(samples.dtype), sigma_grad.to(samples.dtype)


# ============================================================================
# K4 — optimizer steps (fused on device)
# ============================================================================


def clipup_step_(
    velocity: torch.Tensor,
    grad: torch.Tensor,
    *,
    step_size: float,
    max_speed: float,
    momentum: float = 0.9,
) -> torch.Tensor:
    """In-place ClipUp ascent step (Toklu et al. 2020; reference
    optimizers.py:231-357). Normalizes grad to unit L2, takes a step of
    `step_size`, adds momentum-scaled velocity, clips the velocity norm to
    `max_speed`. Returns the updated velocity (= the ascent step)."""
    if velocity.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        mod.clipup_step(velocity, grad, float(step_size), float(max_speed), float(momentum))
        return velocity
    g32 = grad.to(torch.float32)
    gnorm = torch.linalg.vector_norm(g32)
    step = g32 * (step_size / torch.clamp(gnorm, min=1e-30))
    v32 = velocity.to(torch.float32) * momentum + step
    vnorm = torch.linalg.vector_norm(v32)
    scale = torch.clamp(max_speed / torch.clamp(vnorm, min=1e-30), max=1.0)
    v32 = v32 * scale
    velocity.copy_(v32.to(velocity.dtype))
    return velocity


def fused_adam_step_(
    param_step_out: torch.Tensor,
    grad: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    *,
    step_count: int,
    stepsize: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    epsilon: float = 1e-8,
) -> torch.Tensor:
    """In-place Adam *ascent* step: updates moments m, v and writes the
    additive step into `param_step_out`."""
    if grad.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        mod.adam_step(param_step_out, grad, m, v, int(step_count), float(stepsize), float(beta1), float(beta2), float(epsilon))
        return param_step_out
    g32 = grad.to(torch.float32)
    m32 = m.to(torch.float32) * beta1 + (1.0 - beta1) * g32
    v32 = v.to(torch.float32) * beta2 + (1.0 - beta2) * g32 * g32
    m.copy_(m32.to(m.dtype))
    v.copy_(v32.to(v.dtype))
    mhat = m32 / (1.0 - beta1**step_count)
    vhat = v32 / (1.0 - beta2**step_count)
    param_step_out.copy_((stepsize * mhat / (vhat.sqrt() + epsilon)).to(param_step_out.dtype))
    return param_step_out


# ============================================================================
# K7 — NSGA-II non-dominated sorting
# ============================================================================


def domination_counts(utils: torch.Tensor) -> torch.Tensor:
    """Per solution: how many others dominate it. `utils` is (N, M) with
    higher-is-better columns."""
    if utils.device.type == "cuda" and not _allow_eager_on_gpu():
        return hip_required().domination_counts(utils).to(torch.int64)
    a = utils.unsqueeze(1)
    b = utils.unsqueeze(0)
    dom = (a >= b).all(dim=-1) & (a > b).any(dim=-1)
    return dom.sum(dim=0).to(torch.int64)


def pareto_ranks(utils: torch.Tensor, min_assigned: Optional[int] = None) -> torch.Tensor:
    """Front index per solution (0 = non-dominated front), computed on GPU
    by count + front peeling without the N x N matrix. With
    `min_assigned`, peeling stops once that many solutions hold final
    ranks (take_best(n) only needs the fronts crossing n); the rest get a
    shared beyond-last rank."""
    if utils.device.type == "cuda" and not _allow_eager_on_gpu():
        return hip_required().pareto_ranks(utils, int(min_assigned or 0))
    from ..core import _compute_pareto_ranks_eager

    ranks, _ = _compute_pareto_ranks_eager(utils, crowdsort=False, min_assigned=min_assigned)
    return ranks


# ============================================================================
# K5 — fused CMA-ES covariance update
# ============================================================================


def cma_update_c_(
    C: torch.Tensor,
    y: torch.Tensor,
    w_adj: torch.Tensor,
    p_c: torch.Tensor,
    hs_f: torch.Tensor,
    *,
    c1: float,
    cmu: float,
    cc: float,
) -> torch.Tensor:
    """In-place C = scale·C + c1·pc pcᵀ + cμ·Yᵀdiag(w)Y, exactly
    symmetric, with scale = 1 + c1·(1−hs)·cc·(2−cc) − c1 − cμ·Σw computed
    from DEVICE scalars (no host sync per generation).

    On GPU this is ONE fused pass over C (ops/hip/cma.hip — see that file
    for why fp32 VALU, not MFMA, is the right tool at CMA-ES shapes); the
    eager reference is the reference-parity torch chain (cmaes.py:547-553)
    plus explicit symmetrization."""
    wsum = w_adj.sum().to(torch.float32)
    if C.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        mod.cma_update_c(C, y.contiguous().to(torch.float32), w_adj.to(torch.float32), p_c.to(torch.float32),
                         hs_f.reshape(1).to(torch.float32), wsum.reshape(1), float(c1), float(cmu), float(cc))
        return C
    delta_hs = (1.0 - hs_f) * cc * (2.0 - cc)
    scale = 1.0 + c1 * delta_hs - c1 - cmu * wsum
    rank_mu = (y * w_adj.unsqueeze(-1)).T @ y
    rank_one = torch.outer(p_c, p_c)
    C.mul_(scale).add_(rank_one, alpha=c1).add_(rank_mu, alpha=cmu)
    C.copy_(0.5 * (C + C.T))
    return C


def potrf_tile_(A: torch.Tensor, info: Optional[torch.Tensor] = None) -> torch.Tensor:
    """In-place lower Cholesky of one SPD panel (n ≤ 128). On GPU this is
    ONE kernel with the panel LDS-resident (ops/hip/cma.hip
    potrf_panel_kernel) — the building block that removes rocSOLVER's
    small-potf2 chain from the blocked factorization
    (algorithms/cmaes.py _blocked_cholesky). `A` may be a strided view
    (a diagonal block of a larger matrix); its upper triangle is left
    untouched. A non-positive pivot at column j sets `info` (a device
    int32 scalar) to j+1 instead of raising — callers check once per
    factorization, not per panel.

    Eager/CPU reference: torch.linalg.cholesky (raises on failure, which
    the CPU callers rely on; it also zeroes the upper triangle, so
    callers must not rely on the upper half either way)."""
    if A.ndim != 2 or A.shape[0] != A.shape[1]:
        raise ValueError(f"expected a square panel, got {tuple(A.shape)}")
    if A.device.type == "cuda" and not _allow_eager_on_gpu():
        mod = hip_required()
        if info is None:
            info = torch.zeros(1, dtype=torch.int32, device=A.device)
        mod.potrf_tile(A, info)
        return A
    A.copy_(torch.linalg.cholesky(A))
    return A
