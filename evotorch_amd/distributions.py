"""Search distributions: SeparableGaussian (PGPE), SymmetricSeparableGaussian,
ExpSeparableGaussian (SNES), ExpGaussian (XNES).

Behavioral parity with the reference's `distributions.py`
(/root/reference/src/evotorch/distributions.py:40-1018), re-designed for
MI355X:

* `SeparableGaussian.sample` and the ES gradient reductions are dispatched
  through `evotorch_amd.ops` — on a ROCm device these run as hand-written
  CDNA4 HIP kernels (K1/K3 in SURVEY.md §2.9): counter-based philox
  sampling fused with the μ/σ affine map, and a single-pass fused
  (mu_grad, sigma_grad) reduction that reads the population matrix once.
* The antithetic population layout is **halves** (rows [0, n/2) are
  μ + σz, rows [n/2, n) their mirrors) rather than the reference's
  interleaved pairs (distributions.py:616-644) — contiguous halves are the
  natural layout for wave-coalesced HIP writes and for sharding across
  ranks. All in-package consumers (gradients, rank-sharded sampling) use
  the same convention.
"""

import math
from typing import Iterable, Optional, Type, NamedTuple

import torch

from .utils import Device, DType, TensorMakerMixin, to_torch_dtype
from .utils.ranking import rank

__all__ = [
    "GradsWithSamples",
    "GradsWithFitnesses",
    "GradsWithSamplesAndFitnesses",
    "Distribution",
    "SeparableGaussian",
    "SymmetricSeparableGaussian",
    "ExpSeparableGaussian",
    "ExpGaussian",
    "make_functional_sampler",
    "make_functional_grad_estimator",
]


class Distribution(TensorMakerMixin):
    """Base class: a parameter dict + dtype/device; `sample(out=...)`
    in-place fill; `compute_gradients` = rank then `_compute_gradients`;
    `update_parameters` returns a new Distribution following per-parameter
    learning rates or optimizers (reference distributions.py:40-411)."""

    MANDATORY_PARAMETERS: set = set()
    OPTIONAL_PARAMETERS: set = set()
    PARAMETER_NDIMS: dict = {}

    functional_sample = NotImplemented

    def __init__(self, *, solution_length: int, parameters: dict, dtype: Optional[DType] = None, device: Optional[Device] = None):
        for k in parameters:
            if k not in self.MANDATORY_PARAMETERS and k not in self.OPTIONAL_PARAMETERS:
                raise ValueError(f"{type(self).__name__} got an unrecognized parameter {k!r}")
        for k in self.MANDATORY_PARAMETERS:
            if k not in parameters:
                raise ValueError(f"{type(self).__name__} is missing the mandatory parameter {k!r}")
        first_tensor = next(v for v in parameters.values() if isinstance(v, torch.Tensor))
        self._dtype = to_torch_dtype(dtype) if dtype is not None else first_tensor.dtype
        self._device = torch.device(device) if device is not None else first_tensor.device
        self._parameters = {
            k: (v.to(dtype=self._dtype, device=self._device) if isinstance(v, torch.Tensor) else v) for k, v in parameters.items()
        }
        self._solution_length = int(solution_length)

    @property
    def solution_length(self) -> int:
        return self._solution_length

    @property
    def dtype(self) -> torch.dtype:
        return self._dtype

    @property
    def device(self) -> torch.device:
        return self._device

    @property
    def parameters(self) -> dict:
        return self._parameters

    # -- sampling -----------------------------------------------------------

    def sample(
        self,
        num_solutions: Optional[int] = None,
        *,
        out: Optional[torch.Tensor] = None,
        generator=None,
    ) -> torch.Tensor:
        """Sample into `out` (in-place fill) or into a fresh tensor of
        `num_solutions` rows. `generator` may be a torch.Generator or an
        object exposing one via `.generator` (e.g. a Problem)."""
        if (num_solutions is None) == (out is None):
            raise ValueError("Provide exactly one of num_solutions, out")
        if out is None:
            out = torch.empty((int(num_solutions), self._solution_length), dtype=self._dtype, device=self._device)
        if generator is not None and not isinstance(generator, torch.Generator):
            generator = getattr(generator, "generator", None)
        self._fill(out, generator=generator)
        return out

    def _fill(self, out: torch.Tensor, *, generator: Optional[torch.Generator] = None):
        raise NotImplementedError

    # -- gradients ----------------------------------------------------------

    def compute_gradients(
        self,
        samples: torch.Tensor,
        fitnesses: torch.Tensor,
        *,
        objective_sense: str,
        ranking_method: Optional[str] = None,
    ) -> dict:
        if objective_sense not in ("min", "max"):
            raise ValueError(f"objective_sense must be 'min' or 'max', got {objective_sense!r}")
        weights = rank(fitnesses, ranking_method or "raw", higher_is_better=(objective_sense == "max"))
        weights = weights.to(dtype=samples.dtype, device=samples.device)
        return self._compute_gradients(samples, weights, ranking_used=(ranking_method or "raw"))

    def _compute_gradients(self, samples: torch.Tensor, weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        raise NotImplementedError

    # -- SPMD shard-gradient protocol ----------------------------------------
    #
    # The sharded path (Problem._sample_and_compute_gradients_sharded) splits
    # the weight-dependent gradient into three phases so every normalization
    # uses GLOBAL quantities (a per-shard normalization would silently change
    # the merged gradient — ADVICE.md round-1 medium finding):
    #
    #   w_all = prepare_weights_global(all_utils, ranking)   # global center/normalize
    #   sums  = partial_grad_sums(local_samples, w_local)    # raw sums, linear in w
    #   <all-reduce the sums across ranks>
    #   grads = finalize_shard_gradients(sums, w_all, ranking)  # global divisors
    #
    # partial_grad_sums MUST be additive over row blocks of the population so
    # the all-reduced sum equals the single-process gradient exactly.

    def prepare_weights_global(self, all_weights: torch.Tensor, ranking_used: Optional[str]) -> torch.Tensor:
        """Weight preprocessing computed over the GLOBAL utility vector."""
        return all_weights

    def partial_grad_sums(self, samples: torch.Tensor, weights: torch.Tensor) -> dict:
        """Raw, un-normalized gradient sums over a row block (linear in the
        weights; additive across blocks)."""
        raise NotImplementedError

    def finalize_shard_gradients(self, sums: dict, all_weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        """Apply weight-dependent normalization using GLOBAL weights."""
        return sums

    # -- updates ------------------------------------------------------------

    def _follow_gradient(self, param_name: str, grad: torch.Tensor, *, learning_rates: Optional[dict] = None, optimizers: Optional[dict] = None) -> torch.Tensor:
        """The additive step for one parameter: optimizer.ascent(grad) if an
        optimizer is registered for it, else learning_rate * grad."""
        if optimizers is not None and param_name in optimizers and optimizers[param_name] is not None:
            return optimizers[param_name].ascent(grad)
        if learning_rates is not None and param_name in learning_rates:
            return float(learning_rates[param_name]) * grad
        raise ValueError(f"No learning rate or optimizer given for parameter {param_name!r}")

    def _stepped_param(
        self, param_name: str, current: torch.Tensor, grad: torch.Tensor, learning_rates: Optional[dict], optimizers: Optional[dict]
    ) -> torch.Tensor:
        """current + step WITHOUT materializing the step tensor when a
        plain learning rate drives the parameter (one fused add), and
        without the optimizer's defensive clone (the sum is the only
        consumer). At extreme L every L-sized temporary counts — the
        billion-parameter streaming mode is bounded by how many of these
        are alive at the update's peak."""
        if optimizers is not None and param_name in optimizers and optimizers[param_name] is not None:
            return current + optimizers[param_name].ascent(grad, cloned_result=False)
        if learning_rates is not None and param_name in learning_rates:
            return torch.add(current, grad, alpha=float(learning_rates[param_name]))
        raise ValueError(f"No learning rate or optimizer given for parameter {param_name!r}")

    def update_parameters(self, gradients: dict, *, learning_rates: Optional[dict] = None, optimizers: Optional[dict] = None) -> "Distribution":
        raise NotImplementedError

    def modified_copy(self, *, dtype: Optional[DType] = None, device: Optional[Device] = None, **parameters) -> "Distribution":
        new_params = dict(self._parameters)
        new_params.update(parameters)
        return type(self)(new_params, dtype=dtype or self._dtype, device=device or self._device)

    def to(self, device: Device) -> "Distribution":
        device = torch.device(device)
        if device == self._device:
            return self
        params = {k: (v.to(device) if isinstance(v, torch.Tensor) else v) for k, v in self._parameters.items()}
        return type(self)(params, dtype=self._dtype, device=device)

    def relative_entropy(self, other: "Distribution") -> float:
        raise NotImplementedError

    def __repr__(self) -> str:
        return f"<{type(self).__name__} solution_length={self._solution_length} dtype={self._dtype} device={self._device}>"


class SeparableGaussian(Distribution):
    """Diagonal Gaussian (PGPE family). Parameters: mu, sigma (both length
    L). Optional: divide_mu_grad_by / divide_sigma_grad_by
    ('num_solutions' | 'num_directions' | 'total_weight' | 'weight_stdev'),
    parenthood_ratio (CEM elite mode).

    Reference parity: distributions.py:413-614.
    """

    MANDATORY_PARAMETERS = {"mu", "sigma"}
    OPTIONAL_PARAMETERS = {"divide_mu_grad_by", "divide_sigma_grad_by", "parenthood_ratio"}
    PARAMETER_NDIMS = {"mu": 1, "sigma": 1}

    def __init__(self, parameters: dict, *, solution_length: Optional[int] = None, dtype=None, device=None):
        missing = self.MANDATORY_PARAMETERS - set(parameters)
        if missing:
            raise ValueError(f"{type(self).__name__} is missing mandatory parameter(s): {sorted(missing)}")
        unknown = set(parameters) - self.MANDATORY_PARAMETERS - self.OPTIONAL_PARAMETERS
        if unknown:
            raise ValueError(f"{type(self).__name__} got unknown parameter(s): {sorted(unknown)}")
        (mu_len,) = parameters["mu"].shape
        (sigma_len,) = parameters["sigma"].shape
        if mu_len != sigma_len:
            raise ValueError(f"mu and sigma lengths differ: {mu_len} vs {sigma_len}")
        if solution_length is not None and int(solution_length) != mu_len:
            raise ValueError(f"solution_length {solution_length} != len(mu) {mu_len}")
        super().__init__(solution_length=mu_len, parameters=parameters, dtype=dtype, device=device)

    @property
    def mu(self) -> torch.Tensor:
        return self._parameters["mu"]

    @mu.setter
    def mu(self, value: Iterable):
        self._parameters["mu"] = torch.as_tensor(value, dtype=self._dtype, device=self._device)

    @property
    def sigma(self) -> torch.Tensor:
        return self._parameters["sigma"]

    @sigma.setter
    def sigma(self, value: Iterable):
        self._parameters["sigma"] = torch.as_tensor(value, dtype=self._dtype, device=self._device)

    _symmetric = False

    def _fill(self, out: torch.Tensor, *, generator: Optional[torch.Generator] = None):
        from . import ops

        ops.sample_gaussian(out, self.mu, self.sigma, symmetric=self._symmetric, generator=generator)

    def fill_counter_addressed(self, out: torch.Tensor, *, seed: int, row_offset: int = 0):
        """Sample a row-block of the virtual population: direction row d
        draws from philox stream `row_offset + d`, identical however the
        population is partitioned (and identical on CPU and GPU). This is
        what makes (a) the streaming large-L gradient path able to
        REGENERATE noise in pass 2 instead of storing the N×L population,
        and (b) SPMD sharded sampling produce the same virtual population
        for any world size."""
        from . import ops

        ops.sample_gaussian(out, self.mu, self.sigma, symmetric=self._symmetric, seed=seed, row_offset=row_offset)

    def fill_from_noise(self, out: torch.Tensor, z: torch.Tensor):
        """Apply x = mu + sigma*z from pre-generated standard normals
        (side-stream overlap path) — bitwise equal to fill_counter_addressed
        when z holds the same counter-addressed philox draws."""
        from . import ops

        ops.affine_from_noise(out, z, self.mu, self.sigma, symmetric=self._symmetric)

    def accumulate_elite_sums_streamed(self, chunk_iter, is_elite: torch.Tensor):
        """Masked (Σx, Σx²) over elite rows, chunk by chunk (fp64 accum).
        `is_elite` flags LOCAL rows; shards all-reduce the returned sums."""
        sum_x = torch.zeros_like(self.mu, dtype=torch.float64)
        sum_x2 = torch.zeros_like(self.mu, dtype=torch.float64)
        for values_chunk, row0, rows in chunk_iter:
            mask = is_elite[row0 : row0 + rows]
            if bool(mask.any()):
                selected = values_chunk[mask].to(torch.float64)
                sum_x += selected.sum(dim=0)
                sum_x2 += (selected**2).sum(dim=0)
        return sum_x, sum_x2

    def finalize_elite_gradients(self, sum_x: torch.Tensor, sum_x2: torch.Tensor, num_elites: int) -> dict:
        """Turn global elite (Σx, Σx², k) into the CEM gradients —
        algebraically identical to `_elite_gradients`."""
        k = max(int(num_elites), 1)
        mean = sum_x / k
        var = (sum_x2 - k * mean**2) / max(k - 1, 1)
        std = torch.sqrt(torch.clamp(var, min=0.0))
        return {
            "mu": (mean - self.mu.to(torch.float64)).to(self.mu.dtype),
            "sigma": (std - self.sigma.to(torch.float64)).to(self.sigma.dtype),
        }

    @classmethod
    def functional_sample(cls, num_solutions: int, parameters: dict) -> torch.Tensor:
        """Pure out-of-place sampling (vmap-safe under
        randomness='different'; reference distributions.py:431)."""
        mu = parameters["mu"]
        sigma = parameters["sigma"]
        n = int(num_solutions)
        z = torch.randn((n,) + mu.shape[-1:], dtype=mu.dtype, device=mu.device)
        return mu.unsqueeze(-2) + sigma.unsqueeze(-2) * z

    def _centered_weights(self, weights: torch.Tensor, ranking_used: Optional[str]) -> torch.Tensor:
        if ranking_used not in ("centered", "normalized"):
            weights = weights - weights.mean()
        return weights

    def _divide_grad(self, param_name: str, grad: torch.Tensor, weights: torch.Tensor) -> torch.Tensor:
        option = self._parameters.get(f"divide_{param_name}_grad_by", None)
        if option is None:
            return grad
        if option == "num_solutions":
            return grad / weights.shape[0]
        if option == "num_directions":
            return grad / (weights.shape[0] // 2)
        if option == "total_weight":
            return grad / weights.abs().sum()
        if option == "weight_stdev":
            return grad / weights.std()
        raise ValueError(f"Unrecognized grad divisor {option!r}")

    def _elite_gradients(self, samples: torch.Tensor, weights: torch.Tensor) -> dict:
        """CEM-style parenthood-ratio gradients (reference
        distributions.py:538)."""
        num_samples = samples.shape[0]
        num_elites = math.floor(num_samples * float(self._parameters["parenthood_ratio"]))
        elites = samples[weights.argsort(descending=True)[:num_elites]]
        return {
            "mu": elites.mean(dim=0) - self.mu,
            "sigma": elites.std(dim=0, unbiased=True) - self.sigma,
        }

    def _compute_gradients(self, samples: torch.Tensor, weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        if "parenthood_ratio" in self._parameters:
            return self._elite_gradients(samples, weights)
        from . import ops

        weights = self._centered_weights(weights, ranking_used)
        mu_grad, sigma_grad = ops.es_gradients(samples, self.mu, self.sigma, weights, symmetric=False)
        return {
            "mu": self._divide_grad("mu", mu_grad, weights),
            "sigma": self._divide_grad("sigma", sigma_grad, weights),
        }

    # -- SPMD shard-gradient protocol (see Distribution) ---------------------

    def prepare_weights_global(self, all_weights: torch.Tensor, ranking_used: Optional[str]) -> torch.Tensor:
        return self._centered_weights(all_weights, ranking_used)

    def partial_grad_sums(self, samples: torch.Tensor, weights: torch.Tensor) -> dict:
        from . import ops

        mu_grad, sigma_grad = ops.es_gradients(samples, self.mu, self.sigma, weights, symmetric=self._symmetric)
        return {"mu": mu_grad.to(torch.float32), "sigma": sigma_grad.to(torch.float32)}

    def finalize_shard_gradients(self, sums: dict, all_weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        return {
            "mu": self._divide_grad("mu", sums["mu"].to(self.mu.dtype), all_weights),
            "sigma": self._divide_grad("sigma", sums["sigma"].to(self.sigma.dtype), all_weights),
        }

    def update_parameters(self, gradients: dict, *, learning_rates: Optional[dict] = None, optimizers: Optional[dict] = None) -> "SeparableGaussian":
        new_mu = self._stepped_param("mu", self.mu, gradients["mu"], learning_rates, optimizers)
        new_sigma = self._stepped_param("sigma", self.sigma, gradients["sigma"], learning_rates, optimizers)
        return self.modified_copy(mu=new_mu, sigma=new_sigma)

    def relative_entropy(self, other: "SeparableGaussian") -> float:
        """KL(self || other) for diagonal Gaussians (reference
        distributions.py:598)."""
        cov0 = self.sigma.pow(2.0)
        cov1 = other.sigma.pow(2.0)
        mu_delta = other.mu - self.mu
        trace = torch.sum(cov0 / cov1)
        scaled_mu = torch.sum(mu_delta.pow(2.0) / cov1)
        log_det = torch.sum(torch.log(cov1)) - torch.sum(torch.log(cov0))
        return float(0.5 * (trace - self._solution_length + scaled_mu + log_det))


class SymmetricSeparableGaussian(SeparableGaussian):
    """Antithetic diagonal Gaussian (PGPE default). Population layout is
    HALVES: rows [0, n/2) are μ + σz and rows [n/2, n) are the mirrored
    μ − σz (layout deviation from the reference documented in the module
    docstring). Gradients follow the (f⁺−f⁻)/2, (f⁺+f⁻)/2 pair formulas of
    reference distributions.py:708-773."""

    _symmetric = True

    def sample(self, num_solutions: Optional[int] = None, *, out: Optional[torch.Tensor] = None, generator=None) -> torch.Tensor:
        if num_solutions is not None and int(num_solutions) % 2 != 0:
            raise ValueError(f"Symmetric sampling needs an even population, got {num_solutions}")
        if out is not None and out.shape[0] % 2 != 0:
            raise ValueError(f"Symmetric sampling needs an even population, got {out.shape[0]}")
        return super().sample(num_solutions, out=out, generator=generator)

    @classmethod
    def functional_sample(cls, num_solutions: int, parameters: dict) -> torch.Tensor:
        n = int(num_solutions)
        if n % 2 != 0:
            raise ValueError(f"Symmetric sampling needs an even population, got {n}")
        mu = parameters["mu"]
        sigma = parameters["sigma"]
        z = torch.randn((n // 2,) + mu.shape[-1:], dtype=mu.dtype, device=mu.device)
        plus = mu.unsqueeze(-2) + sigma.unsqueeze(-2) * z
        minus = 2.0 * mu.unsqueeze(-2) - plus
        return torch.cat([plus, minus], dim=-2)

    def _compute_gradients(self, samples: torch.Tensor, weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        if "parenthood_ratio" in self._parameters:
            return self._elite_gradients(samples, weights)
        from . import ops

        weights = self._centered_weights(weights, ranking_used)
        mu_grad, sigma_grad = ops.es_gradients(samples, self.mu, self.sigma, weights, symmetric=True)
        return {
            "mu": self._divide_grad("mu", mu_grad, weights),
            "sigma": self._divide_grad("sigma", sigma_grad, weights),
        }


class ExpSeparableGaussian(SeparableGaussian):
    """SNES distribution: raw-noise natural gradient + exponential sigma
    update σ' = σ·exp(½·step) (reference distributions.py:776-811)."""

    MANDATORY_PARAMETERS = {"mu", "sigma"}
    OPTIONAL_PARAMETERS = set()
    PARAMETER_NDIMS = {"mu": 1, "sigma": 1}

    _symmetric = False

    def _compute_gradients(self, samples: torch.Tensor, weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        if ranking_used != "nes":
            weights = weights / weights.abs().sum()
        from . import ops

        mu_grad, sigma_grad = ops.snes_gradients(samples, self.mu, self.sigma, weights)
        return {"mu": mu_grad, "sigma": sigma_grad}

    # -- SPMD shard-gradient protocol: SNES normalizes by the GLOBAL
    # |w| sum (the round-1 per-shard normalization bug) and applies no
    # divisors afterwards.

    def prepare_weights_global(self, all_weights: torch.Tensor, ranking_used: Optional[str]) -> torch.Tensor:
        if ranking_used != "nes":
            all_weights = all_weights / all_weights.abs().sum()
        return all_weights

    def partial_grad_sums(self, samples: torch.Tensor, weights: torch.Tensor) -> dict:
        from . import ops

        mu_grad, sigma_grad = ops.snes_gradients(samples, self.mu, self.sigma, weights)
        return {"mu": mu_grad.to(torch.float32), "sigma": sigma_grad.to(torch.float32)}

    def finalize_shard_gradients(self, sums: dict, all_weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        return {"mu": sums["mu"].to(self.mu.dtype), "sigma": sums["sigma"].to(self.sigma.dtype)}

    def update_parameters(self, gradients: dict, *, learning_rates: Optional[dict] = None, optimizers: Optional[dict] = None) -> "ExpSeparableGaussian":
        new_mu = self._stepped_param("mu", self.mu, gradients["mu"], learning_rates, optimizers)
        # the exponential sigma step owns a fresh tensor (it never aliases
        # optimizer state here), so transform it in place
        step = self._follow_gradient("sigma", gradients["sigma"], learning_rates=learning_rates, optimizers=optimizers)
        new_sigma = self.sigma * step.mul_(0.5).exp_()
        return self.modified_copy(mu=new_mu, sigma=new_sigma)


class ExpGaussian(Distribution):
    """XNES full-covariance distribution in exponential coordinates.
    Parameters: mu (L), sigma = A (L×L factor, covariance = AᵀA), with
    sigma_inv tracked for stability. Reference parity:
    distributions.py:813-1017. The L×L maps run as rocBLAS GEMMs (library
    GEMMs are the right tool for plain dense products; only fused ops get
    hand-written kernels)."""

    MANDATORY_PARAMETERS = {"mu", "sigma"}
    OPTIONAL_PARAMETERS = {"sigma_inv"}
    PARAMETER_NDIMS = {"mu": 1, "sigma": 2, "sigma_inv": 2}

    def __init__(self, parameters: dict, *, solution_length: Optional[int] = None, dtype=None, device=None):
        parameters = dict(parameters)
        if parameters["sigma"].ndim == 1:
            parameters["sigma"] = torch.diag(parameters["sigma"])
        if "sigma_inv" not in parameters:
            parameters["sigma_inv"] = torch.inverse(parameters["sigma"])
        (mu_len,) = parameters["mu"].shape
        super().__init__(solution_length=mu_len, parameters=parameters, dtype=dtype, device=device)
        self.eye = torch.eye(mu_len, dtype=self._dtype, device=self._device)

    @property
    def mu(self) -> torch.Tensor:
        return self._parameters["mu"]

    @property
    def sigma(self) -> torch.Tensor:
        return self._parameters["sigma"]

    @property
    def sigma_inv(self) -> torch.Tensor:
        return self._parameters["sigma_inv"]

    A = sigma
    A_inv = sigma_inv

    @property
    def cov(self) -> torch.Tensor:
        return self.sigma.T @ self.sigma

    def to_global_coordinates(self, z: torch.Tensor) -> torch.Tensor:
        return self.mu.unsqueeze(0) + z @ self.sigma.T

    def to_local_coordinates(self, x: torch.Tensor) -> torch.Tensor:
        return (x - self.mu.unsqueeze(0)) @ self.sigma_inv.T

    def _fill(self, out: torch.Tensor, *, generator: Optional[torch.Generator] = None):
        out.normal_(generator=generator)
        out.copy_(self.to_global_coordinates(out))

    def fill_counter_addressed(self, out: torch.Tensor, *, seed: int, row_offset: int = 0):
        """Counter-addressed full-covariance sampling: row r draws its
        standard normals from philox stream `row_offset + r` and maps them
        through A — world-size-invariant like the separable family (the
        SPMD sharded path requires this entry point)."""
        from . import ops

        z = torch.empty_like(out)
        zeros = torch.zeros_like(self.mu)
        ones = torch.ones_like(self.mu)
        ops.sample_gaussian(z, zeros, ones, symmetric=False, seed=seed, row_offset=row_offset)
        out.copy_(self.to_global_coordinates(z))

    def _compute_gradients(self, samples: torch.Tensor, weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        z = self.to_local_coordinates(samples)
        if ranking_used not in ("centered", "normalized"):
            weights = weights - weights.mean()
        d_grad = torch.mv(z.T, weights)
        # Σ_i w_i (z_i z_iᵀ − I): a weighted syrk — (zᵀ diag(w) z) − Σw·I,
        # MFMA-shaped (K5/K6 in SURVEY.md §2.9), served by rocBLAS.
        M_grad = (z * weights.unsqueeze(-1)).T @ z - weights.sum() * self.eye
        return {"d": d_grad, "M": M_grad}

    # -- SPMD shard-gradient protocol: both d and M are linear in w, so raw
    # local sums all-reduce exactly; centering uses the GLOBAL mean.

    def prepare_weights_global(self, all_weights: torch.Tensor, ranking_used: Optional[str]) -> torch.Tensor:
        if ranking_used not in ("centered", "normalized"):
            all_weights = all_weights - all_weights.mean()
        return all_weights

    def partial_grad_sums(self, samples: torch.Tensor, weights: torch.Tensor) -> dict:
        z = self.to_local_coordinates(samples)
        d_grad = torch.mv(z.T, weights)
        M_grad = (z * weights.unsqueeze(-1)).T @ z - weights.sum() * self.eye
        return {"d": d_grad.to(torch.float32), "M": M_grad.to(torch.float32)}

    def finalize_shard_gradients(self, sums: dict, all_weights: torch.Tensor, ranking_used: Optional[str]) -> dict:
        dt = self.mu.dtype
        return {"d": sums["d"].to(dt), "M": sums["M"].to(dt)}

    def update_parameters(self, gradients: dict, *, learning_rates: Optional[dict] = None, optimizers: Optional[dict] = None) -> "ExpGaussian":
        learning_rates = dict(learning_rates or {})
        optimizers = dict(optimizers or {})
        if "d" not in learning_rates and "mu" in learning_rates:
            learning_rates["d"] = learning_rates["mu"]
        if "M" not in learning_rates and "sigma" in learning_rates:
            learning_rates["M"] = learning_rates["sigma"]
        if "d" not in optimizers and "mu" in optimizers:
            optimizers["d"] = optimizers["mu"]
        update_d = self._follow_gradient("d", gradients["d"], learning_rates=learning_rates, optimizers=optimizers)
        update_M = self._follow_gradient("M", gradients["M"], learning_rates=learning_rates, optimizers=optimizers)
        new_mu = self.mu + torch.mv(self.sigma, update_d)
        new_A = self.sigma @ torch.matrix_exp(0.5 * update_M)
        new_A_inv = torch.matrix_exp(-0.5 * update_M) @ self.sigma_inv
        return self.modified_copy(mu=new_mu, sigma=new_A, sigma_inv=new_A_inv)


# ----------------------------------------------------------------------------
# Functional bridge (reference distributions.py:1023-1622)
# ----------------------------------------------------------------------------


class GradsWithSamples(NamedTuple):
    """Return type option of functional grad estimators (reference
    distributions.py: GradsWithSamples)."""

    grads: tuple
    samples: torch.Tensor


class GradsWithFitnesses(NamedTuple):
    grads: tuple
    fitnesses: torch.Tensor


class GradsWithSamplesAndFitnesses(NamedTuple):
    grads: tuple
    samples: torch.Tensor
    fitnesses: torch.Tensor


def make_functional_sampler(distribution_class: Type[Distribution], *, required_parameters: Iterable[str], fixed_parameters: Optional[dict] = None):
    """Wrap a Distribution class as a pure sampling function
    ``sample(num_solutions, param0, param1, ...) -> samples`` batchable via
    `expects_ndim` (leading batch dims on the parameters run independent
    batched samplers under vmap)."""
    from .decorators import expects_ndim

    required_parameters = list(required_parameters)
    fixed_parameters = dict(fixed_parameters or {})
    ndims = tuple([None] + [distribution_class.PARAMETER_NDIMS.get(p, 1) for p in required_parameters])

    def _sample(num_solutions: int, *args) -> torch.Tensor:
        params = dict(zip(required_parameters, args))
        params.update(fixed_parameters)
        fs = getattr(distribution_class, "functional_sample", NotImplemented)
        if fs is not NotImplemented and fs is not None:
            return fs(num_solutions, params)
        dist = distribution_class(params)
        return dist.sample(int(num_solutions))

    batched = expects_ndim(_sample, ndims, randomness="different")

    def sample(num_solutions: int, *args) -> torch.Tensor:
        return batched(num_solutions, *args)

    sample.__name__ = f"functional_sampler_of_{distribution_class.__name__}"
    return sample


def make_functional_grad_estimator(
    distribution_class: Type[Distribution],
    *,
    required_parameters: Iterable[str],
    fixed_parameters: Optional[dict] = None,
    objective_sense: str = "max",
    ranking_method: Optional[str] = None,
):
    """Wrap a Distribution class as a pure gradient estimator
    ``grad(samples, fitnesses, param0, ...) -> (grad0, grad1, ...)``."""
    from .decorators import expects_ndim

    required_parameters = list(required_parameters)
    fixed_parameters = dict(fixed_parameters or {})
    ndims = tuple([2, 1] + [distribution_class.PARAMETER_NDIMS.get(p, 1) for p in required_parameters])

    def _estimate(samples: torch.Tensor, fitnesses: torch.Tensor, *args):
        params = dict(zip(required_parameters, args))
        params.update(fixed_parameters)
        dist = distribution_class(params)
        grads = dist.compute_gradients(samples, fitnesses, objective_sense=objective_sense, ranking_method=ranking_method)
        return tuple(grads[k] for k in sorted(grads.keys()))

    batched = expects_ndim(_estimate, ndims)

    def estimate(samples: torch.Tensor, fitnesses: torch.Tensor, *args):
        return batched(samples, fitnesses, *args)

    estimate.__name__ = f"functional_grad_estimator_of_{distribution_class.__name__}"
    return estimate
