"""CMA-ES: native torch implementation (Hansen 2016 tutorial formulas).

Reference parity: /root/reference/src/evotorch/algorithms/cmaes.py:90-567 —
rank-μ + rank-1 covariance update, active CMA, separable mode, CSA
step-size control, and Cholesky (not eigen) decomposition for sampling.

MI355X notes: sampling is X = m + σ·(Z Aᵀ) and the rank-μ update is
(√w ⊙ Y)ᵀ(√w ⊙ Y) — both plain dense GEMMs served by rocBLAS (K5 in
SURVEY.md §2.9; library GEMMs are the right tool for unfused dense
products). The Cholesky factorization (d ≤ ~8k) runs on rocSOLVER through
`torch.linalg.cholesky`. The whitened evolution path uses z_w = Σ w_i z_i
directly (A⁻¹y_w = z_w), avoiding any triangular solve.
"""

import math
from typing import Optional

import torch

from .. import ops
from ..core import Problem, SolutionBatch
from ..utils import RealOrVector, to_stdev_init
from ..utils.misc import ensure_tensor_length_and_dtype
from .searchalgorithm import SearchAlgorithm, SinglePopulationAlgorithmMixin

__all__ = ["CMAES"]


class CMAES(SearchAlgorithm, SinglePopulationAlgorithmMixin):
    def __init__(
        self,
        problem: Problem,
        *,
        stdev_init: Optional[RealOrVector] = None,
        radius_init: Optional[RealOrVector] = None,
        popsize: Optional[int] = None,
        center_init: Optional[RealOrVector] = None,
        c_m: float = 1.0,
        c_sigma: Optional[float] = None,
        c_sigma_ratio: float = 1.0,
        damp_sigma: Optional[float] = None,
        damp_sigma_ratio: float = 1.0,
        c_c: Optional[float] = None,
        c_c_ratio: float = 1.0,
        c_1: Optional[float] = None,
        c_1_ratio: float = 1.0,
        c_mu: Optional[float] = None,
        c_mu_ratio: float = 1.0,
        active: bool = True,
        csa_squared: bool = False,
        stdev_min: Optional[float] = None,
        stdev_max: Optional[float] = None,
        separable: bool = False,
        limit_C_decomposition: bool = True,
        obj_index: Optional[int] = None,
    ):
        problem.ensure_numeric()
        problem.ensure_unbounded()
        SearchAlgorithm.__init__(
            self,
            problem,
            center=lambda: self._m,
            stdev=lambda: self._sigma * (self._sigma_diag() if self._separable else torch.sqrt(torch.diagonal(self._C))),
            sigma=lambda: float(self._sigma),
            mean_eval=self._get_mean_eval,
        )
        n = problem.solution_length
        self._n = n
        self._obj_index = 0 if obj_index is None else int(obj_index)
        device = problem.device
        dtype = problem.dtype

        stdev_spec = to_stdev_init(solution_length=n, stdev_init=stdev_init, radius_init=radius_init)
        sigma_vec = ensure_tensor_length_and_dtype(stdev_spec, n, dtype, about="stdev_init", device=device)
        self._sigma = torch.as_tensor(float(sigma_vec.mean()), dtype=dtype, device=device)
        init_scale = (sigma_vec / self._sigma) ** 2  # fold anisotropy into C

        if center_init is None:
            self._m = problem.generate_values(1).reshape(-1)
        else:
            self._m = ensure_tensor_length_and_dtype(center_init, n, dtype, about="center_init", device=device)

        if popsize is None:
            popsize = int(4 + math.floor(3 * math.log(n)))
        self._popsize = int(popsize)
        mu = self._popsize // 2

        # recombination weights (Hansen 2016 eq. 49-53; active CMA keeps the
        # negative tail)
        raw = torch.tensor([math.log((self._popsize + 1) / 2.0) - math.log(i + 1) for i in range(self._popsize)], dtype=dtype, device=device)
        pos = raw[:mu]
        self._mu_eff = float(pos.sum() ** 2 / (pos**2).sum())
        w_pos = pos / pos.sum()

        self._c_sigma = c_sigma if c_sigma is not None else c_sigma_ratio * (self._mu_eff + 2.0) / (n + self._mu_eff + 5.0)
        self._damp_sigma = (
            damp_sigma
            if damp_sigma is not None
            else damp_sigma_ratio * (1.0 + 2.0 * max(0.0, math.sqrt((self._mu_eff - 1.0) / (n + 1.0)) - 1.0) + self._c_sigma)
        )
        self._c_c = c_c if c_c is not None else c_c_ratio * (4.0 + self._mu_eff / n) / (n + 4.0 + 2.0 * self._mu_eff / n)
        alpha_cov = 2.0
        self._c_1 = c_1 if c_1 is not None else c_1_ratio * alpha_cov / ((n + 1.3) ** 2 + self._mu_eff)
        self._c_mu = (
            c_mu
            if c_mu is not None
            else c_mu_ratio
            * min(
                1.0 - self._c_1,
                alpha_cov * (0.25 + self._mu_eff + 1.0 / self._mu_eff - 2.0) / ((n + 2.0) ** 2 + alpha_cov * self._mu_eff / 2.0),
            )
        )
        self._active = bool(active)
        if self._active:
            neg = raw[mu:]
            mu_eff_neg = float(neg.sum() ** 2 / (neg**2).sum())
            alpha_mu_neg = 1.0 + self._c_1 / self._c_mu
            alpha_mueff_neg = 1.0 + 2.0 * mu_eff_neg / (self._mu_eff + 2.0)
            alpha_posdef_neg = (1.0 - self._c_1 - self._c_mu) / (n * self._c_mu)
            w_neg = neg / neg.abs().sum() * min(alpha_mu_neg, alpha_mueff_neg, alpha_posdef_neg)
            self._weights = torch.cat([w_pos, w_neg])
        else:
            self._weights = torch.cat([w_pos, torch.zeros(self._popsize - mu, dtype=dtype, device=device)])
        self._mu = mu
        self._c_m = float(c_m)
        self._csa_squared = bool(csa_squared)
        self._stdev_min = stdev_min
        self._stdev_max = stdev_max
        self._separable = bool(separable)

        self._p_sigma = torch.zeros(n, dtype=dtype, device=device)
        self._p_c = torch.zeros(n, dtype=dtype, device=device)
        if self._separable:
            self._C = init_scale.clone()  # diagonal
            self._A = torch.sqrt(self._C)
        else:
            self._C = torch.diag(init_scale.clone())
            if self._device_is_gpu():
                # factorize the initial (diagonal) C on device: numerically
                # the same as diag(sqrt(.)), and it absorbs rocSOLVER's
                # one-time init (~220 ms at d=4096) at construction instead
                # of inside a user's timed loop
                self._A = self._cholesky(self._C)
            else:
                self._A = torch.diag(torch.sqrt(init_scale.clone()))
        self._chi_n = math.sqrt(n) * (1.0 - 1.0 / (4.0 * n) + 1.0 / (21.0 * n * n))
        # amortize the O(n^3) Cholesky (reference limit_C_decomposition)
        self._decompose_interval = max(1, int(1.0 / ((self._c_1 + self._c_mu) * n * 10.0))) if limit_C_decomposition else 1
        self._steps_since_decompose = 0
        self._decomp_stream = None  # side-stream pipelined potrf state
        self._decomp_ready = None
        self._pending_A = None
        self._decomp_info = None  # deferred non-PD flag of the async factorization
        self._pending_info = None
        self._population: Optional[SolutionBatch] = None
        SinglePopulationAlgorithmMixin.__init__(self, exclude={"mean_eval"})

    # -- properties ---------------------------------------------------------

    @property
    def population(self) -> Optional[SolutionBatch]:
        return self._population

    @property
    def obj_index(self) -> int:
        return self._obj_index

    def _sigma_diag(self):
        return torch.sqrt(self._C) if self._separable else torch.sqrt(torch.diagonal(self._C))

    def _get_mean_eval(self):
        if self._population is None:
            return None
        return float(torch.mean(torch.Tensor.as_subclass(self._population.evals, torch.Tensor)[:, self._obj_index]))

    # -- stepping -----------------------------------------------------------

    def _sample(self, num_samples: Optional[int] = None):
        n = self._n
        lam = self._popsize if num_samples is None else int(num_samples)
        problem = self.problem
        g = problem.generator
        z = torch.empty((lam, n), dtype=self._m.dtype, device=self._m.device)
        z.normal_(generator=g if (g is not None and g.device == z.device) else None)
        if self._separable:
            y = z * self._A  # A is the diagonal sqrt
        else:
            y = z @ self._A.T  # y ~ N(0, C) with C = A Aᵀ
        return z, y, self._m + self._sigma * y

    def sample_distribution(self, num_samples: Optional[int] = None):
        """Draw (z, y, x): standard normals, correlated steps y = z Aᵀ, and
        solutions x = m + σ·y (reference cmaes.py: sample_distribution)."""
        return self._sample() if num_samples is None else self._sample(num_samples)

    def get_population_weights(self, order: torch.Tensor) -> torch.Tensor:
        """Recombination weights aligned to the fitness-sorted population
        (reference cmaes.py: get_population_weights)."""
        return self._weights

    def update_m(self, y_w: torch.Tensor):
        """Mean update m += c_m σ y_w (reference cmaes.py: update_m)."""
        self._m = self._m + self._c_m * self._sigma * y_w

    def update_p_sigma(self, z_w: torch.Tensor):
        """CSA path update (reference cmaes.py: update_p_sigma)."""
        cs = self._c_sigma
        self._p_sigma = (1.0 - cs) * self._p_sigma + math.sqrt(cs * (2.0 - cs) * self._mu_eff) * z_w

    def update_sigma(self):
        """CSA step-size update with clamping (reference cmaes.py:
        update_sigma)."""
        n = self._n
        cs = self._c_sigma
        ps_norm = torch.linalg.vector_norm(self._p_sigma)
        if self._csa_squared:
            exponent = (cs / self._damp_sigma) * ((ps_norm**2 / n) - 1.0) / 2.0
        else:
            exponent = (cs / self._damp_sigma) * (ps_norm / self._chi_n - 1.0)
        self._sigma = self._sigma * torch.exp(torch.clamp(exponent, max=1.0))
        if self._stdev_min is not None:
            self._sigma = torch.clamp(self._sigma, min=self._stdev_min)
        if self._stdev_max is not None:
            self._sigma = torch.clamp(self._sigma, max=self._stdev_max)
        return ps_norm

    def update_p_c(self, y_w: torch.Tensor, ps_norm: torch.Tensor) -> torch.Tensor:
        """Rank-1 path update; returns the stall indicator hs as a device
        tensor (no host sync) — reference cmaes.py: update_p_c."""
        n = self._n
        cc, cs = self._c_c, self._c_sigma
        hs_threshold = (1.4 + 2.0 / (n + 1.0)) * self._chi_n * math.sqrt(1.0 - (1.0 - cs) ** (2 * (self._steps_count + 1)))
        hs_f = (ps_norm < hs_threshold).to(self._p_c.dtype)
        self._p_c = (1.0 - cc) * self._p_c + hs_f * math.sqrt(cc * (2.0 - cc) * self._mu_eff) * y_w
        return hs_f

    def update_C(self, z: torch.Tensor, y: torch.Tensor, hs_f: torch.Tensor):
        """Rank-1 + rank-μ covariance update (active CMA negative-weight
        scaling included) — reference cmaes.py: update_C."""
        n = self._n
        cc = self._c_c
        c1, cmu = self._c_1, self._c_mu
        delta_hs = (1.0 - hs_f) * cc * (2.0 - cc)
        w = self._weights
        w_adj = w.clone()
        if self._active:
            neg_mask = w < 0
            z_norm_sq = (z**2).sum(dim=-1)
            w_adj = torch.where(neg_mask, w * n / torch.clamp(z_norm_sq, min=1e-12), w)
        if self._separable:
            rank_mu = (w_adj.unsqueeze(-1) * y**2).sum(dim=0)
            self._C = (1.0 + c1 * delta_hs - c1 - cmu * w_adj.sum()) * self._C + c1 * self._p_c**2 + cmu * rank_mu
            self._C = torch.clamp(self._C, min=1e-20)
        else:
            # fused K5 kernel: one pass over C applying all three terms
            # with exact symmetry (ops/hip/cma.hip; eager torch chain on
            # CPU); the hs stall correction and Σw stay device-side.
            from .. import ops

            self._C = self._C.contiguous()
            ops.cma_update_c_(self._C, y, w_adj, self._p_c, torch.as_tensor(hs_f, device=y.device),
                              c1=float(c1), cmu=float(cmu), cc=float(cc))

    def decompose_C(self):
        """Refresh the Cholesky factor A, amortized over
        `_decompose_interval` generations (reference cmaes.py:
        decompose_C; separable mode is a plain sqrt).

        MI355X pipelining: rocSOLVER potrf at d=4096 costs ~13 ms but
        leaves most of the chip idle, so on GPU the factorization runs on
        a SIDE stream over a snapshot of C taken at the HALF-interval
        mark, and the result is adopted exactly at the interval boundary
        (stream-wait, so the adoption generation is deterministic). The
        factor's staleness matches the reference's amortization; the
        potrf cost disappears into the idle gaps of the intervening
        generations."""
        if self._separable:
            self._A = torch.sqrt(self._C)
            return
        self._steps_since_decompose += 1
        interval = self._decompose_interval
        if interval >= 4 and self._device_is_gpu():
            if self._steps_since_decompose >= max(1, interval // 2) and self._pending_A is None:
                # deferred non-PD check of the PREVIOUS async factorization:
                # by now its side-stream work finished long ago, so the read
                # costs no wait; on failure, repair synchronously through
                # the jitter path
                if self._decomp_info is not None and int(self._decomp_info.item()) != 0:
                    self._decomp_info.zero_()
                    self._A = self._cholesky(self._C)
                # snapshot + async factorization on the side stream (no
                # host sync anywhere: the failure flag stays on device)
                if self._decomp_stream is None:
                    self._decomp_stream = torch.cuda.Stream(device=self._C.device)
                    self._decomp_ready = torch.cuda.Event()
                if self._decomp_info is None:
                    self._decomp_info = torch.zeros(1, dtype=torch.int32, device=self._C.device)
                snap_ready = torch.cuda.Event()
                snap_ready.record()
                with torch.cuda.stream(self._decomp_stream):
                    self._decomp_stream.wait_event(snap_ready)
                    c_snap = self._C.clone()
                    self._pending_A = self._cholesky(c_snap, info_out=self._decomp_info)
                    self._decomp_ready.record()
            if self._steps_since_decompose >= interval and self._pending_A is not None:
                torch.cuda.current_stream().wait_event(self._decomp_ready)
                self._A = self._pending_A
                self._pending_A = None
                self._steps_since_decompose = 0
            return
        if self._steps_since_decompose >= interval:
            self._A = self._cholesky(self._C)
            self._steps_since_decompose = 0

    def _step(self):
        problem = self.problem
        lam = self._popsize
        z, y, x = self.sample_distribution()
        batch = SolutionBatch(problem, popsize=lam, device=self._m.device, empty=True)
        batch.access_values().copy_(x)
        problem.evaluate(batch)
        self._population = batch
        order = batch.argsort(obj_index=self._obj_index)
        z = z[order]
        y = y[order]

        w_pos = self._weights[: self._mu]
        y_w = w_pos @ y[: self._mu]
        z_w = w_pos @ z[: self._mu]

        self.update_m(y_w)
        self.update_p_sigma(z_w)
        ps_norm = self.update_sigma()
        hs_f = self.update_p_c(y_w, ps_norm)
        self.update_C(z, y, hs_f)
        self.decompose_C()

    def _state_items(self) -> dict:
        return {
            "m": self._m, "sigma": self._sigma, "C": self._C, "A": self._A,
            "p_sigma": self._p_sigma, "p_c": self._p_c,
            "steps_since_decompose": self._steps_since_decompose,
        }

    def _load_state_items(self, items: dict):
        for name in ("m", "sigma", "C", "A", "p_sigma", "p_c"):
            target = getattr(self, f"_{name}")
            value = torch.as_tensor(items[name]).to(target.device, target.dtype)
            if target.ndim == 0:
                setattr(self, f"_{name}", value)
            else:
                target.copy_(value)
        self._steps_since_decompose = int(items.get("steps_since_decompose", 0))

    def _device_is_gpu(self) -> bool:
        return self._m.device.type == "cuda"

    @staticmethod
    def _blocked_cholesky(
        C: torch.Tensor, block: int = 512, panel: int = 128, info_out: Optional[torch.Tensor] = None
    ) -> torch.Tensor:
        """Right-looking blocked Cholesky composed from rocBLAS trsm + gemm
        (the O(n³) mass) with the hand in-LDS panel kernel
        (ops.potrf_tile_) on the ≤128-wide diagonal panels. rocSOLVER's
        monolithic potrf at d=4096 runs at ~1.8 TFLOP/s on MI355X and its
        unblocked potf2 chain of tiny kernels dominated the CMA-ES
        refresh profile; this keeps the trailing syrk-style updates in
        rocBLAS and leaves NO rocSOLVER kernel in the factorization.
        Non-PD input raises (after one device read at the end), matching
        torch.linalg.cholesky for the caller's jitter-retry path — UNLESS
        `info_out` is given, in which case the failure flag is left in
        that device scalar and nothing host-syncs (for the side-stream
        pipelined refresh, which checks it one interval later)."""
        A = C.clone()
        n = A.shape[0]
        on_gpu = A.is_cuda
        if info_out is not None:
            info = info_out
            info.zero_()
        else:
            info = torch.zeros(1, dtype=torch.int32, device=A.device) if on_gpu else None
        for k in range(0, n, block):
            e = min(k + block, n)
            if on_gpu:
                blk = A[k:e, k:e]
                m = e - k
                for p in range(0, m, panel):
                    q = min(p + panel, m)
                    ops.potrf_tile_(blk[p:q, p:q], info)
                    if q < m:
                        # trailing of the diagonal block (the .mT views read
                        # only our L; the panel's stale upper half is inert)
                        blk[q:, p:q] = torch.linalg.solve_triangular(
                            blk[p:q, p:q].mT, blk[q:, p:q], upper=True, left=False
                        )
                        Lp = blk[q:, p:q]
                        blk[q:, q:] -= Lp @ Lp.T
            else:
                A[k:e, k:e] = torch.linalg.cholesky(A[k:e, k:e])
            if e < n:
                # L21 = A21 · L11⁻ᵀ  ⇔  X · L11ᵀ = A21 (right triangular solve)
                A[e:, k:e] = torch.linalg.solve_triangular(
                    A[k:e, k:e].mT, A[e:, k:e], upper=True, left=False
                )
                L21 = A[e:, k:e]
                A[e:, e:] -= L21 @ L21.T
        if info_out is None and info is not None and int(info.item()) != 0:
            raise RuntimeError(f"cholesky: matrix not positive-definite (panel column {int(info.item()) - 1})")
        return torch.tril(A)

    def _cholesky(self, C: torch.Tensor, info_out: Optional[torch.Tensor] = None) -> torch.Tensor:
        try:
            if self._device_is_gpu() and C.shape[0] > 1024:
                return self._blocked_cholesky(C, info_out=info_out)
            return torch.linalg.cholesky(C)
        except Exception:
            # regularize on failure
            n = C.shape[0]
            jitter = 1e-12
            eye = torch.eye(n, dtype=C.dtype, device=C.device)
            for _ in range(12):
                try:
                    return torch.linalg.cholesky(C + jitter * eye)
                except Exception:
                    jitter *= 10.0
            raise
