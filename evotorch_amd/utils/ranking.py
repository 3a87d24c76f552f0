"""Fitness shaping (ranking) utilities.

Re-design of the reference's `tools/ranking.py`
(/root/reference/src/evotorch/tools/ranking.py:24-216). On ROCm devices the
sort+map is dispatched through `evotorch_amd.ops` so the utility map is fused
with the argsort postprocessing in one HIP kernel (K2 in SURVEY.md §2.9);
this module is the device-agnostic eager implementation and the public API.

All ranking functions accept `higher_is_better` and return utilities with the
same shape as the input fitnesses. For batched input (ndim > 1), ranking is
applied along the last dimension independently for each leading index.
"""

from typing import Optional

import torch

__all__ = ["centered", "linear", "nes", "normalized", "raw", "rank", "ranking_method_exists"]


def _ranks_ascending(fitnesses: torch.Tensor, higher_is_better: bool) -> torch.Tensor:
    """Integer ranks in [0, n): rank 0 = worst solution, n-1 = best."""
    x = fitnesses if higher_is_better else -fitnesses
    order = x.argsort(dim=-1)
    ranks = torch.empty_like(order)
    n = fitnesses.shape[-1]
    src = torch.arange(n, device=fitnesses.device).expand_as(order)
    ranks.scatter_(-1, order, src)
    return ranks


def centered(fitnesses: torch.Tensor, *, higher_is_better: bool = True) -> torch.Tensor:
    """Centered ranks in [-0.5, 0.5] — the default fitness shaping of PGPE."""
    n = fitnesses.shape[-1]
    ranks = _ranks_ascending(fitnesses, higher_is_better).to(torch.float32)
    if n == 1:
        return torch.zeros_like(fitnesses, dtype=torch.float32)
    return (ranks / (n - 1)) - 0.5


def linear(fitnesses: torch.Tensor, *, higher_is_better: bool = True) -> torch.Tensor:
    """Linearly spaced utilities in [0, 1]."""
    n = fitnesses.shape[-1]
    ranks = _ranks_ascending(fitnesses, higher_is_better).to(torch.float32)
    if n == 1:
        return torch.zeros_like(fitnesses, dtype=torch.float32)
    return ranks / (n - 1)


def nes(fitnesses: torch.Tensor, *, higher_is_better: bool = True) -> torch.Tensor:
    """NES log-utilities (Wierstra et al. 2014), shifted to sum to ~0:
    ``u_i = max(0, log(n/2+1) - log(rank_from_best_i)) / Z - 1/n``."""
    import math

    n = fitnesses.shape[-1]
    ranks = _ranks_ascending(fitnesses, higher_is_better)
    # rank-from-best: best solution gets 1. The log(n/2+1) constant stays a
    # python float (a device-tensor construction here would be an H2D copy,
    # which hipGraph capture forbids).
    rank_from_best = (n - ranks).to(torch.float32)
    util = torch.clamp(math.log(n / 2.0 + 1.0) - torch.log(rank_from_best), min=0.0)
    denom = util.sum(dim=-1, keepdim=True)
    return util / denom - 1.0 / n


def normalized(fitnesses: torch.Tensor, *, higher_is_better: bool = True) -> torch.Tensor:
    """(f - mean) / stdev, negated when lower is better."""
    f = fitnesses.to(torch.float32)
    if not higher_is_better:
        f = -f
    mean = f.mean(dim=-1, keepdim=True)
    std = f.std(dim=-1, keepdim=True)
    return (f - mean) / std


def raw(fitnesses: torch.Tensor, *, higher_is_better: bool = True) -> torch.Tensor:
    """Identity shaping (negated when lower is better)."""
    f = fitnesses.to(torch.float32)
    return f if higher_is_better else -f


rankers = {
    "centered": centered,
    "linear": linear,
    "nes": nes,
    "normalized": normalized,
    "raw": raw,
}


def ranking_method_exists(method: str) -> bool:
    return method in rankers


_FUSED_METHODS = {"centered": 0, "linear": 1, "nes": 2}


def rank(fitnesses: torch.Tensor, ranking_method: Optional[str] = "raw", *, higher_is_better: bool) -> torch.Tensor:
    """Apply the named ranking method; ``None`` means raw.

    On ROCm, sort-based methods for 1-D fitness vectors up to 8192 run as
    ONE fused HIP kernel (bitonic sort + utility map, K2 in SURVEY.md
    §2.9) instead of the ~6-dispatch torch chain."""
    if ranking_method is None:
        ranking_method = "raw"
    try:
        f = rankers[ranking_method]
    except KeyError:
        raise ValueError(f"Unknown ranking method {ranking_method!r}; available: {sorted(rankers)}") from None
    fitnesses = torch.as_tensor(fitnesses)
    if (
        fitnesses.is_cuda
        and fitnesses.ndim == 1
        and 1 < fitnesses.shape[0] <= 8192
        and ranking_method in _FUSED_METHODS
    ):
        from ..ops.dispatch import _allow_eager_on_gpu, hip_required

        if not _allow_eager_on_gpu():
            mod = hip_required()
            return mod.fused_rank(fitnesses, _FUSED_METHODS[ranking_method], bool(higher_is_better))
    return f(fitnesses, higher_is_better=higher_is_better)
