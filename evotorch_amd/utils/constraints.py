"""Constraint penalization utilities: violation / log_barrier / penalty.

Re-design of the reference's `tools/constraints.py`
(/root/reference/src/evotorch/tools/constraints.py:22-281). These are plain
tensor functions batchable via `evotorch_amd.decorators.expects_ndim`.
"""

from typing import Union

import torch

__all__ = ["violation", "log_barrier", "penalty"]

_COMPARISONS = ("<=", "==", ">=")


def _as_pair(lhs, rhs):
    lhs = torch.as_tensor(lhs)
    rhs = torch.as_tensor(rhs, dtype=lhs.dtype if lhs.is_floating_point() else None, device=lhs.device if isinstance(lhs, torch.Tensor) else None)
    return lhs, rhs


def violation(lhs: Union[float, torch.Tensor], comparison: str, rhs: Union[float, torch.Tensor]) -> torch.Tensor:
    """Amount of constraint violation (0 when satisfied).

    ``violation(lhs, "<=", rhs) = max(lhs - rhs, 0)``;
    ``">="`` mirrors; ``"=="`` returns ``|lhs - rhs|``.
    """
    if comparison not in _COMPARISONS:
        raise ValueError(f"Unknown comparison {comparison!r}; expected one of {_COMPARISONS}")
    lhs, rhs = _as_pair(lhs, rhs)
    if comparison == "<=":
        return torch.clamp(lhs - rhs, min=0)
    if comparison == ">=":
        return torch.clamp(rhs - lhs, min=0)
    return (lhs - rhs).abs()


def log_barrier(
    lhs: Union[float, torch.Tensor],
    comparison: str,
    rhs: Union[float, torch.Tensor],
    *,
    penalty_sign: str,
    sharpness: Union[float, torch.Tensor] = 1.0,
    inf: Union[float, torch.Tensor] = float("inf"),
) -> torch.Tensor:
    """Logarithmic barrier penalty for an inequality constraint.

    Inside the feasible region returns ``log(margin) * sharpness`` with the
    requested sign; at/over the boundary returns ``-inf`` (or ``+inf`` for
    positive penalty sign), clamped to ``inf`` magnitude if given.
    """
    if comparison not in ("<=", ">="):
        raise ValueError("log_barrier supports only '<=' and '>='")
    if penalty_sign not in ("+", "-"):
        raise ValueError("penalty_sign must be '+' or '-'")
    lhs, rhs = _as_pair(lhs, rhs)
    margin = (rhs - lhs) if comparison == "<=" else (lhs - rhs)
    sharpness = torch.as_tensor(sharpness, dtype=margin.dtype, device=margin.device)
    raw = torch.where(margin > 0, torch.log(margin.clamp(min=1e-45)), torch.full_like(margin, float("-inf")))
    raw = raw * sharpness
    inf_t = torch.as_tensor(inf, dtype=margin.dtype, device=margin.device)
    raw = torch.clamp(raw, min=-inf_t.abs())
    return raw if penalty_sign == "-" else -raw


def penalty(
    lhs: Union[float, torch.Tensor],
    comparison: str,
    rhs: Union[float, torch.Tensor],
    *,
    penalty_sign: str,
    linear: Union[float, torch.Tensor, None] = None,
    step: Union[float, torch.Tensor, None] = None,
    exp: Union[float, torch.Tensor, None] = None,
    exp_inf: Union[float, torch.Tensor, None] = None,
) -> torch.Tensor:
    """Combined penalty: ``linear * v + step * (v > 0) + v ** exp`` where
    `v` is the violation amount; the exponential term is clamped to
    `exp_inf` magnitude if given. The result carries `penalty_sign`."""
    if penalty_sign not in ("+", "-"):
        raise ValueError("penalty_sign must be '+' or '-'")
    v = violation(lhs, comparison, rhs)
    total = torch.zeros_like(v)
    if linear is not None:
        total = total + torch.as_tensor(linear, dtype=v.dtype, device=v.device) * v
    if step is not None:
        total = total + torch.as_tensor(step, dtype=v.dtype, device=v.device) * (v > 0).to(v.dtype)
    if exp is not None:
        e = torch.as_tensor(exp, dtype=v.dtype, device=v.device)
        powed = v**e
        if exp_inf is not None:
            powed = torch.clamp(powed, max=torch.as_tensor(exp_inf, dtype=v.dtype, device=v.device).abs())
        total = total + powed
    if linear is None and step is None and exp is None:
        total = v
    return -total if penalty_sign == "-" else total
