"""Functional PGPE: pgpe() / pgpe_ask() / pgpe_tell().

Reference parity: /root/reference/src/evotorch/algorithms/functional/
funcpgpe.py:67-400. All math is batched over leading dimensions of the
state tensors, so a stacked state runs independent searches (the batched
hyperparameter-sweep pattern of the reference's Functional API notebooks).
"""

from typing import NamedTuple, Optional, Union

import torch

from ...utils import ranking as _ranking
from .funcoptimizers import get_functional_optimizer

__all__ = ["PGPEState", "pgpe", "pgpe_ask", "pgpe_tell"]


class PGPEState(NamedTuple):
    optimizer: tuple            # (name, opt_state) — name is a static str
    optimizer_state: NamedTuple
    stdev: torch.Tensor
    stdev_learning_rate: torch.Tensor
    stdev_min: Optional[torch.Tensor]
    stdev_max: Optional[torch.Tensor]
    stdev_max_change: Optional[torch.Tensor]
    ranking_method: str
    objective_sense: str
    symmetric: bool


def pgpe(
    *,
    center_init: torch.Tensor,
    center_learning_rate: float,
    stdev_learning_rate: float,
    objective_sense: str,
    stdev_init: Optional[Union[float, torch.Tensor]] = None,
    radius_init: Optional[float] = None,
    ranking_method: str = "centered",
    optimizer: Union[str, tuple] = "clipup",
    optimizer_config: Optional[dict] = None,
    stdev_min: Optional[Union[float, torch.Tensor]] = None,
    stdev_max: Optional[Union[float, torch.Tensor]] = None,
    stdev_max_change: Optional[Union[float, torch.Tensor]] = 0.2,
    symmetric: bool = True,
) -> PGPEState:
    center = torch.as_tensor(center_init)
    if (stdev_init is None) == (radius_init is None):
        raise ValueError("Provide exactly one of stdev_init, radius_init")
    if radius_init is not None:
        length = center.shape[-1]
        stdev_init = float(radius_init) ** 2 / length
        stdev_init = stdev_init**0.5
    stdev = torch.as_tensor(stdev_init, dtype=center.dtype, device=center.device)
    if stdev.ndim == 0:
        stdev = stdev.expand(center.shape).clone()
    opt_name = optimizer if isinstance(optimizer, str) else "custom"
    opt_init, _, _ = get_functional_optimizer(optimizer)
    opt_state = opt_init(center_init=center, stepsize=center_learning_rate, **(optimizer_config or {}))

    def opt_tensor(x):
        if x is None:
            return None
        return torch.as_tensor(x, dtype=center.dtype, device=center.device)

    return PGPEState(
        optimizer=(optimizer if isinstance(optimizer, (str, tuple)) else "clipup"),
        optimizer_state=opt_state,
        stdev=stdev,
        stdev_learning_rate=torch.as_tensor(stdev_learning_rate, dtype=center.dtype, device=center.device),
        stdev_min=opt_tensor(stdev_min),
        stdev_max=opt_tensor(stdev_max),
        stdev_max_change=opt_tensor(stdev_max_change),
        ranking_method=str(ranking_method),
        objective_sense=str(objective_sense),
        symmetric=bool(symmetric),
    )


def pgpe_ask(state: PGPEState, *, popsize: int, generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Sample a population (..., popsize, L). Symmetric mode uses the
    halves layout (see evotorch_amd/distributions.py)."""
    _, opt_ask, _ = get_functional_optimizer(state.optimizer)
    center = opt_ask(state.optimizer_state)
    shape = center.shape
    if state.symmetric:
        if popsize % 2 != 0:
            raise ValueError("symmetric PGPE needs an even popsize")
        half = popsize // 2
        z = torch.randn(shape[:-1] + (half, shape[-1]), dtype=center.dtype, device=center.device, generator=generator)
        plus = center.unsqueeze(-2) + state.stdev.unsqueeze(-2) * z
        minus = 2.0 * center.unsqueeze(-2) - plus
        return torch.cat([plus, minus], dim=-2)
    z = torch.randn(shape[:-1] + (popsize, shape[-1]), dtype=center.dtype, device=center.device, generator=generator)
    return center.unsqueeze(-2) + state.stdev.unsqueeze(-2) * z


def pgpe_tell(state: PGPEState, values: torch.Tensor, evals: torch.Tensor) -> PGPEState:
    """Consume the evaluated population and return the updated state."""
    opt_init, opt_ask, opt_tell = get_functional_optimizer(state.optimizer)
    center = opt_ask(state.optimizer_state)
    stdev = state.stdev
    weights = _ranking.rank(evals, state.ranking_method, higher_is_better=(state.objective_sense == "max")).to(values.dtype)

    n = values.shape[-2]
    if state.symmetric:
        # zero-center raw/nes-style weights here too (same rule as the
        # non-symmetric branch and the object API; round-1 ADVICE fix —
        # uncentered weights bias the sigma gradient's (w⁺+w⁻)/2 terms)
        if state.ranking_method not in ("centered", "normalized"):
            weights = weights - weights.mean(dim=-1, keepdim=True)
        d = n // 2
        noises = values[..., :d, :] - center.unsqueeze(-2)
        w_plus = weights[..., :d]
        w_minus = weights[..., d:]
        mu_grad = torch.einsum("...d,...dl->...l", (w_plus - w_minus) / 2.0, noises) / d
        sigma_grad = torch.einsum("...d,...dl->...l", (w_plus + w_minus) / 2.0, (noises**2 - stdev.unsqueeze(-2) ** 2) / stdev.unsqueeze(-2)) / d
    else:
        if state.ranking_method not in ("centered", "normalized"):
            weights = weights - weights.mean(dim=-1, keepdim=True)
        noises = values - center.unsqueeze(-2)
        mu_grad = torch.einsum("...n,...nl->...l", weights, noises) / n
        sigma_grad = torch.einsum("...n,...nl->...l", weights, (noises**2 - stdev.unsqueeze(-2) ** 2) / stdev.unsqueeze(-2)) / n

    new_opt_state = opt_tell(state.optimizer_state, follow_grad=mu_grad)
    new_stdev = stdev + state.stdev_learning_rate * sigma_grad
    if state.stdev_max_change is not None:
        allowed = stdev.abs() * state.stdev_max_change
        new_stdev = torch.clamp(new_stdev, stdev - allowed, stdev + allowed)
    if state.stdev_min is not None:
        new_stdev = torch.maximum(new_stdev, state.stdev_min)
    if state.stdev_max is not None:
        new_stdev = torch.minimum(new_stdev, state.stdev_max)
    return state._replace(optimizer_state=new_opt_state, stdev=new_stdev)
