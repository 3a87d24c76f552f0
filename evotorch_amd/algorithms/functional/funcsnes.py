"""Functional SNES: snes() / snes_ask() / snes_tell().

A green-field extension of the functional family (the reference ships
functional pgpe and cem only — algorithms/functional/__init__.py:15-50);
the state-transition style matches funcpgpe.py, so stacked states run
batched hyperparameter sweeps under vmap just the same.

Math follows the object API's SNES (gaussian.py / ExpSeparableGaussian):
natural gradients on (mu, sigma) from raw noise z = (x - mu) / sigma, with
the exponential sigma update sigma' = sigma * exp(0.5 * lr * grad).
"""

import math
from typing import NamedTuple, Optional, Union

import torch

from ...utils import ranking as _ranking

__all__ = ["SNESState", "snes", "snes_ask", "snes_tell"]


class SNESState(NamedTuple):
    center: torch.Tensor
    stdev: torch.Tensor
    center_learning_rate: torch.Tensor
    stdev_learning_rate: torch.Tensor
    ranking_method: str
    objective_sense: str


def snes(
    *,
    center_init: torch.Tensor,
    objective_sense: str,
    stdev_init: Optional[Union[float, torch.Tensor]] = None,
    radius_init: Optional[float] = None,
    center_learning_rate: float = 1.0,
    stdev_learning_rate: Optional[float] = None,
    ranking_method: str = "nes",
) -> SNESState:
    center = torch.as_tensor(center_init)
    length = center.shape[-1]
    if (stdev_init is None) == (radius_init is None):
        raise ValueError("Provide exactly one of stdev_init, radius_init")
    if radius_init is not None:
        stdev_init = math.sqrt(float(radius_init) ** 2 / length)
    stdev = torch.as_tensor(stdev_init, dtype=center.dtype, device=center.device)
    if stdev.ndim == 0:
        stdev = stdev.expand(center.shape).clone()
    if stdev_learning_rate is None:
        stdev_learning_rate = 0.2 * (3 + math.log(length)) / math.sqrt(length)
    return SNESState(
        center=center,
        stdev=stdev,
        center_learning_rate=torch.as_tensor(center_learning_rate, dtype=center.dtype, device=center.device),
        stdev_learning_rate=torch.as_tensor(stdev_learning_rate, dtype=center.dtype, device=center.device),
        ranking_method=str(ranking_method),
        objective_sense=str(objective_sense),
    )


def snes_ask(state: SNESState, *, popsize: int, generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Sample a population (..., popsize, L)."""
    shape = state.center.shape
    z = torch.randn(shape[:-1] + (popsize, shape[-1]), dtype=state.center.dtype,
                    device=state.center.device, generator=generator)
    return state.center.unsqueeze(-2) + state.stdev.unsqueeze(-2) * z


def snes_tell(state: SNESState, values: torch.Tensor, evals: torch.Tensor) -> SNESState:
    """Consume the evaluated population and return the updated state."""
    weights = _ranking.rank(evals, state.ranking_method,
                            higher_is_better=(state.objective_sense == "max")).to(values.dtype)
    if state.ranking_method != "nes":
        weights = weights / weights.abs().sum(dim=-1, keepdim=True)
    z = (values - state.center.unsqueeze(-2)) / state.stdev.unsqueeze(-2)
    mu_grad = torch.einsum("...n,...nl->...l", weights, z)
    sigma_grad = torch.einsum("...n,...nl->...l", weights, z**2 - 1.0)
    lr_c = state.center_learning_rate
    while lr_c.ndim < mu_grad.ndim:
        lr_c = lr_c.unsqueeze(-1)
    lr_s = state.stdev_learning_rate
    while lr_s.ndim < sigma_grad.ndim:
        lr_s = lr_s.unsqueeze(-1)
    new_center = state.center + lr_c * state.stdev * mu_grad
    new_stdev = state.stdev * torch.exp(0.5 * lr_s * sigma_grad)
    return state._replace(center=new_center, stdev=new_stdev)
