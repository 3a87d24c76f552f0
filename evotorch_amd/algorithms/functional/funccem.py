"""Functional CEM: cem() / cem_ask() / cem_tell().

Reference parity: /root/reference/src/evotorch/algorithms/functional/
funccem.py:34-260. Batched over leading state dimensions.
"""

from typing import NamedTuple, Optional, Union

import torch

__all__ = ["CEMState", "cem", "cem_ask", "cem_tell"]


class CEMState(NamedTuple):
    center: torch.Tensor
    stdev: torch.Tensor
    parenthood_ratio: float
    objective_sense: str
    stdev_min: Optional[torch.Tensor]
    stdev_max: Optional[torch.Tensor]
    stdev_max_change: Optional[torch.Tensor]


def cem(
    *,
    center_init: torch.Tensor,
    parenthood_ratio: float,
    objective_sense: str,
    stdev_init: Optional[Union[float, torch.Tensor]] = None,
    radius_init: Optional[float] = None,
    stdev_min: Optional[Union[float, torch.Tensor]] = None,
    stdev_max: Optional[Union[float, torch.Tensor]] = None,
    stdev_max_change: Optional[Union[float, torch.Tensor]] = None,
) -> CEMState:
    center = torch.as_tensor(center_init)
    if (stdev_init is None) == (radius_init is None):
        raise ValueError("Provide exactly one of stdev_init, radius_init")
    if radius_init is not None:
        stdev_init = (float(radius_init) ** 2 / center.shape[-1]) ** 0.5
    stdev = torch.as_tensor(stdev_init, dtype=center.dtype, device=center.device)
    if stdev.ndim == 0:
        stdev = stdev.expand(center.shape).clone()

    def opt_tensor(x):
        return None if x is None else torch.as_tensor(x, dtype=center.dtype, device=center.device)

    if objective_sense not in ("min", "max"):
        raise ValueError(f"objective_sense must be 'min' or 'max', got {objective_sense!r}")
    return CEMState(
        center=center.clone(),
        stdev=stdev,
        parenthood_ratio=float(parenthood_ratio),
        objective_sense=str(objective_sense),
        stdev_min=opt_tensor(stdev_min),
        stdev_max=opt_tensor(stdev_max),
        stdev_max_change=opt_tensor(stdev_max_change),
    )


def cem_ask(state: CEMState, *, popsize: int, generator: Optional[torch.Generator] = None) -> torch.Tensor:
    z = torch.randn(state.center.shape[:-1] + (popsize, state.center.shape[-1]), dtype=state.center.dtype, device=state.center.device, generator=generator)
    return state.center.unsqueeze(-2) + state.stdev.unsqueeze(-2) * z


def cem_tell(state: CEMState, values: torch.Tensor, evals: torch.Tensor) -> CEMState:
    n = values.shape[-2]
    num_parents = max(1, int(n * state.parenthood_ratio))
    utils = evals if state.objective_sense == "max" else -evals
    top = utils.argsort(dim=-1, descending=True)[..., :num_parents]
    parents = torch.gather(values, -2, top.unsqueeze(-1).expand(top.shape + (values.shape[-1],)))
    new_center = parents.mean(dim=-2)
    new_stdev = parents.std(dim=-2, unbiased=True)
    if state.stdev_max_change is not None:
        allowed = state.stdev.abs() * state.stdev_max_change
        new_stdev = torch.clamp(new_stdev, state.stdev - allowed, state.stdev + allowed)
    if state.stdev_min is not None:
        new_stdev = torch.maximum(new_stdev, state.stdev_min)
    if state.stdev_max is not None:
        new_stdev = torch.minimum(new_stdev, state.stdev_max)
    return state._replace(center=new_center, stdev=new_stdev)
