"""Functional optimizers: adam / clipup / sgd as pure state-transition
triples (reference funcadam.py:34, funcclipup.py:31, funcsgd.py:30).
All state fields are tensors, batched over leading dimensions."""

from typing import Callable, NamedTuple, Optional, Union

import torch

__all__ = ["OptimizerFunctions", "get_functional_optimizer", "AdamState", "adam", "adam_ask", "adam_tell", "ClipUpState", "clipup", "clipup_ask", "clipup_tell", "SGDState", "sgd", "sgd_ask", "sgd_tell"]


def _t(x, like: torch.Tensor) -> torch.Tensor:
    return torch.as_tensor(x, dtype=like.dtype, device=like.device)


class AdamState(NamedTuple):
    center: torch.Tensor
    stepsize: torch.Tensor
    beta1: torch.Tensor
    beta2: torch.Tensor
    epsilon: torch.Tensor
    m: torch.Tensor
    v: torch.Tensor
    t: torch.Tensor


def adam(*, center_init: torch.Tensor, stepsize: float = 0.001, beta1: float = 0.9, beta2: float = 0.999, epsilon: float = 1e-8) -> AdamState:
    center = torch.as_tensor(center_init)
    return AdamState(
        center=center.clone(),
        stepsize=_t(stepsize, center),
        beta1=_t(beta1, center),
        beta2=_t(beta2, center),
        epsilon=_t(epsilon, center),
        m=torch.zeros_like(center),
        v=torch.zeros_like(center),
        t=torch.zeros((), dtype=center.dtype, device=center.device),
    )


def adam_ask(state: AdamState) -> torch.Tensor:
    return state.center


def adam_tell(state: AdamState, *, follow_grad: torch.Tensor) -> AdamState:
    """Ascent step along follow_grad."""
    g = follow_grad
    t = state.t + 1.0
    m = state.beta1 * state.m + (1.0 - state.beta1) * g
    v = state.beta2 * state.v + (1.0 - state.beta2) * g * g
    mhat = m / (1.0 - state.beta1**t)
    vhat = v / (1.0 - state.beta2**t)
    center = state.center + state.stepsize * mhat / (vhat.sqrt() + state.epsilon)
    return state._replace(center=center, m=m, v=v, t=t)


class ClipUpState(NamedTuple):
    center: torch.Tensor
    stepsize: torch.Tensor
    momentum: torch.Tensor
    max_speed: torch.Tensor
    velocity: torch.Tensor


def clipup(*, center_init: torch.Tensor, stepsize: float, momentum: float = 0.9, max_speed: Optional[float] = None) -> ClipUpState:
    center = torch.as_tensor(center_init)
    if max_speed is None:
        max_speed = 2.0 * float(stepsize)
    return ClipUpState(
        center=center.clone(),
        stepsize=_t(stepsize, center),
        momentum=_t(momentum, center),
        max_speed=_t(max_speed, center),
        velocity=torch.zeros_like(center),
    )


def clipup_ask(state: ClipUpState) -> torch.Tensor:
    return state.center


def clipup_tell(state: ClipUpState, *, follow_grad: torch.Tensor) -> ClipUpState:
    g = follow_grad
    gnorm = torch.linalg.vector_norm(g, dim=-1, keepdim=True).clamp(min=1e-30)
    step = g * (state.stepsize / gnorm)
    velocity = state.momentum * state.velocity + step
    vnorm = torch.linalg.vector_norm(velocity, dim=-1, keepdim=True).clamp(min=1e-30)
    scale = torch.clamp(state.max_speed / vnorm, max=1.0)
    velocity = velocity * scale
    return state._replace(center=state.center + velocity, velocity=velocity)


class SGDState(NamedTuple):
    center: torch.Tensor
    stepsize: torch.Tensor
    momentum: torch.Tensor
    velocity: torch.Tensor


def sgd(*, center_init: torch.Tensor, stepsize: float, momentum: Optional[float] = None) -> SGDState:
    center = torch.as_tensor(center_init)
    return SGDState(
        center=center.clone(),
        stepsize=_t(stepsize, center),
        momentum=_t(momentum if momentum is not None else 0.0, center),
        velocity=torch.zeros_like(center),
    )


def sgd_ask(state: SGDState) -> torch.Tensor:
    return state.center


def sgd_tell(state: SGDState, *, follow_grad: torch.Tensor) -> SGDState:
    velocity = state.momentum * state.velocity + state.stepsize * follow_grad
    return state._replace(center=state.center + velocity, velocity=velocity)


_OPTIMIZERS = {
    "adam": (adam, adam_ask, adam_tell),
    "clipup": (clipup, clipup_ask, clipup_tell),
    "sgd": (sgd, sgd_ask, sgd_tell),
}


class OptimizerFunctions(NamedTuple):
    """(initialize, ask, tell) triple returned by get_functional_optimizer
    (reference algorithms/functional/misc.py:26)."""

    initialize: Callable
    ask: Callable
    tell: Callable


def get_functional_optimizer(name: Union[str, tuple]):
    """Resolve 'adam'/'clipup'/'sgd' (or a custom (init, ask, tell)
    triple) — reference functional/misc.py:26."""
    if isinstance(name, tuple):
        return OptimizerFunctions(*name)
    try:
        return OptimizerFunctions(*_OPTIMIZERS[str(name).lower()])
    except KeyError:
        raise ValueError(f"Unknown functional optimizer {name!r}") from None
