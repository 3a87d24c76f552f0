"""ES step-followers: ClipUp, Adam, SGD.

Reference parity: /root/reference/src/evotorch/optimizers.py:101-357. Unlike
the reference (which wraps `torch.optim` objects around a dummy parameter),
these are direct stateful implementations over a gradient stream — the
`ascent(grad)` method returns the additive step. On ROCm devices both the
moment updates and ClipUp's global-norm clipping run as fused HIP kernels
(K4 in SURVEY.md §2.9) so the L-length state never round-trips to host.
"""

from typing import Optional, Type, Union

import torch

from . import ops
from .utils import Device, DType, to_torch_dtype

__all__ = ["Adam", "SGD", "ClipUp", "ClipUpParameterGroup", "TorchOptimizer", "get_optimizer_class"]


class _Optimizer:
    def __init__(self, *, solution_length: int, dtype: DType = torch.float32, device: Device = "cpu", stepsize: float):
        self._length = int(solution_length)
        self._dtype = to_torch_dtype(dtype)
        self._device = torch.device(device)
        self._stepsize = float(stepsize)
        # param_groups-style view for hyperparameter read/write compatibility
        self.param_groups = [_ParamGroup(self)]

    @property
    def contained_optimizer(self) -> "_Optimizer":
        return self

    def _new_state(self) -> torch.Tensor:
        return torch.zeros(self._length, dtype=self._dtype, device=self._device)

    def ascent(self, globalg: torch.Tensor, *, cloned_result: bool = True) -> torch.Tensor:
        raise NotImplementedError


class _ParamGroup(dict):
    def __init__(self, opt: _Optimizer):
        super().__init__()
        self._opt = opt

    def __getitem__(self, k):
        if k == "lr":
            return self._opt._stepsize
        if k == "max_speed" and hasattr(self._opt, "_max_speed"):
            return self._opt._max_speed
        return super().__getitem__(k)

    def __setitem__(self, k, v):
        if k == "lr":
            self._opt._stepsize = float(v)
        elif k == "max_speed" and hasattr(self._opt, "_max_speed"):
            self._opt._max_speed = float(v)
        else:
            super().__setitem__(k, v)


class Adam(_Optimizer):
    """Adam over the (ascent-direction) gradient stream."""

    def __init__(
        self,
        *,
        solution_length: int,
        dtype: DType = torch.float32,
        device: Device = "cpu",
        stepsize: float,
        beta1: float = 0.9,
        beta2: float = 0.999,
        epsilon: float = 1e-8,
        amsgrad: bool = False,
    ):
        super().__init__(solution_length=solution_length, dtype=dtype, device=device, stepsize=stepsize)
        self._beta1 = float(beta1)
        self._beta2 = float(beta2)
        self._epsilon = float(epsilon)
        if amsgrad:
            raise NotImplementedError("amsgrad is not supported")
        self._m = self._new_state()
        self._v = self._new_state()
        self._t = 0

    def ascent(self, globalg: torch.Tensor, *, cloned_result: bool = True) -> torch.Tensor:
        self._t += 1
        step = torch.empty_like(self._m)
        ops.fused_adam_step_(
            step,
            globalg.to(dtype=self._dtype, device=self._device),
            self._m,
            self._v,
            step_count=self._t,
            stepsize=self._stepsize,
            beta1=self._beta1,
            beta2=self._beta2,
            epsilon=self._epsilon,
        )
        return step


class SGD(_Optimizer):
    """Plain (optionally momentum) SGD over the gradient stream."""

    def __init__(
        self,
        *,
        solution_length: int,
        dtype: DType = torch.float32,
        device: Device = "cpu",
        stepsize: float,
        momentum: Optional[float] = None,
    ):
        super().__init__(solution_length=solution_length, dtype=dtype, device=device, stepsize=stepsize)
        self._momentum = None if momentum is None else float(momentum)
        self._velocity = self._new_state() if self._momentum is not None else None

    def ascent(self, globalg: torch.Tensor, *, cloned_result: bool = True) -> torch.Tensor:
        g = globalg.to(dtype=self._dtype, device=self._device)
        if self._momentum is None:
            return self._stepsize * g
        self._velocity.mul_(self._momentum).add_(g, alpha=self._stepsize)
        return self._velocity.clone() if cloned_result else self._velocity


class ClipUp(_Optimizer):
    """ClipUp (Toklu et al. 2020, arXiv:2008.02387): normalized gradient
    step + momentum velocity clipped to a maximum speed. The default
    `max_speed` is `2 * stepsize` when not given, matching the reference
    (optimizers.py:231)."""

    def __init__(
        self,
        *,
        solution_length: int,
        dtype: DType = torch.float32,
        device: Device = "cpu",
        stepsize: float,
        momentum: float = 0.9,
        max_speed: Optional[float] = None,
    ):
        super().__init__(solution_length=solution_length, dtype=dtype, device=device, stepsize=stepsize)
        self._momentum = float(momentum)
        self._max_speed = float(max_speed) if max_speed is not None else 2.0 * float(stepsize)
        self._velocity = self._new_state()

    @property
    def max_speed(self) -> float:
        return self._max_speed

    def ascent(self, globalg: torch.Tensor, *, cloned_result: bool = True) -> torch.Tensor:
        g = globalg.to(dtype=self._dtype, device=self._device)
        ops.clipup_step_(
            self._velocity,
            g,
            step_size=self._stepsize,
            max_speed=self._max_speed,
            momentum=self._momentum,
        )
        return self._velocity.clone() if cloned_result else self._velocity


class TorchOptimizer:
    """Adapter exposing any `torch.optim` optimizer through the framework's
    `ascent(gradient)` interface (reference optimizers.py:31-100): the
    gradient-ASCENT direction is handed to a torch optimizer that performs
    descent on an internal parameter copy; the returned step is the
    parameter delta.

    Example:
        opt = TorchOptimizer(torch.optim.RMSprop, solution_length=L,
                             stepsize=0.01, config={"alpha": 0.95})
        new_center = center + opt.ascent(grad)
    """

    def __init__(
        self,
        torch_optimizer,
        *,
        solution_length: int,
        stepsize: float,
        dtype: DType = torch.float32,
        device: Device = "cpu",
        config: Optional[dict] = None,
    ):
        self._length = int(solution_length)
        self._dtype = to_torch_dtype(dtype)
        self._device = torch.device(device)
        self._params = torch.zeros(self._length, dtype=self._dtype, device=self._device, requires_grad=False)
        cfg = dict(config or {})
        cfg["lr"] = float(stepsize)
        self._opt = torch_optimizer([self._params], **cfg)
        self.param_groups = self._opt.param_groups

    @property
    def contained_optimizer(self):
        return self._opt

    def ascent(self, globalg: torch.Tensor, *, cloned_result: bool = True) -> torch.Tensor:
        before = self._params.clone()
        self._params.grad = -globalg.to(dtype=self._dtype, device=self._device).reshape(-1)
        self._opt.step()
        self._opt.zero_grad(set_to_none=True)
        step = self._params - before
        return step.clone() if cloned_result else step


ClipUpParameterGroup = _ParamGroup  # reference exposes the ClipUp param-group view type


def get_optimizer_class(s: str, optimizer_config: Optional[dict] = None) -> Union[Type, callable]:
    """Resolve an optimizer name ('adam', 'sgd', 'clipup', ...) to a class
    or pre-configured factory (reference optimizers.py:359)."""
    lowered = str(s).lower()
    if lowered == "adam":
        cls = Adam
    elif lowered in ("sgd", "momentum", "nesterov"):
        cls = SGD
    elif lowered == "clipup":
        cls = ClipUp
    else:
        raise ValueError(f"Unknown optimizer {s!r}; expected 'adam', 'sgd', or 'clipup'")
    if optimizer_config:
        def factory(**kwargs):
            merged = {**optimizer_config, **kwargs}
            return cls(**merged)

        return factory
    return cls
