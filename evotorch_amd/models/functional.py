"""Functional-ized modules: run an nn.Module from a flat parameter vector,
and over a whole POPULATION of parameter vectors at once.

Reference parity: /root/reference/src/evotorch/neuroevolution/net/
functional.py (ModuleExpectingFlatParameters :46, make_functional_module
:203) and the vmapped population forward of net/vecrl.py:1261-1265.

MI355X note: the population-batched forward of Linear-family policies is
GEMM-shaped work (K10 in SURVEY.md §2.9). `torch.func.vmap` over
`functional_call` lowers population-batched Linear layers to rocBLAS
batched GEMMs (bmm); the fully-fused LDS-resident path for linear policies
lives in evotorch_amd/ops/hip/rollout.hip.
"""

import copy
import torch
from torch import nn

__all__ = ["ModuleExpectingFlatParameters", "make_functional_module"]


class ModuleExpectingFlatParameters:
    """Wraps an nn.Module as a pure function `f(flat_params, *inputs)`.

    With a 2-D `flat_params` (popsize × L) the call is vmapped across the
    population: `f(params_2d, batched_inputs)` evaluates every member's
    network in one batched pass.
    """

    def __init__(self, net: nn.Module, *, disable_autograd_tracking: bool = True):
        self._net = copy.deepcopy(net)
        self._param_shapes = [(name, p.shape, p.numel()) for name, p in self._net.named_parameters()]
        self._length = sum(n for _, _, n in self._param_shapes)
        self._buffers = dict(self._net.named_buffers())
        if disable_autograd_tracking:
            for p in self._net.parameters():
                p.requires_grad_(False)

    @property
    def parameter_count(self) -> int:
        return self._length

    def parameter_dict(self, flat: torch.Tensor) -> dict:
        out = {}
        offset = 0
        for name, shape, numel in self._param_shapes:
            out[name] = flat[offset : offset + numel].reshape(shape)
            offset += numel
        return out

    def _single(self, flat: torch.Tensor, *inputs):
        params = self.parameter_dict(flat)
        return torch.func.functional_call(self._net, {**params, **self._buffers}, inputs)

    def __call__(self, flat_params: torch.Tensor, *inputs):
        if flat_params.ndim == 1:
            return self._single(flat_params, *inputs)
        if flat_params.ndim == 2:
            return torch.func.vmap(self._single)(flat_params, *inputs)
        raise ValueError(f"flat_params must be 1-D or 2-D, got ndim={flat_params.ndim}")


def make_functional_module(net: nn.Module, *, disable_autograd_tracking: bool = True) -> ModuleExpectingFlatParameters:
    return ModuleExpectingFlatParameters(net, disable_autograd_tracking=disable_autograd_tracking)
