"""Flat-parameter-vector ⇄ nn.Module utilities.

Reference parity: /root/reference/src/evotorch/neuroevolution/net/misc.py
(fill_parameters :26, parameter_vector :50, count_parameters :73).
"""

import torch
from torch import nn

__all__ = ["count_parameters", "parameter_vector", "fill_parameters", "device_of_module"]


def count_parameters(net: nn.Module) -> int:
    return sum(p.numel() for p in net.parameters())


def parameter_vector(net: nn.Module, *, device=None) -> torch.Tensor:
    parts = [p.detach().reshape(-1) for p in net.parameters()]
    result = torch.cat(parts) if parts else torch.empty(0)
    if device is not None:
        result = result.to(device)
    return result


@torch.no_grad()
def fill_parameters(net: nn.Module, vector: torch.Tensor):
    """Load a flat vector into the module's parameters (in order)."""
    offset = 0
    for p in net.parameters():
        n = p.numel()
        p.copy_(vector[offset : offset + n].reshape(p.shape).to(p.device, p.dtype))
        offset += n
    if offset != vector.numel():
        raise ValueError(f"Vector length {vector.numel()} does not match parameter count {offset}")


def device_of_module(net: nn.Module, default=None):
    for p in net.parameters():
        return p.device
    for b in net.buffers():
        return b.device
    return torch.device(default) if default is not None else torch.device("cpu")
