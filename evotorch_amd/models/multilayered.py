"""MultiLayered: a sequential container that threads optional recurrent
hidden states through its layers.

Reference parity: /root/reference/src/evotorch/neuroevolution/net/
multilayered.py:21. A layer whose forward takes (x, h) and returns
(y, new_h) participates in the state dict keyed by its index.
"""

import inspect
from typing import Optional, Tuple, Union

import torch
from torch import nn

__all__ = ["MultiLayered"]


def _wants_state(module: nn.Module) -> bool:
    try:
        sig = inspect.signature(module.forward)
    except (TypeError, ValueError):
        return False
    return len(sig.parameters) >= 2


class MultiLayered(nn.Module):
    def __init__(self, *layers: nn.Module):
        super().__init__()
        self._submodules = nn.ModuleList(layers)

    def __iter__(self):
        return iter(self._submodules)

    def __len__(self):
        return len(self._submodules)

    def __getitem__(self, i):
        return self._submodules[i]

    def forward(self, x: torch.Tensor, h: Optional[dict] = None) -> Union[torch.Tensor, Tuple[torch.Tensor, dict]]:
        new_h = {}
        for i, layer in enumerate(self._submodules):
            if _wants_state(layer):
                layer_h = None if h is None else h.get(i, None)
                x, layer_new_h = layer(x, layer_h)
                new_h[i] = layer_new_h
            else:
                x = layer(x)
        if len(new_h) == 0:
            return x
        return x, new_h

    def append(self, module: nn.Module):
        self._submodules.append(module)
