"""StatefulModule / ensure_stateful: hides a recurrent module's explicit
hidden state in an attribute, for plain `module(x)` call sites.

Reference parity: /root/reference/src/evotorch/neuroevolution/net/
statefulmodule.py:21,89.
"""

from typing import Any, Optional

import torch
from torch import nn

__all__ = ["StatefulModule", "ensure_stateful"]


class StatefulModule(nn.Module):
    def __init__(self, wrapped: nn.Module):
        super().__init__()
        self.wrapped_module = wrapped
        self._state: Optional[Any] = None

    @property
    def state(self) -> Optional[Any]:
        return self._state

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._state is None:
            out = self.wrapped_module(x)
        else:
            out = self.wrapped_module(x, self._state)
        if isinstance(out, tuple):
            y, self._state = out
            return y
        self._state = None
        return out

    def reset(self):
        self._state = None


def ensure_stateful(net: nn.Module) -> StatefulModule:
    if isinstance(net, StatefulModule):
        return net
    return StatefulModule(net)
