"""Policy: a functional-ized network evaluated over a whole population,
with auto-managed (and maskable-reset) recurrent state.

Reference parity: the `Policy` class of
/root/reference/src/evotorch/neuroevolution/net/vecrl.py:1019-1315
(vmapped call :1261-1265, masked state reset via reset_tensors :866).

The parameters tensor may be 1-D (single member) or 2-D (population);
2-D parameters run the whole population per step as a batched forward
(rocBLAS batched GEMMs for Linear stacks — K10 in SURVEY.md §2.9).
"""

from typing import Any, Optional, Union

import torch
from torch import nn

from .functional import make_functional_module
from .parser import str_to_net

__all__ = ["Policy", "reset_tensors"]


def reset_tensors(container: Any, mask: torch.Tensor) -> Any:
    """Recursively zero the rows of every tensor in `container` selected by
    the boolean `mask` (reference net/vecrl.py:866)."""
    if isinstance(container, torch.Tensor):
        m = mask
        while m.ndim < container.ndim:
            m = m.unsqueeze(-1)
        return torch.where(m, torch.zeros_like(container), container)
    if isinstance(container, dict):
        return {k: reset_tensors(v, mask) for k, v in container.items()}
    if isinstance(container, (list, tuple)):
        items = [reset_tensors(v, mask) for v in container]
        return type(container)(items) if isinstance(container, list) else tuple(items)
    return container


class Policy:
    def __init__(self, net: Union[str, nn.Module], **constants):
        if isinstance(net, str):
            net = str_to_net(net, **constants)
        self._net = net
        self._fmodule = make_functional_module(net)
        self._params: Optional[torch.Tensor] = None
        self._state: Optional[Any] = None

    @property
    def parameter_count(self) -> int:
        return self._fmodule.parameter_count

    @property
    def parameters(self) -> Optional[torch.Tensor]:
        return self._params

    @property
    def parameter_length(self) -> int:  # reference vecrl.py alias
        return self.parameter_count

    @property
    def wrapped_module(self) -> nn.Module:
        """The underlying torch module (reference vecrl.py: wrapped_module)."""
        return self.net

    @property
    def net(self) -> nn.Module:
        return self._net

    @property
    def h(self) -> Optional[Any]:
        return self._state

    def set_parameters(self, params: torch.Tensor, *, reset: bool = True):
        if params.shape[-1] != self.parameter_count:
            raise ValueError(f"Expected parameter vectors of length {self.parameter_count}, got {params.shape[-1]}")
        self._params = params
        if reset:
            self._state = None

    def __call__(self, obs: torch.Tensor) -> torch.Tensor:
        if self._params is None:
            raise RuntimeError("Call set_parameters first")
        if self._state is None:
            out = self._fmodule(self._params, obs)
        else:
            out = self._call_with_state(obs)
        if isinstance(out, tuple):
            y, self._state = out
            return y
        # probe: maybe the module is stateful but state was None
        return out

    def _call_with_state(self, obs: torch.Tensor):
        if self._params.ndim == 2:
            return torch.func.vmap(self._fmodule._single)(self._params, obs, self._state)
        return self._fmodule._single(self._params, obs, self._state)

    def reset(self, mask: Optional[torch.Tensor] = None, *, copy: bool = True):
        """Reset recurrent state: fully (mask=None) or only the rows where
        mask is True (per-env episode restarts)."""
        if mask is None or self._state is None:
            self._state = None
            return
        self._state = reset_tensors(self._state, mask)

    def to_torch_module(self, parameter_vector: torch.Tensor) -> nn.Module:
        """A standalone stateful nn.Module with the given parameters."""
        import copy as _copy

        from .misc import fill_parameters
        from .statefulmodule import ensure_stateful

        net = _copy.deepcopy(self._net)
        fill_parameters(net, torch.as_tensor(parameter_vector))
        return ensure_stateful(net)
