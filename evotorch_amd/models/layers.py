"""Custom policy modules: Clip, Bin, Slice, Round, Apply, stateless-signature
RNN/LSTM, FeedForwardNet, StructuredControlNet, LocomotorNet.

Reference parity: /root/reference/src/evotorch/neuroevolution/net/
layers.py:24-568. The recurrent modules take and return their hidden state
explicitly (`forward(x, h) -> (y, h)`), which is what makes them usable
under vmapped population forwards.
"""

import math
from typing import Callable, Optional, Tuple, Union

import torch
from torch import nn

__all__ = [
    "Clip",
    "Bin",
    "Slice",
    "Round",
    "Apply",
    "RNN",
    "LSTM",
    "RecurrentNet",
    "FeedForwardNet",
    "StructuredControlNet",
    "LocomotorNet",
]


class Clip(nn.Module):
    def __init__(self, lb: float, ub: float):
        super().__init__()
        self._lb = float(lb)
        self._ub = float(ub)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x.clamp(self._lb, self._ub)

    def extra_repr(self) -> str:
        return f"lb={self._lb}, ub={self._ub}"


class Bin(nn.Module):
    """Maps input to one of two values by sign."""

    def __init__(self, lb: float, ub: float):
        super().__init__()
        self._lb = float(lb)
        self._ub = float(ub)
        self._interval_size = self._ub - self._lb
        self._shrink_amount = self._interval_size / 2.0
        self._shift_amount = (self._ub + self._lb) / 2.0

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.where(x < 0, torch.full_like(x, self._lb), torch.full_like(x, self._ub))

    def extra_repr(self) -> str:
        return f"lb={self._lb}, ub={self._ub}"


class Slice(nn.Module):
    def __init__(self, from_index: int, to_index: int):
        super().__init__()
        self._from = int(from_index)
        self._to = int(to_index)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x[..., self._from : self._to]

    def extra_repr(self) -> str:
        return f"from_index={self._from}, to_index={self._to}"


class Round(nn.Module):
    def __init__(self, ndigits: int = 0):
        super().__init__()
        self._ndigits = int(ndigits)
        self._q = 10.0**self._ndigits

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.round(x * self._q) / self._q

    def extra_repr(self) -> str:
        return f"ndigits={self._ndigits}"


class Apply(nn.Module):
    """Applies a named torch function (e.g. 'tanh', 'relu') or an operator
    with a constant second argument."""

    def __init__(self, fn: Union[str, Callable], *args):
        super().__init__()
        self._fn = getattr(torch, fn) if isinstance(fn, str) else fn
        self._args = args

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self._fn(x, *self._args)

    def extra_repr(self) -> str:
        return f"fn={getattr(self._fn, '__name__', self._fn)}"


class RecurrentNet(nn.Module):
    """Base for hidden-state-in-signature recurrent layers."""

    def reset(self):  # compatibility no-op: state lives in the caller
        pass


class RNN(RecurrentNet):
    """Elman RNN with explicit hidden state: `forward(x, h=None) -> (y, h)`
    (reference layers.py:161)."""

    def __init__(self, input_size: int, hidden_size: int, nonlinearity: str = "tanh", *, dtype: torch.dtype = torch.float32):
        super().__init__()
        self.input_size = int(input_size)
        self.hidden_size = int(hidden_size)
        self._act = {"tanh": torch.tanh, "relu": torch.relu}[nonlinearity]
        k = 1.0 / math.sqrt(hidden_size)
        self.W_ih = nn.Parameter(torch.empty(hidden_size, input_size, dtype=dtype).uniform_(-k, k))
        self.W_hh = nn.Parameter(torch.empty(hidden_size, hidden_size, dtype=dtype).uniform_(-k, k))
        self.b_ih = nn.Parameter(torch.zeros(hidden_size, dtype=dtype))
        self.b_hh = nn.Parameter(torch.zeros(hidden_size, dtype=dtype))

    def forward(self, x: torch.Tensor, h: Optional[torch.Tensor] = None) -> Tuple[torch.Tensor, torch.Tensor]:
        if h is None:
            h = torch.zeros(x.shape[:-1] + (self.hidden_size,), dtype=x.dtype, device=x.device)
        new_h = self._act(x @ self.W_ih.T + self.b_ih + h @ self.W_hh.T + self.b_hh)
        return new_h, new_h

    def __repr__(self) -> str:
        return f"RNN(input_size={self.input_size}, hidden_size={self.hidden_size})"


class LSTM(RecurrentNet):
    """LSTM with explicit (h, c) state: `forward(x, hc=None) -> (y, (h, c))`
    (reference layers.py:221)."""

    def __init__(self, input_size: int, hidden_size: int, *, dtype: torch.dtype = torch.float32):
        super().__init__()
        self.input_size = int(input_size)
        self.hidden_size = int(hidden_size)
        k = 1.0 / math.sqrt(hidden_size)
        self.W_ih = nn.Parameter(torch.empty(4 * hidden_size, input_size, dtype=dtype).uniform_(-k, k))
        self.W_hh = nn.Parameter(torch.empty(4 * hidden_size, hidden_size, dtype=dtype).uniform_(-k, k))
        self.b_ih = nn.Parameter(torch.zeros(4 * hidden_size, dtype=dtype))
        self.b_hh = nn.Parameter(torch.zeros(4 * hidden_size, dtype=dtype))

    def forward(self, x: torch.Tensor, hc: Optional[Tuple[torch.Tensor, torch.Tensor]] = None):
        hs = self.hidden_size
        if hc is None:
            h = torch.zeros(x.shape[:-1] + (hs,), dtype=x.dtype, device=x.device)
            c = torch.zeros_like(h)
        else:
            h, c = hc
        gates = x @ self.W_ih.T + self.b_ih + h @ self.W_hh.T + self.b_hh
        i, f, g, o = gates.split(hs, dim=-1)
        i = torch.sigmoid(i)
        f = torch.sigmoid(f)
        g = torch.tanh(g)
        o = torch.sigmoid(o)
        new_c = f * c + i * g
        new_h = o * torch.tanh(new_c)
        return new_h, (new_h, new_c)

    def __repr__(self) -> str:
        return f"LSTM(input_size={self.input_size}, hidden_size={self.hidden_size})"


class FeedForwardNet(nn.Module):
    """MLP described by a list of (size, activation) layer tuples
    (reference layers.py:305)."""

    LengthActTuple = Tuple[int, Union[str, Callable]]
    LengthActBiasTuple = Tuple[int, Union[str, Callable], Union[bool]]

    def __init__(self, input_size: int, layers, *, dtype: torch.dtype = torch.float32):
        super().__init__()
        self._layers = nn.ModuleList()
        self._acts = []
        current = int(input_size)
        for layer in layers:
            if len(layer) == 2:
                size, act = layer
                bias = True
            else:
                size, act, bias = layer
            self._layers.append(nn.Linear(current, int(size), bias=bool(bias)).to(dtype))
            if isinstance(act, str):
                act = getattr(torch, act) if act != "none" else None
            self._acts.append(act)
            current = int(size)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for linear, act in zip(self._layers, self._acts):
            x = linear(x)
            if act is not None:
                x = act(x)
        return x


class StructuredControlNet(nn.Module):
    """Structured Control Net (Srouji et al. 2018): parallel linear and
    nonlinear (MLP) control streams, summed (reference layers.py:386)."""

    def __init__(self, *, in_features: int, out_features: int, num_layers: int, hidden_size: int, bias: bool = True, nonlinearity=torch.tanh):
        super().__init__()
        self._linear = nn.Linear(in_features, out_features, bias=bias)
        mlp_layers = []
        current = in_features
        for _ in range(num_layers):
            mlp_layers.append(nn.Linear(current, hidden_size, bias=bias))
            current = hidden_size
        mlp_layers.append(nn.Linear(current, out_features, bias=bias))
        self._mlp = nn.ModuleList(mlp_layers)
        self._nonlinearity = nonlinearity

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        linear_out = self._linear(x)
        y = x
        for i, layer in enumerate(self._mlp):
            y = layer(y)
            if i < len(self._mlp) - 1:
                y = self._nonlinearity(y)
        return linear_out + y


class LocomotorNet(nn.Module):
    """StructuredControlNet variant whose nonlinear stream is a sum of
    sinusoids (reference layers.py:487).

    Note: the sinusoid phase advances an internal time buffer per forward
    call, so this layer suits per-solution rollouts (GymNE-style); for
    vmapped population rollouts use an explicit-state design instead."""

    def __init__(self, *, in_features: int, out_features: int, bias: bool = True, num_sinusoids: int = 16):
        super().__init__()
        self._linear = nn.Linear(in_features, out_features, bias=bias)
        self._num_sinusoids = int(num_sinusoids)
        self._amplitudes = nn.Parameter(torch.randn(self._num_sinusoids, out_features) * 0.1)
        self._frequencies = nn.Parameter(torch.randn(self._num_sinusoids, out_features) * 0.1)
        self._phases = nn.Parameter(torch.randn(self._num_sinusoids, out_features) * 0.1)
        self.register_buffer("_t", torch.zeros(1))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        linear_out = self._linear(x)
        t = self._t
        sinusoid = (self._amplitudes * torch.sin(self._frequencies * t + self._phases)).sum(dim=0)
        self._t = t + 1.0
        return linear_out + sinusoid
