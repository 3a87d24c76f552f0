"""ReadOnlyTensor: a torch.Tensor subclass that blocks in-place mutation.

Re-design of the reference's `tools/readonlytensor.py`
(/root/reference/src/evotorch/tools/readonlytensor.py:27-226). Status dicts
and `SolutionBatch` accessors hand these out so that user code cannot
silently corrupt population storage shared with HIP kernels.
"""

from typing import Union

import torch

__all__ = ["ReadOnlyTensor", "read_only_tensor", "as_read_only_tensor"]


def _err(op: str):
    raise TypeError(f"ReadOnlyTensor does not allow the in-place/mutating operation {op!r}. Use `.clone()` to obtain a mutable copy.")


class ReadOnlyTensor(torch.Tensor):
    """A tensor view that raises on in-place mutation. Indexing/slicing
    returns ReadOnlyTensor views; arithmetic returns plain tensors."""

    @classmethod
    def __torch_function__(cls, func, types, args=(), kwargs=None):
        if kwargs is None:
            kwargs = {}
        name = getattr(func, "__name__", str(func))
        # Block in-place ops on a ReadOnlyTensor receiver, and out= into one.
        if name.endswith("_") and not name.endswith("__") and len(args) > 0 and isinstance(args[0], ReadOnlyTensor):
            _err(name)
        out = kwargs.get("out", None)
        outs = out if isinstance(out, (tuple, list)) else (out,)
        for o in outs:
            if isinstance(o, ReadOnlyTensor):
                _err(f"{name}(out=...)")
        if name in ("__setitem__", "copy_", "set_", "fill_", "zero_", "scatter_", "index_put_", "masked_fill_"):
            if len(args) > 0 and isinstance(args[0], ReadOnlyTensor):
                _err(name)
        result = super().__torch_function__(func, types, args, kwargs)
        # Keep views read-only; convert computed results to plain tensors.
        view_funcs = {
            "__getitem__", "view", "reshape", "expand", "expand_as", "permute", "transpose",
            "t", "squeeze", "unsqueeze", "narrow", "select", "split", "chunk", "flatten",
            "detach", "as_subclass", "contiguous", "to", "cpu", "cuda", "clone",
        }
        def demote(x):
            if isinstance(x, ReadOnlyTensor) and name not in view_funcs:
                return torch.Tensor.as_subclass(x, torch.Tensor)
            return x
        if name in ("clone",):
            # clone yields a mutable copy
            if isinstance(result, ReadOnlyTensor):
                return torch.Tensor.as_subclass(result, torch.Tensor)
            return result
        if isinstance(result, torch.Tensor):
            return demote(result)
        if isinstance(result, (tuple, list)):
            return type(result)(demote(r) for r in result)
        return result

    def __setitem__(self, *args, **kwargs):
        _err("__setitem__")

    def numpy(self):
        """Return a read-only numpy view."""
        arr = torch.Tensor.as_subclass(self, torch.Tensor).detach().cpu().numpy()
        arr.flags.writeable = False
        return arr

    def __repr__(self) -> str:
        inner = torch.Tensor.as_subclass(self, torch.Tensor).__repr__()
        return inner.replace("tensor(", "ReadOnlyTensor(", 1)


def as_read_only_tensor(x: Union[torch.Tensor, "ReadOnlyTensor"], *, dtype=None, device=None) -> ReadOnlyTensor:
    """Wrap `x` as a ReadOnlyTensor without copying (shares storage)."""
    t = torch.as_tensor(x, dtype=dtype, device=device)
    if isinstance(t, ReadOnlyTensor):
        return t
    return t.as_subclass(ReadOnlyTensor)


def read_only_tensor(x, *, dtype=None, device=None) -> ReadOnlyTensor:
    """Copying constructor for a ReadOnlyTensor."""
    t = torch.as_tensor(x, dtype=dtype, device=device).clone()
    return t.as_subclass(ReadOnlyTensor)
