"""Deep cloning with memoization, and Clonable/Serializable mixins.

Re-design of the reference's `tools/cloning.py`
(/root/reference/src/evotorch/tools/cloning.py:25-289).
"""

import copy as _py_copy
from typing import Any, Optional

import numpy as np
import torch

from .immutable import ImmutableContainer, mutable_copy
from .readonlytensor import ReadOnlyTensor

__all__ = ["deep_clone", "clone", "Clonable", "Serializable", "ReadOnlyClonable"]


def deep_clone(
    x: Any,
    *,
    otherwise_deepcopy: bool = False,
    otherwise_return: bool = False,
    otherwise_fail: bool = False,
    memo: Optional[dict] = None,
) -> Any:
    """Recursively clone tensors/arrays/containers/Clonables; non-clonable
    leaves are handled per the `otherwise_*` policy (exactly one may be
    True; default policy is `otherwise_deepcopy`)."""
    if memo is None:
        memo = {}
    k = id(x)
    if k in memo:
        return memo[k]
    chosen = [otherwise_deepcopy, otherwise_return, otherwise_fail]
    if sum(bool(c) for c in chosen) == 0:
        otherwise_deepcopy = True
    elif sum(bool(c) for c in chosen) > 1:
        raise ValueError("At most one `otherwise_*` policy may be chosen")

    if isinstance(x, ReadOnlyTensor):
        result = torch.Tensor.as_subclass(torch.Tensor.as_subclass(x, torch.Tensor).clone(), ReadOnlyTensor)
    elif isinstance(x, torch.Tensor):
        result = x.clone()
    elif isinstance(x, np.ndarray):
        result = x.copy()
    elif isinstance(x, Clonable):
        result = x.clone(memo=memo)
    elif isinstance(x, (int, float, complex, bool, str, bytes, type(None), type(Ellipsis))):
        result = x
    elif isinstance(x, dict):
        result = {}
        memo[k] = result
        for key, val in x.items():
            result[key] = deep_clone(val, otherwise_deepcopy=otherwise_deepcopy, otherwise_return=otherwise_return, otherwise_fail=otherwise_fail, memo=memo)
        return result
    elif isinstance(x, list):
        result = []
        memo[k] = result
        for val in x:
            result.append(deep_clone(val, otherwise_deepcopy=otherwise_deepcopy, otherwise_return=otherwise_return, otherwise_fail=otherwise_fail, memo=memo))
        return result
    elif isinstance(x, tuple):
        result = tuple(
            deep_clone(val, otherwise_deepcopy=otherwise_deepcopy, otherwise_return=otherwise_return, otherwise_fail=otherwise_fail, memo=memo) for val in x
        )
    elif isinstance(x, ImmutableContainer):
        result = mutable_copy(x)
    elif otherwise_deepcopy:
        result = _py_copy.deepcopy(x, memo)
    elif otherwise_return:
        result = x
    else:
        raise TypeError(f"Do not know how to clone {type(x)}")
    memo[k] = result
    return result


def clone(x: Any, **kwargs) -> Any:
    """Clone `x`: uses `.clone()` if available, otherwise deep_clone."""
    if isinstance(x, (torch.Tensor, np.ndarray)) or hasattr(x, "clone"):
        if isinstance(x, np.ndarray):
            return x.copy()
        return x.clone(**kwargs) if isinstance(x, Clonable) else x.clone()
    return deep_clone(x, **kwargs)


class Clonable:
    """Mixin providing memo-aware `clone()` and `__deepcopy__` via a
    subclass-implemented `_get_cloned_state(memo=...)`."""

    def _get_cloned_state(self, *, memo: dict) -> dict:
        raise NotImplementedError

    def clone(self, *, memo: Optional[dict] = None):
        if memo is None:
            memo = {}
        if id(self) in memo:
            return memo[id(self)]
        new_obj = object.__new__(type(self))
        memo[id(self)] = new_obj
        state = self._get_cloned_state(memo=memo)
        new_obj.__dict__.update(state)
        return new_obj

    def __deepcopy__(self, memo: Optional[dict]):
        if memo is None:
            memo = {}
        return self.clone(memo=memo)


class Serializable(Clonable):
    """Clonable that pickles through its cloned state."""

    def __getstate__(self) -> dict:
        memo = {id(self): self}
        return self._get_cloned_state(memo=memo)

    def __setstate__(self, state: dict):
        self.__dict__.update(state)


class ReadOnlyClonable(Clonable):
    """Clonable whose `clone()` returns a read-only clone by default;
    `clone(preserve_read_only=False)` gives a mutable one."""

    def _get_cloned_state(self, *, memo: dict) -> dict:
        raise NotImplementedError

    def clone(self, *, memo: Optional[dict] = None, preserve_read_only: bool = True):
        result = Clonable.clone(self, memo=memo)
        if not preserve_read_only:
            result = result._as_mutable() if hasattr(result, "_as_mutable") else result
        return result
