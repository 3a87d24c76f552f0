"""Immutable container types and `as_immutable`.

Re-design of the reference's `tools/immutable.py`
(/root/reference/src/evotorch/tools/immutable.py:50-289).
"""

from collections.abc import Mapping, Sequence, Set
from typing import Any

import numpy as np
import torch

from .readonlytensor import ReadOnlyTensor, as_read_only_tensor

__all__ = [
    "ImmutableSequence",
    "is_immutable_container_or_tensor","as_immutable", "mutable_copy", "ImmutableContainer", "ImmutableList", "ImmutableSet", "ImmutableDict"]


class ImmutableContainer:
    """Marker base class for immutable containers."""


class ImmutableList(ImmutableContainer, Sequence):
    def __init__(self, items):
        self._items = tuple(as_immutable(x) for x in items)

    def __getitem__(self, i):
        if isinstance(i, slice):
            return ImmutableList(self._items[i])
        return self._items[i]

    def __len__(self):
        return len(self._items)

    def __iter__(self):
        return iter(self._items)

    def __eq__(self, other):
        if isinstance(other, (list, tuple, ImmutableList)):
            return len(self) == len(other) and all(_eq(a, b) for a, b in zip(self, other))
        return NotImplemented

    def __hash__(self):
        return hash(tuple(id(type(x)) for x in self._items))

    def __repr__(self):
        return f"ImmutableList({list(self._items)!r})"


class ImmutableSet(ImmutableContainer, Set):
    def __init__(self, items):
        self._items = frozenset(as_immutable(x) for x in items)

    def __contains__(self, x):
        return x in self._items

    def __iter__(self):
        return iter(self._items)

    def __len__(self):
        return len(self._items)

    def __repr__(self):
        return f"ImmutableSet({set(self._items)!r})"


class ImmutableDict(ImmutableContainer, Mapping):
    def __init__(self, mapping):
        self._data = {k: as_immutable(v) for k, v in dict(mapping).items()}

    def __getitem__(self, k):
        return self._data[k]

    def __iter__(self):
        return iter(self._data)

    def __len__(self):
        return len(self._data)

    def __repr__(self):
        return f"ImmutableDict({self._data!r})"


ImmutableSequence = ImmutableList  # reference alias (immutable.py:195)


def is_immutable_container_or_tensor(x) -> bool:
    """True for ImmutableList/Set/Dict and read-only tensors (reference
    tools/immutable.py)."""
    from .readonlytensor import ReadOnlyTensor

    return isinstance(x, (ImmutableContainer, ReadOnlyTensor))


def _eq(a, b) -> bool:
    if isinstance(a, torch.Tensor) and isinstance(b, torch.Tensor):
        return a.shape == b.shape and bool(torch.equal(torch.Tensor.as_subclass(a, torch.Tensor), torch.Tensor.as_subclass(b, torch.Tensor)))
    try:
        return bool(a == b)
    except Exception:
        return a is b


def as_immutable(x: Any) -> Any:
    """Deep-convert `x` into an immutable equivalent: tensors become
    read-only clones, containers become Immutable* wrappers, scalars pass
    through."""
    if isinstance(x, ReadOnlyTensor):
        return x
    if isinstance(x, torch.Tensor):
        return as_read_only_tensor(x.clone())
    if hasattr(x, "get_read_only_view") and hasattr(x, "is_read_only"):
        # TensorFrame (and compatible containers): immutable = read-only view
        return x if x.is_read_only else x.get_read_only_view()
    if isinstance(x, np.ndarray):
        if x.dtype == object:
            return ImmutableList(list(x))
        arr = x.copy()
        arr.flags.writeable = False
        return arr
    if isinstance(x, Mapping):
        return ImmutableDict(x)
    if isinstance(x, (set, frozenset)):
        return ImmutableSet(x)
    if isinstance(x, (list, tuple)):
        return ImmutableList(x)
    return x


def mutable_copy(x: Any) -> Any:
    """Deep-convert an immutable object back to a plain mutable one."""
    if isinstance(x, ReadOnlyTensor):
        return torch.Tensor.as_subclass(x, torch.Tensor).clone()
    if isinstance(x, np.ndarray):
        return x.copy()
    if isinstance(x, ImmutableDict):
        return {k: mutable_copy(v) for k, v in x.items()}
    if isinstance(x, ImmutableList):
        return [mutable_copy(v) for v in x]
    if isinstance(x, ImmutableSet):
        return {mutable_copy(v) for v in x}
    if isinstance(x, dict):
        return {k: mutable_copy(v) for k, v in x.items()}
    if isinstance(x, (list, tuple)):
        return type(x)(mutable_copy(v) for v in x)
    return x
