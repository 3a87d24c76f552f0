"""ObjectArray: a 1-D container of arbitrary python objects with torch-like
indexing semantics.

Re-design of the reference's `tools/objectarray.py`
(/root/reference/src/evotorch/tools/objectarray.py:39-534). Stored items are
converted to immutables on write (aliasing safety — the reference's
store-immutable-clones policy), and `__getitem__` with a slice returns a
*view* sharing storage, mirroring SolutionBatch's shared-memory slicing.
"""

from typing import Iterable, Optional, Union

import numpy as np

from .immutable import as_immutable, mutable_copy

__all__ = ["ObjectArray", "as_object_array"]


class ObjectArray:
    dtype = object

    def __init__(self, size: Optional[int] = None, *, slice_of: Optional[tuple] = None):
        if slice_of is not None:
            source, start, stop = slice_of
            self._data = source._data
            self._start = start
            self._stop = stop
            self._read_only = source._read_only
        else:
            n = int(size) if size is not None else 0
            self._data = np.empty(n, dtype=object)
            self._start = 0
            self._stop = n
            self._read_only = False

    # -- properties ---------------------------------------------------------
    @property
    def shape(self):
        return (len(self),)

    @property
    def ndim(self) -> int:
        return 1

    @property
    def is_read_only(self) -> bool:
        return self._read_only

    def size(self):
        """torch.Size-style shape tuple (reference objectarray.py:208)."""
        import torch

        return torch.Size([len(self)])

    def dim(self) -> int:
        return 1

    def numel(self) -> int:
        return len(self)

    def set_item(self, i, x):
        """Explicit setter (reference objectarray.py:344): same as
        `self[i] = x` (stores an immutable clone)."""
        self[i] = x

    def repeat(self, count: int) -> "ObjectArray":
        """Concatenate `count` copies of this array (reference
        objectarray.py:244)."""
        count = int(count)
        out = ObjectArray(len(self) * count)
        for rep in range(count):
            for i in range(len(self)):
                out[rep * len(self) + i] = self[i]
        return out

    @staticmethod
    def from_numpy(arr) -> "ObjectArray":
        """Build from a numpy object (or any) 1-D array (reference
        objectarray.py:302)."""
        out = ObjectArray(len(arr))
        for i, item in enumerate(arr):
            out[i] = item
        return out

    @property
    def device(self) -> str:
        return "cpu"

    def set_read_only_(self):
        self._read_only = True
        return self

    def get_read_only_view(self) -> "ObjectArray":
        view = ObjectArray(slice_of=(self, self._start, self._stop))
        view._read_only = True
        return view

    def storage_ptr(self) -> int:
        return self._data.ctypes.data

    def __len__(self) -> int:
        return self._stop - self._start

    def _check_writable(self):
        if self._read_only:
            raise ValueError("This ObjectArray is read-only")

    def __getitem__(self, i):
        if isinstance(i, slice):
            start, stop, step = i.indices(len(self))
            if step != 1:
                result = ObjectArray(len(range(start, stop, step)))
                for j, src in enumerate(range(start, stop, step)):
                    result._data[j] = self._data[self._start + src]
                result._read_only = self._read_only
                return result
            return ObjectArray(slice_of=(self, self._start + start, self._start + stop))
        if isinstance(i, (list, np.ndarray)) or (hasattr(i, "ndim") and getattr(i, "ndim", 0) == 1):
            idx = np.asarray([int(j) for j in i])
            result = ObjectArray(len(idx))
            for j, src in enumerate(idx):
                result._data[j] = self._data[self._start + int(src)]
            result._read_only = self._read_only
            return result
        j = int(i)
        if j < 0:
            j += len(self)
        if not (0 <= j < len(self)):
            raise IndexError(f"index {i} out of range for ObjectArray of length {len(self)}")
        return self._data[self._start + j]

    def __setitem__(self, i, value):
        self._check_writable()
        if isinstance(i, slice):
            start, stop, step = i.indices(len(self))
            values = list(value)
            targets = list(range(start, stop, step))
            if len(values) != len(targets):
                raise ValueError("Length mismatch in slice assignment")
            for t, v in zip(targets, values):
                self._data[self._start + t] = as_immutable(v)
            return
        j = int(i)
        if j < 0:
            j += len(self)
        if not (0 <= j < len(self)):
            raise IndexError(f"index {i} out of range")
        self._data[self._start + j] = as_immutable(value)

    def __iter__(self):
        for i in range(len(self)):
            yield self[i]

    def clone(self, *, memo: Optional[dict] = None) -> "ObjectArray":
        """Deep clone: the elements themselves are cloned (reference
        objectarray.py:404 — `x.clone()[0] is not x[0]` for container
        elements), and the clone is always writable."""
        if memo is None:
            memo = {}
        if id(self) in memo:
            return memo[id(self)]
        result = ObjectArray(len(self))
        memo[id(self)] = result
        for i in range(len(self)):
            item = self._data[self._start + i]
            result._data[i] = as_immutable(mutable_copy(item)) if item is not None else None
        return result

    def __copy__(self) -> "ObjectArray":
        return self.clone()

    def __deepcopy__(self, memo: Optional[dict]) -> "ObjectArray":
        return self.clone(memo=memo if memo is not None else {})

    def untyped_storage(self):
        return self.storage()

    def storage(self):
        """torch-like storage handle exposing data_ptr() (reference
        objectarray.py:479; used by tools.storage_ptr)."""

        class _Storage:
            def __init__(self, ptr):
                self._ptr = ptr

            def data_ptr(self):
                return self._ptr

        return _Storage(self._data.ctypes.data)

    def numpy(self) -> np.ndarray:
        out = np.empty(len(self), dtype=object)
        for i in range(len(self)):
            out[i] = mutable_copy(self[i])
        return out

    def __eq__(self, other):
        if isinstance(other, (ObjectArray, list, tuple, np.ndarray)):
            if len(self) != len(other):
                return False
            import torch
            flags = []
            for a, b in zip(self, other):
                try:
                    r = a == b
                    if isinstance(r, (np.ndarray, torch.Tensor)):
                        r = bool(np.all(np.asarray(r)))
                    flags.append(bool(r))
                except Exception:
                    flags.append(a is b)
            return np.array(flags)
        return NotImplemented

    def __repr__(self) -> str:
        return f"ObjectArray({[self[i] for i in range(len(self))]!r})"


def as_object_array(x: Union[Iterable, ObjectArray]) -> ObjectArray:
    if isinstance(x, ObjectArray):
        return x
    items = list(x)
    result = ObjectArray(len(items))
    for i, item in enumerate(items):
        result[i] = item
    return result
