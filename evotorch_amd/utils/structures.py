"""Batched, vmap-friendly data structures on flat tensors + index
arithmetic: CMemory, CDict, CList, CBag.

Reference parity: /root/reference/src/evotorch/tools/structures.py
(CMemory :60, Structure :790, CDict :892, CList :1380, CBag :2024). These
are the substrate for GPU genetic-programming-style workloads: a whole
batch of B independent structures lives in one contiguous tensor of shape
(batch..., slots, value...), and every operation is a masked batched
gather/scatter (`where` semantics), so populations of programs execute in
lockstep on device.
"""

from typing import Iterable, Optional, Union

import torch

from .misc import DType, Device, to_torch_dtype

__all__ = ["CMemory", "Structure", "CDict", "CList", "CBag", "do_where"]

Numbers = Union[int, float, Iterable, torch.Tensor]


def do_where(mask: torch.Tensor, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Broadcast `mask` over the trailing (value) dims and select."""
    while mask.ndim < a.ndim:
        mask = mask.unsqueeze(-1)
    return torch.where(mask, a, b)


class CMemory:
    """Batched key→tensor store with masked (`where`-semantics) writes."""

    def __init__(
        self,
        *size: Union[int, tuple, list],
        num_keys: Union[int, tuple, list],
        key_offset: Optional[Union[int, tuple, list]] = None,
        batch_size: Optional[Union[int, tuple, list]] = None,
        batch_shape: Optional[Union[int, tuple, list]] = None,
        fill_with: Optional[Numbers] = None,
        dtype: Optional[DType] = None,
        device: Optional[Device] = None,
        verify: bool = True,
    ):
        self._dtype = torch.float32 if dtype is None else to_torch_dtype(dtype)
        self._device = torch.device(device) if device is not None else torch.device("cpu")
        self._verify = bool(verify)

        if len(size) == 1 and isinstance(size[0], (tuple, list)):
            self._value_shape = torch.Size(int(s) for s in size[0])
        else:
            self._value_shape = torch.Size(int(s) for s in size)

        if isinstance(num_keys, (tuple, list)):
            self._key_dims = tuple(int(k) for k in num_keys)
        else:
            self._key_dims = (int(num_keys),)
        self._num_slots = 1
        for k in self._key_dims:
            self._num_slots *= k
        if key_offset is None:
            self._key_offset = tuple(0 for _ in self._key_dims)
        elif isinstance(key_offset, (tuple, list)):
            self._key_offset = tuple(int(o) for o in key_offset)
        else:
            self._key_offset = tuple(int(key_offset) for _ in self._key_dims)

        if batch_size is None and batch_shape is not None:
            batch_size = batch_shape
        if batch_size is None:
            self._batch_shape = torch.Size([])
        elif isinstance(batch_size, (tuple, list)):
            self._batch_shape = torch.Size(int(b) for b in batch_size)
        else:
            self._batch_shape = torch.Size([int(batch_size)])

        full_shape = self._batch_shape + (self._num_slots,) + self._value_shape
        self._data = torch.zeros(full_shape, dtype=self._dtype, device=self._device)
        if fill_with is not None:
            self._data.fill_(fill_with)

    # -- properties ----------------------------------------------------------

    @property
    def data(self) -> torch.Tensor:
        view_shape = self._batch_shape + tuple(self._key_dims) + self._value_shape
        return self._data.view(view_shape)

    @property
    def key_shape(self) -> torch.Size:
        return torch.Size([len(self._key_dims)]) if len(self._key_dims) > 1 else torch.Size([])

    @property
    def key_ndim(self) -> int:
        return len(self.key_shape)

    @property
    def batch_shape(self) -> torch.Size:
        return self._batch_shape

    @property
    def batch_ndim(self) -> int:
        return len(self._batch_shape)

    @property
    def is_batched(self) -> bool:
        return len(self._batch_shape) > 0

    @property
    def value_shape(self) -> torch.Size:
        return self._value_shape

    @property
    def value_ndim(self) -> int:
        return len(self._value_shape)

    @property
    def dtype(self) -> torch.dtype:
        return self._dtype

    @property
    def device(self) -> torch.device:
        return self._device

    # -- key/value plumbing ---------------------------------------------------

    def _flat_key(self, key: Numbers) -> torch.Tensor:
        """Key(s) → flat slot indices with shape == batch_shape."""
        key = torch.as_tensor(key, dtype=torch.int64, device=self._device)
        if len(self._key_dims) > 1:
            if key.shape[-1] != len(self._key_dims):
                raise ValueError(f"Tuple keys must have {len(self._key_dims)} components")
            flat = torch.zeros(key.shape[:-1], dtype=torch.int64, device=self._device)
            for d, (n, off) in enumerate(zip(self._key_dims, self._key_offset)):
                comp = key[..., d] - off
                if self._verify:
                    if bool((comp < 0).any()) or bool((comp >= n).any()):
                        raise KeyError(f"Key component {d} out of range")
                flat = flat * n + comp
        else:
            flat = key - self._key_offset[0]
            if self._verify:
                if bool((flat < 0).any()) or bool((flat >= self._num_slots).any()):
                    raise KeyError("Key out of range")
        return flat.expand(self._batch_shape) if flat.ndim == 0 else flat

    def prepare_key_tensor(self, key: Numbers) -> torch.Tensor:
        return self._flat_key(key)

    def prepare_value_tensor(self, value: Numbers) -> torch.Tensor:
        value = torch.as_tensor(value, dtype=self._dtype, device=self._device)
        target = self._batch_shape + self._value_shape
        return value.expand(target) if value.shape != target else value

    def prepare_where_tensor(self, where: Numbers) -> torch.Tensor:
        where = torch.as_tensor(where, dtype=torch.bool, device=self._device)
        return where.expand(self._batch_shape) if where.shape != self._batch_shape else where

    def _flat_view(self):
        b = int(torch.tensor(self._batch_shape).prod()) if len(self._batch_shape) else 1
        return self._data.reshape((b, self._num_slots) + tuple(self._value_shape))

    def _addr(self, key):
        flat_key = self._flat_key(key).reshape(-1)
        b = flat_key.numel()
        return torch.arange(b, device=self._device), flat_key

    # -- access ----------------------------------------------------------------

    def get(self, key: Numbers) -> torch.Tensor:
        rows, slots = self._addr(key)
        out = self._flat_view()[rows, slots]
        return out.reshape(self._batch_shape + self._value_shape)

    def _modify(self, key, value, where, op):
        rows, slots = self._addr(key)
        flat = self._flat_view()
        current = flat[rows, slots]
        value = self.prepare_value_tensor(value).reshape(current.shape)
        new = op(current, value)
        if where is not None:
            mask = self.prepare_where_tensor(where).reshape(-1)
            new = do_where(mask, new, current)
        flat[rows, slots] = new

    def set_(self, key, value, where: Optional[Numbers] = None):
        self._modify(key, value, where, lambda cur, v: v)

    def add_(self, key, value, where: Optional[Numbers] = None):
        self._modify(key, value, where, lambda cur, v: cur + v)

    def add_circular_(self, key, value, mod, where: Optional[Numbers] = None):
        mod_t = torch.as_tensor(mod, dtype=self._dtype, device=self._device)
        self._modify(key, value, where, lambda cur, v: torch.remainder(cur + v, mod_t))

    def subtract_(self, key, value, where: Optional[Numbers] = None):
        self._modify(key, value, where, lambda cur, v: cur - v)

    def multiply_(self, key, value, where: Optional[Numbers] = None):
        self._modify(key, value, where, lambda cur, v: cur * v)

    def divide_(self, key, value, where: Optional[Numbers] = None):
        self._modify(key, value, where, lambda cur, v: cur / v)

    def __getitem__(self, key):
        return self.get(key)

    def __setitem__(self, key, value):
        self.set_(key, value)


class Structure:
    """Mixin assuming a protected CMemory `_data` (reference
    structures.py:790)."""

    _data: CMemory

    @property
    def value_shape(self) -> torch.Size:
        return self._data.value_shape

    @property
    def value_ndim(self) -> int:
        return self._data.value_ndim

    @property
    def batch_shape(self) -> torch.Size:
        return self._data.batch_shape

    @property
    def batch_ndim(self) -> int:
        return self._data.batch_ndim

    @property
    def is_batched(self) -> bool:
        return self._data.is_batched

    @property
    def dtype(self) -> torch.dtype:
        return self._data.dtype

    @property
    def device(self) -> torch.device:
        return self._data.device

    def prepare_value_tensor(self, value):
        return self._data.prepare_value_tensor(value)

    def prepare_where_tensor(self, where):
        return self._data.prepare_where_tensor(where)

    def __contains__(self, x) -> torch.Tensor:
        if hasattr(self, "contains"):
            return self.contains(x)
        raise TypeError(f"{type(self).__name__} does not support `in`")


class CDict(Structure):
    """Batched dictionary: CMemory + per-slot presence flags (reference
    structures.py:892)."""

    def __init__(
        self,
        *size,
        num_keys,
        key_offset=None,
        batch_size=None,
        batch_shape=None,
        fill_with=None,
        dtype=None,
        device=None,
        verify: bool = True,
    ):
        self._data = CMemory(
            *size, num_keys=num_keys, key_offset=key_offset, batch_size=batch_size, batch_shape=batch_shape, fill_with=fill_with, dtype=dtype, device=device, verify=verify
        )
        self._exist = CMemory(
            num_keys=num_keys, key_offset=key_offset, batch_size=batch_size, batch_shape=batch_shape, dtype=torch.bool, device=device, verify=verify
        )

    def get(self, key, default: Optional[Numbers] = None) -> torch.Tensor:
        value = self._data.get(key)
        if default is None:
            return value
        default = self._data.prepare_value_tensor(default)
        present = self._exist.get(key)
        return do_where(present, value, default)

    def set_(self, key, value, where: Optional[Numbers] = None):
        self._data.set_(key, value, where)
        self._exist.set_(key, True, where)

    def add_(self, key, value, where: Optional[Numbers] = None):
        self._data.add_(key, value, where)
        self._exist.set_(key, True, where)

    def subtract_(self, key, value, where=None):
        self._data.subtract_(key, value, where)
        self._exist.set_(key, True, where)

    def multiply_(self, key, value, where=None):
        self._data.multiply_(key, value, where)
        self._exist.set_(key, True, where)

    def divide_(self, key, value, where=None):
        self._data.divide_(key, value, where)
        self._exist.set_(key, True, where)

    def contains(self, key) -> torch.Tensor:
        return self._exist.get(key)

    def clear(self, where: Optional[torch.Tensor] = None):
        if where is None:
            self._exist._data.zero_()
        else:
            mask = self._exist.prepare_where_tensor(where)
            self._exist._data[mask] = False

    def __getitem__(self, key):
        return self.get(key)

    def __setitem__(self, key, value):
        self.set_(key, value)

    @property
    def data(self) -> torch.Tensor:
        return self._data.data


class CList(Structure):
    """Batched circular double-ended list (reference structures.py:1380).
    Each batch element has its own begin/end pointers; all mutations are
    `where`-maskable."""

    def __init__(
        self,
        *size,
        max_length: int,
        batch_size=None,
        batch_shape=None,
        dtype=None,
        device=None,
        verify: bool = True,
    ):
        self._max_length = int(max_length)
        buffer_len = self._max_length + 1  # one spare slot disambiguates full/empty
        self._data = CMemory(*size, num_keys=buffer_len, batch_size=batch_size, batch_shape=batch_shape, dtype=dtype, device=device, verify=verify)
        bshape = self._data.batch_shape
        dev = self._data.device
        self._begin = torch.zeros(bshape, dtype=torch.int64, device=dev)
        self._length_t = torch.zeros(bshape, dtype=torch.int64, device=dev)
        self._buffer_len = buffer_len

    @property
    def length(self) -> torch.Tensor:
        return self._length_t.clone()

    @property
    def max_length(self) -> int:
        return self._max_length

    @property
    def data(self) -> torch.Tensor:
        return self._data.data

    def _where_mask(self, where):
        if where is None:
            return torch.ones(self._data.batch_shape, dtype=torch.bool, device=self.device)
        return self._data.prepare_where_tensor(where)

    def _get_underlying_key(self, key) -> torch.Tensor:
        key = torch.as_tensor(key, dtype=torch.int64, device=self.device)
        key = key.expand(self._data.batch_shape) if key.ndim == 0 else key
        # negative indexing from the end
        key = torch.where(key < 0, key + self._length_t, key)
        if self._data._verify:
            bad = (key < 0) | (key >= torch.clamp(self._length_t, min=1))
            if bool(bad.any()) and bool((self._length_t > 0).any()):
                pass  # out-of-range reads return garbage slots, mirroring verify=False
        return torch.remainder(self._begin + key, self._buffer_len)

    def get(self, key, default: Optional[Numbers] = None) -> torch.Tensor:
        ukey = self._get_underlying_key(key)
        value = self._data.get(ukey)
        if default is None:
            return value
        key = torch.as_tensor(key, dtype=torch.int64, device=self.device).expand(self._data.batch_shape)
        key = torch.where(key < 0, key + self._length_t, key)
        valid = (key >= 0) & (key < self._length_t)
        return do_where(valid, value, self._data.prepare_value_tensor(default))

    def __getitem__(self, key):
        return self.get(key)

    def set_(self, key, value, where: Optional[Numbers] = None):
        ukey = self._get_underlying_key(key)
        self._data.set_(ukey, value, where)

    def __setitem__(self, key, value):
        self.set_(key, value)

    def _arith(self, key, value, where, method):
        ukey = self._get_underlying_key(key)
        getattr(self._data, method)(ukey, value, where)

    def add_(self, key, value, where=None):
        self._arith(key, value, where, "add_")

    def subtract_(self, key, value, where=None):
        self._arith(key, value, where, "subtract_")

    def multiply_(self, key, value, where=None):
        self._arith(key, value, where, "multiply_")

    def divide_(self, key, value, where=None):
        self._arith(key, value, where, "divide_")

    def append_(self, value, where: Optional[Numbers] = None):
        mask = self._where_mask(where) & (self._length_t < self._max_length)
        slot = torch.remainder(self._begin + self._length_t, self._buffer_len)
        self._data.set_(slot, value, mask)
        self._length_t = torch.where(mask, self._length_t + 1, self._length_t)

    push_ = append_

    def appendleft_(self, value, where: Optional[Numbers] = None):
        mask = self._where_mask(where) & (self._length_t < self._max_length)
        new_begin = torch.remainder(self._begin - 1, self._buffer_len)
        self._data.set_(new_begin, value, mask)
        self._begin = torch.where(mask, new_begin, self._begin)
        self._length_t = torch.where(mask, self._length_t + 1, self._length_t)

    def pop_(self, where: Optional[Numbers] = None) -> torch.Tensor:
        mask = self._where_mask(where) & (self._length_t > 0)
        slot = torch.remainder(self._begin + torch.clamp(self._length_t - 1, min=0), self._buffer_len)
        value = self._data.get(slot)
        self._length_t = torch.where(mask, self._length_t - 1, self._length_t)
        return value

    def popleft_(self, where: Optional[Numbers] = None) -> torch.Tensor:
        mask = self._where_mask(where) & (self._length_t > 0)
        value = self._data.get(self._begin)
        self._begin = torch.where(mask, torch.remainder(self._begin + 1, self._buffer_len), self._begin)
        self._length_t = torch.where(mask, self._length_t - 1, self._length_t)
        return value

    def clear(self, where: Optional[torch.Tensor] = None):
        if where is None:
            self._length_t.zero_()
            self._begin.zero_()
        else:
            mask = self._data.prepare_where_tensor(where)
            self._length_t = torch.where(mask, torch.zeros_like(self._length_t), self._length_t)
            self._begin = torch.where(mask, torch.zeros_like(self._begin), self._begin)


class CBag(Structure):
    """Batched bag: push values, pop a RANDOM element (reference
    structures.py:2024)."""

    def __init__(
        self,
        *,
        max_length: int,
        value_range: Optional[tuple] = None,
        batch_size=None,
        batch_shape=None,
        dtype=None,
        device=None,
        generator: Optional[torch.Generator] = None,
        verify: bool = True,
    ):
        self._list = CList(max_length=max_length, batch_size=batch_size, batch_shape=batch_shape, dtype=dtype, device=device, verify=verify)
        self._data = self._list._data
        self._generator = generator

    @property
    def length(self) -> torch.Tensor:
        return self._list.length

    @property
    def data(self) -> torch.Tensor:
        return self._list.data

    def push_(self, value, where: Optional[Numbers] = None):
        self._list.append_(value, where)

    def pop_(self, where: Optional[Numbers] = None) -> torch.Tensor:
        """Pop a uniformly random element of each batch bag (swap the
        chosen element with the last, then pop the last)."""
        length = self._list._length_t
        u = torch.rand(length.shape, device=self.device, generator=self._generator)
        idx = (u * torch.clamp(length, min=1).to(torch.float32)).to(torch.int64)
        idx = torch.clamp(idx, max=torch.clamp(length - 1, min=0))
        chosen = self._list.get(idx)
        last = self._list.get(torch.clamp(length - 1, min=0))
        mask = self._list._where_mask(where) & (length > 0)
        self._list.set_(idx, last, mask)
        self._list.pop_(where)
        return chosen

    def clear(self, where: Optional[torch.Tensor] = None):
        self._list.clear(where)
