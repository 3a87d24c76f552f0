"""TensorFrame: a pandas-like columnar container of torch tensors.

Reference parity: /root/reference/src/evotorch/tools/tensorframe.py:53-1338
(sort/argsort :807/:868, hstack/vstack :881/:922, `each` vmapped row-fn
:953, `pick` indexer :1270). All columns live on one device; row-wise
operations are batched tensor ops, never python loops.
"""

from collections.abc import Mapping
from typing import Callable, Optional, Union

import torch

from .recursiveprintable import RecursivePrintable

__all__ = ["TensorFrame"]


class _PickIndexer:
    """frame.pick[rows] / frame.pick[rows, cols] — getter returns a new
    (copied) frame; SETTER writes in place into the frame's own storage
    (reference tensorframe.py:1270: `tbl.pick[1:4, "X"] = ...`)."""

    def __init__(self, frame: "TensorFrame"):
        self._frame = frame

    @staticmethod
    def _normalize_cols(frame: "TensorFrame", cols) -> list:
        if isinstance(cols, str):
            return [cols]
        if isinstance(cols, slice):
            return list(frame._columns.keys())[cols]
        return list(cols)

    def __getitem__(self, index) -> "TensorFrame":
        frame = self._frame
        if isinstance(index, tuple) and len(index) == 2:
            rows, cols = index
            names = self._normalize_cols(frame, cols)
            sub = TensorFrame({c: frame._columns[c] for c in names})
            return sub.pick[rows]
        if isinstance(index, (int,)):
            index = slice(index, index + 1)
        return TensorFrame({name: col[index] for name, col in frame._columns.items()})

    def __setitem__(self, index, value):
        frame = self._frame
        if frame._read_only:
            raise TypeError("This TensorFrame is read-only")
        if isinstance(index, tuple) and len(index) == 2:
            rows, cols = index
            names = self._normalize_cols(frame, cols)
        else:
            rows = index
            names = list(frame._columns.keys())
        if isinstance(value, TensorFrame):
            value = value._columns
        if isinstance(value, Mapping):
            for name in names:
                frame._columns[name][rows] = torch.as_tensor(value[name])
            return
        if len(names) != 1:
            raise ValueError("Assigning a single tensor requires exactly one target column")
        frame._columns[names[0]][rows] = torch.as_tensor(value)


class TensorFrame(RecursivePrintable):
    def __init__(self, data: Optional[Union[Mapping, "TensorFrame"]] = None, *, read_only: bool = False, device=None, _copy: bool = True, **kwargs):
        if isinstance(data, TensorFrame):
            columns = dict(data._columns)
        elif data is not None:
            columns = dict(data)
        else:
            columns = {}
        columns.update(kwargs)
        self._columns = {}
        n = None
        for name, col in columns.items():
            if isinstance(col, torch.Tensor):
                # own our storage: pandas-style isolation — in-place writes
                # through pick[...] must never touch the caller's tensor
                t = torch.Tensor.as_subclass(col, torch.Tensor)
                t = t.clone() if _copy else t
            else:
                t = torch.as_tensor(col)
            if device is not None:
                t = t.to(device)
            if t.ndim == 0:
                t = t.reshape(1)
            if n is None:
                n = t.shape[0]
            elif t.shape[0] != n:
                if t.shape[0] == 1:
                    t = t.expand((n,) + t.shape[1:]).clone()
                else:
                    raise ValueError(f"Column {name!r} has {t.shape[0]} rows, expected {n}")
            self._columns[str(name)] = t
        self._read_only = bool(read_only)
        if read_only:
            from .readonlytensor import as_read_only_tensor

            self._columns = {k: as_read_only_tensor(v) for k, v in self._columns.items()}

    # -- basics --------------------------------------------------------------

    @property
    def columns(self) -> list:
        return list(self._columns.keys())

    @property
    def is_read_only(self) -> bool:
        return self._read_only

    def __len__(self) -> int:
        for col in self._columns.values():
            return col.shape[0]
        return 0

    @property
    def device(self):
        for col in self._columns.values():
            return col.device
        return torch.device("cpu")

    def __getitem__(self, name: str) -> torch.Tensor:
        if isinstance(name, str):
            return self._columns[name]
        return self.pick[name]

    def __setitem__(self, name: str, value):
        if self._read_only:
            raise TypeError("This TensorFrame is read-only")
        value = torch.as_tensor(value)
        if value.ndim == 0 and len(self) > 0:
            value = value.expand(len(self)).clone()
        if len(self._columns) > 0 and value.shape[0] != len(self):
            raise ValueError(f"Column of {value.shape[0]} rows cannot join a frame of {len(self)} rows")
        self._columns[str(name)] = value

    def __contains__(self, name: str) -> bool:
        return name in self._columns

    def __getattr__(self, name: str):
        cols = object.__getattribute__(self, "_columns") if "_columns" in self.__dict__ else {}
        if name in cols:
            return cols[name]
        raise AttributeError(name)

    @property
    def pick(self) -> _PickIndexer:
        """Row indexer: frame.pick[rows] / frame.pick[rows, cols]."""
        return _PickIndexer(self)

    # -- transforms -----------------------------------------------------------

    def with_columns(self, **kwargs) -> "TensorFrame":
        """New frame with the given columns added or replaced; the
        read-only flag is preserved (reference tensorframe.py)."""
        new = dict(self._columns)
        for k, v in kwargs.items():
            new[k] = torch.as_tensor(v)
        return TensorFrame(new, read_only=self._read_only)

    def without_columns(self, *names: str) -> "TensorFrame":
        return TensorFrame({k: v for k, v in self._columns.items() if k not in names})

    def to(self, device) -> "TensorFrame":
        return TensorFrame({k: v.to(device) for k, v in self._columns.items()})

    def argsort(self, by: str, *, descending: bool = False) -> torch.Tensor:
        return self._columns[by].argsort(descending=descending)

    def sort(self, by: str, *, descending: bool = False) -> "TensorFrame":
        order = self.argsort(by, descending=descending)
        return self.pick[order]

    def sort_values(self, by: str, *, ascending: bool = True) -> "TensorFrame":
        return self.sort(by, descending=not ascending)

    def hstack(self, other: "TensorFrame", *, override: bool = False) -> "TensorFrame":
        new = dict(self._columns)
        for k, v in other._columns.items():
            if k in new and not override:
                raise ValueError(f"Column {k!r} exists in both frames (use override=True)")
            new[k] = v
        return TensorFrame(new)

    def vstack(self, other: "TensorFrame") -> "TensorFrame":
        if set(self.columns) != set(other.columns):
            raise ValueError("vstack requires identical column sets")
        for k in self._columns:
            a, b = self._columns[k], other._columns[k]
            if a.ndim != b.ndim or a.shape[1:] != b.shape[1:]:
                raise ValueError(f"Column {k!r}: row shapes {tuple(a.shape[1:])} vs {tuple(b.shape[1:])} cannot vstack")
        return TensorFrame({k: torch.cat([self._columns[k], other._columns[k]], dim=0) for k in self._columns})

    def each(self, fn: Callable, *, join: bool = False, override: bool = False) -> "TensorFrame":
        """Apply a row-wise function (receiving a dict of row tensors) via
        vmap; returns a frame of the (dict) outputs. With join=True the
        outputs are hstacked onto self."""

        keys = list(self._columns.keys())

        def row_fn(*tensors):
            row = dict(zip(keys, tensors))
            out = fn(row)
            if not isinstance(out, Mapping):
                raise TypeError("each(fn): fn must return a dict")
            return tuple(out[k] for k in sorted(out.keys()))

        # probe output keys with row 0
        probe = fn({k: v[0] for k, v in self._columns.items()})
        out_keys = sorted(probe.keys())
        outputs = torch.func.vmap(row_fn)(*[self._columns[k] for k in keys])
        result = TensorFrame({k: t for k, t in zip(out_keys, outputs)})
        if join:
            return self.hstack(result, override=override)
        return result

    def cpu(self) -> "TensorFrame":
        return self.to("cpu")

    def cuda(self, device=None) -> "TensorFrame":
        return self.to(torch.device("cuda") if device is None else torch.device(device))

    def as_tensor(self, x, *, to_work_with: Optional[str] = None, broadcast_if_scalar: bool = False) -> torch.Tensor:
        """Coerce `x` to a tensor on this frame's device; with
        `to_work_with` (a column name or tensor), match that column's dtype
        and, if `broadcast_if_scalar`, expand a 0-dim result to the
        column's leading length (reference tensorframe.py:304)."""
        ref = None
        if to_work_with is not None:
            ref = self[to_work_with] if isinstance(to_work_with, str) else to_work_with
        out = torch.as_tensor(x, dtype=(ref.dtype if ref is not None else None),
                              device=(ref.device if ref is not None else self.device))
        if broadcast_if_scalar and out.ndim == 0 and ref is not None:
            out = out.expand(ref.shape[0])
        return out

    def join(self, other) -> "TensorFrame":
        """pandas-style alias of hstack (reference tensorframe.py:1092)."""
        if isinstance(other, (list, tuple)):
            out = self
            for item in other:
                out = out.hstack(item)
            return out
        return self.hstack(other)

    def drop(self, *, columns) -> "TensorFrame":
        """Drop the named column(s) (reference tensorframe.py:1107)."""
        names = [columns] if isinstance(columns, str) else list(columns)
        return self.without_columns(*names)

    def nlargest(self, n: int, columns) -> "TensorFrame":
        """The n rows with the largest values in the given column
        (reference tensorframe.py:1060)."""
        by = columns if isinstance(columns, str) else list(columns)[0]
        idx = self.argsort(by, descending=True)[: int(n)]
        return self.pick[idx]

    def nsmallest(self, n: int, columns) -> "TensorFrame":
        by = columns if isinstance(columns, str) else list(columns)[0]
        idx = self.argsort(by, descending=False)[: int(n)]
        return self.pick[idx]

    def with_enforced_device(self, device) -> "TensorFrame":
        """Shallow copy whose columns are moved to (and future columns
        coerced onto) `device` (reference tensorframe.py:432)."""
        out = self.to(device)
        out._enforced_device = torch.device(device)
        return out

    def without_enforced_device(self) -> "TensorFrame":
        out = self.to(self.device)
        out._enforced_device = None
        return out

    def get_read_only_view(self) -> "TensorFrame":
        """Read-only view SHARING storage: mutating the original is
        reflected, writes through the view raise TypeError (the columns
        are ReadOnlyTensors)."""
        return TensorFrame(self._columns, read_only=True, _copy=False)

    def clone(self, *, memo: Optional[dict] = None) -> "TensorFrame":
        """Mutable deep copy (cloning a read-only frame yields a writable
        one, reference parity)."""
        return TensorFrame({k: torch.Tensor.as_subclass(v, torch.Tensor).clone() for k, v in self._columns.items()}, _copy=False)

    def to_pandas(self):
        import pandas as pd

        return pd.DataFrame({k: v.detach().cpu().numpy() if v.ndim == 1 else list(v.detach().cpu().numpy()) for k, v in self._columns.items()})

    def to_string(self, *, max_depth: int = 10) -> str:
        cols = ", ".join(f"{k}: {tuple(v.shape)}" for k, v in self._columns.items())
        return f"TensorFrame({len(self)} rows; {cols})"
