"""Hook: an ordered collection of callbacks whose dict/list returns are
accumulated. Re-design of the reference's `tools/hook.py`
(/root/reference/src/evotorch/tools/hook.py:25-130)."""

from typing import Any, Callable, Iterable, Optional

__all__ = ["Hook"]


class Hook:
    def __init__(
        self,
        callbacks: Optional[Iterable[Callable]] = None,
        *,
        args: Optional[Iterable] = None,
        kwargs: Optional[dict] = None,
    ):
        self._funcs = list(callbacks) if callbacks is not None else []
        self._args = list(args) if args is not None else []
        self._kwargs = dict(kwargs) if kwargs is not None else {}

    @property
    def args(self) -> list:
        return self._args

    @property
    def kwargs(self) -> dict:
        return self._kwargs

    def __call__(self, *args, **kwargs) -> Optional[Any]:
        """Call every registered callback. If any returns a dict, the dicts
        are merged and returned; if any returns a list, the lists are
        concatenated and returned."""
        all_args = list(args) + self._args
        all_kwargs = {**kwargs, **self._kwargs}
        result_dict = None
        result_list = None
        for f in self._funcs:
            out = f(*all_args, **all_kwargs)
            if out is None:
                continue
            if isinstance(out, dict):
                if result_dict is None:
                    result_dict = {}
                result_dict.update(out)
            elif isinstance(out, (list, tuple)):
                if result_list is None:
                    result_list = []
                result_list.extend(out)
            else:
                raise TypeError(
                    f"Hook callback {f} returned {type(out)}; expected None, dict, or list"
                )
        if result_dict is not None and result_list is not None:
            raise TypeError("Hook callbacks returned a mix of dicts and lists")
        return result_dict if result_dict is not None else result_list

    def accumulate_dict(self, *args, **kwargs) -> dict:
        out = self(*args, **kwargs)
        if out is None:
            return {}
        if not isinstance(out, dict):
            raise TypeError(f"Expected dict accumulation, got {type(out)}")
        return out

    def accumulate_sequence(self, *args, **kwargs) -> list:
        out = self(*args, **kwargs)
        if out is None:
            return []
        if not isinstance(out, list):
            raise TypeError(f"Expected list accumulation, got {type(out)}")
        return out

    # list-like interface
    def append(self, f: Callable):
        self._funcs.append(f)

    def remove(self, f: Callable):
        self._funcs.remove(f)

    def insert(self, i: int, f: Callable):
        self._funcs.insert(i, f)

    def clear(self):
        self._funcs.clear()

    def __len__(self) -> int:
        return len(self._funcs)

    def __iter__(self):
        return iter(self._funcs)

    def __getitem__(self, i):
        return self._funcs[i]

    def __repr__(self) -> str:
        return f"Hook({self._funcs!r})"
