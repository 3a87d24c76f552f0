"""Pretty-printing mixin (reference tools/recursiveprintable.py:24)."""

from collections.abc import Iterable, Mapping

__all__ = ["RecursivePrintable"]

_MAX_DEPTH = 10


class RecursivePrintable:
    def to_string(self, *, max_depth: int = _MAX_DEPTH) -> str:
        if max_depth <= 0:
            return "<...>"

        def fmt(x, depth):
            if isinstance(x, RecursivePrintable):
                return x.to_string(max_depth=depth - 1)
            return repr(x)

        name = type(self).__name__
        if isinstance(self, Mapping):
            items = ", ".join(f"{k!r}: {fmt(v, max_depth)}" for k, v in self.items())
            return f"{name}({{{items}}})"
        if isinstance(self, Iterable):
            try:
                items = ", ".join(fmt(v, max_depth) for v in self)
                return f"{name}([{items}])"
            except Exception:
                pass
        return f"{name}(...)"

    def __str__(self) -> str:
        return self.to_string()

    def __repr__(self) -> str:
        return self.to_string()
