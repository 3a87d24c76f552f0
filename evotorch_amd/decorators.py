"""Decorators: @pass_info, @on_device/@on_cuda/@on_aux_device, @vectorized,
@expects_ndim, @rowwise.

Reference parity: /root/reference/src/evotorch/decorators.py:170-965.
`expects_ndim` is the backbone of the functional API: each positional
argument declares its expected ndim, and any extra leftmost dimensions are
mapped over with `torch.func.vmap`, so a whole searcher or operator written
for one population runs as B independent batched searches.
"""

import functools
from typing import Callable, Iterable, Optional, Union

import torch

__all__ = ["pass_info", "on_device", "on_cuda", "on_aux_device", "vectorized", "expects_ndim", "rowwise"]


def pass_info(fn_or_class: Callable) -> Callable:
    """Mark a function/class as wanting problem info keyword arguments
    (obs_length, act_length, obs_space, ...) injected at call time."""
    fn_or_class.__evotorch_pass_info__ = True
    return fn_or_class


def on_device(device) -> Callable:
    """Declare the device on which a fitness function wants its batches."""

    def decorator(fn: Callable) -> Callable:
        fn.__evotorch_device__ = str(device)
        return fn

    return decorator


def on_cuda(fn_or_index: Union[Callable, int, None] = None):
    """@on_cuda or @on_cuda(i): shorthand for @on_device('cuda[:i]')."""
    if callable(fn_or_index):
        fn_or_index.__evotorch_device__ = "cuda"
        return fn_or_index
    index = fn_or_index

    def decorator(fn: Callable) -> Callable:
        fn.__evotorch_device__ = "cuda" if index is None else f"cuda:{int(index)}"
        return fn

    return decorator


def on_aux_device(fn: Callable) -> Callable:
    """Declare that a fitness function wants batches on the Problem's
    aux_device (the first visible accelerator)."""
    fn.__evotorch_on_aux_device__ = True
    return fn


def vectorized(fn: Callable) -> Callable:
    """Declare that a fitness function takes the whole 2-D batch at once."""
    fn.__evotorch_vectorized__ = True
    return fn


def _call_with_ndims(fn: Callable, ndims: tuple, args: tuple, randomness: str):
    prepared = []
    extras = []
    for a, nd in zip(args, ndims):
        if nd is None:
            prepared.append(a)
            extras.append(0)
        else:
            t = torch.as_tensor(a)
            if t.ndim < nd:
                raise ValueError(f"Argument with expected ndim {nd} has only ndim {t.ndim}")
            prepared.append(t)
            extras.append(t.ndim - nd)
    max_extra = max(extras) if extras else 0
    if max_extra == 0:
        return fn(*prepared)
    in_dims = tuple(0 if (nd is not None and e == max_extra) else None for nd, e in zip(ndims, extras))

    def inner(*inner_args):
        return _call_with_ndims(fn, ndims, inner_args, randomness)

    return torch.func.vmap(inner, in_dims=in_dims, randomness=randomness)(*prepared)


def expects_ndim(
    fn: Optional[Callable] = None,
    expected_ndims: Optional[Iterable[Optional[int]]] = None,
    *,
    allow_smaller_ndim: bool = False,
    randomness: str = "error",
) -> Callable:
    """Wrap `fn` so that each positional argument is validated against its
    expected ndim and extra leftmost dimensions are vmapped over.

    Usable both as `expects_ndim(fn, (None, 1, 1))` and as a decorator
    `@expects_ndim(1, 1)`.
    """
    # decorator-style: expects_ndim(1, 1, ...) or expects_ndim((1,1))
    if fn is not None and not callable(fn):
        if expected_ndims is None:
            expected_ndims = (fn,)
        else:
            expected_ndims = (fn, expected_ndims)
        fn = None
    if fn is None:
        ndims_outer = expected_ndims

        def decorator(inner_fn: Callable, _nd=ndims_outer) -> Callable:
            return expects_ndim(inner_fn, _nd, allow_smaller_ndim=allow_smaller_ndim, randomness=randomness)

        # Support @expects_ndim(1, None, 2) with multiple scalar args
        def flexible_decorator(*args, **kwargs):
            if len(args) == 1 and callable(args[0]) and not kwargs:
                return decorator(args[0])
            raise TypeError("expects_ndim decorator takes exactly the function")

        return flexible_decorator

    ndims = tuple(expected_ndims) if expected_ndims is not None else ()

    @functools.wraps(fn)
    def wrapped(*args):
        if len(args) != len(ndims):
            raise TypeError(f"{fn.__name__} expects {len(ndims)} positional arguments, got {len(args)}")
        return _call_with_ndims(fn, ndims, args, randomness)

    wrapped.__expects_ndim__ = ndims
    return wrapped


def _expects_ndim_varargs(fn: Callable, per_arg_ndim: int, randomness: str) -> Callable:
    @functools.wraps(fn)
    def wrapped(*args):
        ndims = tuple(per_arg_ndim if isinstance(a, (torch.Tensor,)) or _is_numeric(a) else None for a in args)
        return _call_with_ndims(fn, ndims, args, randomness)

    return wrapped


def _is_numeric(a) -> bool:
    import numpy as np

    return isinstance(a, (int, float, np.ndarray, list, tuple))


def rowwise(fn: Optional[Callable] = None, *, randomness: str = "error") -> Callable:
    """Write per-row (1-D) logic; calls with 2-D/3-D/... inputs are
    auto-vmapped over all leading dimensions (reference decorators.py:877).
    All tensor arguments are treated as rows."""

    def decorator(inner_fn: Callable) -> Callable:
        wrapped = _expects_ndim_varargs(inner_fn, 1, randomness)
        wrapped.__evotorch_rowwise__ = True
        return wrapped

    if fn is not None:
        return decorator(fn)
    return decorator
