"""Decorators: @pass_info, @on_device/@on_cuda/@on_aux_device, @vectorized,
@expects_ndim, @rowwise.

Reference parity: /root/reference/src/evotorch/decorators.py:170-965.
`expects_ndim` is the backbone of the functional API: each positional
argument declares its expected ndim, and any extra leftmost dimensions are
mapped over with `torch.func.vmap`, so a whole searcher or operator written
for one population runs as B independent batched searches.
"""

import functools
from typing import Callable, Optional, Union

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


def expects_ndim(*spec, allow_smaller_ndim: bool = False, randomness: str = "error") -> Callable:
    """Wrap a function so that each positional argument is validated
    against its expected ndim and extra leftmost dimensions are vmapped
    over (reference decorators.py:613).

    Usable in all of the reference's forms:
      * `expects_ndim(fn, (None, 1, 1))`           — direct wrap
      * `@expects_ndim(1, 1)` / `@expects_ndim(0, (1,), 2)` — decorator
        with per-argument ndims as positional arguments (tuples are
        flattened to their single element when length-1; `None` = leave
        the argument untouched)
    """

    def norm(nd):
        if isinstance(nd, (tuple, list)):
            if len(nd) != 1:
                raise ValueError(f"Cannot interpret the ndim spec {nd!r}")
            return int(nd[0])
        return None if nd is None else int(nd)

    if len(spec) >= 1 and callable(spec[0]):
        fn = spec[0]
        if len(spec) == 2 and isinstance(spec[1], (tuple, list)):
            ndims = tuple(None if n is None else int(n) for n in spec[1])
        else:
            ndims = tuple(norm(n) for n in spec[1:])

        @functools.wraps(fn)
        def wrapped(*args):
            if len(args) != len(ndims):
                raise TypeError(f"{fn.__name__} expects {len(ndims)} positional arguments, got {len(args)}")
            return _call_with_ndims(fn, ndims, args, randomness)

        wrapped.__expects_ndim__ = ndims
        return wrapped

    # decorator form: the positional args are the ndim spec
    ndim_spec = spec[0] if (len(spec) == 1 and isinstance(spec[0], (tuple, list)) and len(spec[0]) != 1) else spec

    def decorator(inner_fn: Callable) -> Callable:
        return expects_ndim(inner_fn, tuple(ndim_spec), allow_smaller_ndim=allow_smaller_ndim, randomness=randomness)

    return decorator


def _expects_ndim_varargs(fn: Callable, per_arg_ndim: int, randomness: str) -> Callable:
    @functools.wraps(fn)
    def wrapped(*args):
        # tensors/arrays are treated as rows to map over; python scalars
        # pass through untouched
        ndims = tuple(per_arg_ndim if _is_rowlike(a) else None for a in args)
        return _call_with_ndims(fn, ndims, args, randomness)

    return wrapped


def _is_rowlike(a) -> bool:
    import numpy as np

    if isinstance(a, torch.Tensor):
        return True
    if isinstance(a, np.ndarray):
        return a.ndim >= 1
    return isinstance(a, (list, tuple))


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
