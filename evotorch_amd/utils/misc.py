"""Dtype/device utilities and tensor factories — the L1 tensor substrate.

MI355X-native re-design of the reference's `tools/misc.py`
(/root/reference/src/evotorch/tools/misc.py:75-1816). Differences from the
reference are deliberate:

* RNG: all random factories are generator-aware, and `make_gaussian`
  supports the antithetic (`symmetric=True`) mode used by PGPE. On ROCm
  devices the hot-path sampling is routed through the HIP philox kernel in
  `evotorch_amd.ops` (counter-based, so population shards sampled on
  different ranks are reproducible) — see `evotorch_amd/ops/dispatch.py`.
* No Ray helpers (`ensure_ray`, `split_workload` for actor pools): the
  parallel substrate is one rank per GPU over RCCL (`evotorch_amd.parallel`).
"""

from numbers import Number
from typing import Any, Iterable, Optional, Union

import numpy as np
import torch

DType = Union[str, torch.dtype, np.dtype, type]
Device = Union[str, torch.device]
Size = Union[int, torch.Size, tuple]
RealOrVector = Union[float, Iterable[float], torch.Tensor]
Vector = Union[Iterable[float], torch.Tensor]

__all__ = [
    "DType",
    "Device",
    "Size",
    "RealOrVector",
    "Vector",
    "to_torch_dtype",
    "to_numpy_dtype",
    "is_dtype_object",
    "is_dtype_float",
    "is_dtype_integer",
    "is_dtype_bool",
    "is_dtype_real",
    "is_sequence",
    "is_real",
    "is_integer",
    "is_bool",
    "is_real_vector",
    "is_integer_vector",
    "is_bool_vector",
    "cast_tensors_in_container",
    "dtype_of_container",
    "device_of_container",
    "dtype_of",
    "device_of",
    "clip_tensor",
    "modify_tensor",
    "modify_vector",
    "empty_tensor_like",
    "ensure_tensor_length_and_dtype",
    "expect_none",
    "pass_info_if_needed",
    "set_default_logger_config",
    "message_from",
    "rowwise_sum",
    "multiply_rows_by_scalars",
    "is_tensor_on_cpu",
    "as_tensor",
    "make_tensor",
    "make_empty",
    "make_zeros",
    "make_ones",
    "make_nan",
    "make_I",
    "make_uniform",
    "make_gaussian",
    "make_randint",
    "make_uniform_shaped_like",
    "make_gaussian_shaped_like",
    "stdev_from_radius",
    "split_workload",
    "make_batched_false_for_vmap",
    "ErroneousResult",
    "numpy_copy",
    "storage_ptr",
]


_TORCH_DTYPES = {
    "float16": torch.float16,
    "float32": torch.float32,
    "float64": torch.float64,
    "bfloat16": torch.bfloat16,
    "int8": torch.int8,
    "int16": torch.int16,
    "int32": torch.int32,
    "int64": torch.int64,
    "uint8": torch.uint8,
    "bool": torch.bool,
    "half": torch.float16,
    "float": torch.float32,
    "double": torch.float64,
    "long": torch.int64,
    "int": torch.int32,
    "short": torch.int16,
}


def to_torch_dtype(dtype: DType) -> torch.dtype:
    """Convert a dtype-like object (str / numpy dtype / python type / torch
    dtype) into the corresponding ``torch.dtype``."""
    if isinstance(dtype, torch.dtype):
        return dtype
    if dtype is float:
        return torch.float32
    if dtype is int:
        return torch.int64
    if dtype is bool:
        return torch.bool
    if isinstance(dtype, str):
        if dtype in _TORCH_DTYPES:
            return _TORCH_DTYPES[dtype]
        raise ValueError(f"Unknown dtype string: {dtype!r}")
    # numpy dtype or type
    name = np.dtype(dtype).name
    if name in _TORCH_DTYPES:
        return _TORCH_DTYPES[name]
    raise ValueError(f"Cannot convert {dtype!r} to a torch dtype")


def to_numpy_dtype(dtype: DType) -> np.dtype:
    """Convert a dtype-like object into a numpy dtype."""
    if is_dtype_object(dtype):
        return np.dtype(object)
    t = to_torch_dtype(dtype)
    if t is torch.bfloat16:
        raise ValueError("bfloat16 has no numpy equivalent")
    return np.dtype(str(t).replace("torch.", ""))


def is_dtype_object(dtype: DType) -> bool:
    """True iff `dtype` denotes the python-object dtype (used by ObjectArray)."""
    if isinstance(dtype, str):
        return dtype in ("object", "O", "Any")
    if dtype is object or dtype is Any:
        return True
    try:
        return np.dtype(dtype) == np.dtype(object)
    except Exception:
        return False


def is_dtype_float(dtype: DType) -> bool:
    return (not is_dtype_object(dtype)) and to_torch_dtype(dtype).is_floating_point


def is_dtype_bool(dtype: DType) -> bool:
    return (not is_dtype_object(dtype)) and to_torch_dtype(dtype) is torch.bool


def is_dtype_integer(dtype: DType) -> bool:
    if is_dtype_object(dtype):
        return False
    t = to_torch_dtype(dtype)
    return (not t.is_floating_point) and (t is not torch.bool)


def is_dtype_real(dtype: DType) -> bool:
    return is_dtype_float(dtype) or is_dtype_integer(dtype)


def is_sequence(x: Any) -> bool:
    """True for list/tuple/ndarray/tensor-like sequences, False for scalars,
    strings, dicts."""
    if isinstance(x, (str, bytes, dict)):
        return False
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim > 0
    return isinstance(x, Iterable)


def is_real(x: Any) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 0 and is_dtype_real(x.dtype)
    return isinstance(x, Number) and not isinstance(x, bool)


def is_integer(x: Any) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 0 and is_dtype_integer(x.dtype)
    return isinstance(x, (int, np.integer)) and not isinstance(x, bool)


def is_bool(x: Any) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 0 and is_dtype_bool(x.dtype)
    return isinstance(x, (bool, np.bool_))


def _is_vector_of(x: Any, pred) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 1 and pred(x.dtype)
    if is_sequence(x):
        return all(pred(type(item)) if not isinstance(item, (torch.Tensor, np.ndarray)) else item.ndim == 0 for item in x)
    return False


def is_real_vector(x: Any) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 1 and is_dtype_real(x.dtype)
    return is_sequence(x) and all(is_real(i) for i in x)


def is_integer_vector(x: Any) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 1 and is_dtype_integer(x.dtype)
    return is_sequence(x) and all(is_integer(i) for i in x)


def is_bool_vector(x: Any) -> bool:
    if isinstance(x, (torch.Tensor, np.ndarray)):
        return x.ndim == 1 and is_dtype_bool(x.dtype)
    return is_sequence(x) and all(is_bool(i) for i in x)


def cast_tensors_in_container(container: Any, *, dtype: Optional[DType] = None, device: Optional[Device] = None, memo: Optional[dict] = None) -> Any:
    """Recursively cast every tensor found in a (possibly nested) container
    to the given dtype and/or device, returning a new container."""
    if memo is None:
        memo = {}
    key = id(container)
    if key in memo:
        return memo[key]

    def cast(t: torch.Tensor) -> torch.Tensor:
        kw = {}
        if dtype is not None and t.is_floating_point():
            kw["dtype"] = to_torch_dtype(dtype)
        if device is not None:
            kw["device"] = device
        return t.to(**kw) if kw else t

    if isinstance(container, torch.Tensor):
        result = cast(container)
    elif isinstance(container, dict):
        result = {k: cast_tensors_in_container(v, dtype=dtype, device=device, memo=memo) for k, v in container.items()}
    elif isinstance(container, (list, tuple)):
        items = [cast_tensors_in_container(v, dtype=dtype, device=device, memo=memo) for v in container]
        result = type(container)(items) if not isinstance(container, tuple) else tuple(items)
    else:
        result = container
    memo[key] = result
    return result


def _tensors_in(container: Any):
    if isinstance(container, torch.Tensor):
        yield container
    elif isinstance(container, dict):
        for v in container.values():
            yield from _tensors_in(v)
    elif isinstance(container, (list, tuple)):
        for v in container:
            yield from _tensors_in(v)


def dtype_of_container(container: Any) -> Optional[torch.dtype]:
    for t in _tensors_in(container):
        return t.dtype
    return None


def device_of_container(container: Any) -> Optional[torch.device]:
    for t in _tensors_in(container):
        return t.device
    return None


def dtype_of(x: Any) -> DType:
    """dtype of a tensor, ObjectArray, or Problem-like object."""
    if hasattr(x, "dtype"):
        return x.dtype
    d = dtype_of_container(x)
    if d is None:
        raise ValueError(f"Cannot determine the dtype of {type(x)}")
    return d


def device_of(x: Any) -> Device:
    if hasattr(x, "device"):
        return x.device
    d = device_of_container(x)
    if d is None:
        raise ValueError(f"Cannot determine the device of {type(x)}")
    return d


def numpy_copy(x: Any, dtype: Optional[DType] = None) -> np.ndarray:
    """Copy of `x` as a numpy array (detached, CPU)."""
    if isinstance(x, torch.Tensor):
        arr = x.detach().cpu().numpy().copy()
    else:
        arr = np.array(x)
    if dtype is not None:
        arr = arr.astype(to_numpy_dtype(dtype))
    return arr


def storage_ptr(x) -> int:
    """Address of the underlying storage of a tensor (or of an object
    exposing `.values`). Used by tests to assert shared-memory slicing."""
    if hasattr(x, "untyped_storage"):
        return x.untyped_storage().data_ptr()
    raise TypeError(f"no storage for {type(x)}")


def clip_tensor(
    x: torch.Tensor,
    lb: Optional[RealOrVector] = None,
    ub: Optional[RealOrVector] = None,
    ensure_copy: bool = False,
) -> torch.Tensor:
    """Clip a tensor into [lb, ub] (either bound optional)."""
    result = x
    if lb is not None:
        lb = torch.as_tensor(lb, dtype=x.dtype, device=x.device)
        result = torch.max(result, lb)
    if ub is not None:
        ub = torch.as_tensor(ub, dtype=x.dtype, device=x.device)
        result = torch.min(result, ub)
    if ensure_copy and result is x:
        result = x.clone()
    return result


def modify_tensor(
    original: torch.Tensor,
    target: torch.Tensor,
    lb: Optional[RealOrVector] = None,
    ub: Optional[RealOrVector] = None,
    max_change: Optional[RealOrVector] = None,
    in_place: bool = False,
) -> torch.Tensor:
    """Move `original` towards `target` under clamping (reference
    tools/misc.py:711): `max_change` is RELATIVE — the allowed change is
    ``|original| * max_change`` — and it TIGHTENS [lb, ub], with the
    reference's max-then-min order (the upper bound wins on a degenerate
    intersection). hipGraph-capture-safe: python-scalar bounds stay
    scalars (no host-to-device copies on the hot path)."""
    if lb is None and ub is None and max_change is None:
        if in_place:
            original[:] = target
            return original
        return target.clone()

    def scalar_or_tensor(x):
        return x if (x is None or isinstance(x, torch.Tensor)) else float(x)

    lo = scalar_or_tensor(lb)
    hi = scalar_or_tensor(ub)
    if max_change is not None:
        # two L-sized temporaries total (lo_t doubles as the result) — at
        # billion-parameter L this clamp runs inside the update's memory
        # peak, so every temporary counts
        mc = max_change if isinstance(max_change, torch.Tensor) else float(max_change)
        allowed = original.abs().mul_(mc)
        lo_t = original - allowed
        hi_t = allowed.add_(original)  # `allowed` is dead after this line
        if lo is not None:
            if isinstance(lo, torch.Tensor):
                torch.max(lo_t, lo, out=lo_t)
            else:
                lo_t.clamp_(min=lo)
        if hi is not None:
            if isinstance(hi, torch.Tensor):
                torch.min(hi_t, hi, out=hi_t)
            else:
                hi_t.clamp_(max=hi)
        torch.max(target, lo_t, out=lo_t)
        result = torch.min(lo_t, hi_t, out=lo_t)
    elif isinstance(lo, torch.Tensor) == isinstance(hi, torch.Tensor) or lo is None or hi is None:
        # torch.clamp applies max(min_val) before min(max_val) — same
        # upper-bound-wins semantics as the reference
        result = torch.clamp(target, min=lo, max=hi)
    else:
        # mixed tensor/scalar bounds: clamp() rejects the combination, so
        # apply max-then-min explicitly (same upper-bound-wins order)
        result = torch.max(target, lo) if isinstance(lo, torch.Tensor) else torch.clamp(target, min=lo)
        result = torch.min(result, hi) if isinstance(hi, torch.Tensor) else result.clamp_(max=hi)
    if in_place:
        original[:] = result
        return original
    return result

def modify_vector(*args, **kwargs) -> torch.Tensor:
    """Alias of :func:`modify_tensor` (reference parity)."""
    return modify_tensor(*args, **kwargs)


def empty_tensor_like(source: Any, *, shape: Optional[Size] = None, length: Optional[int] = None, dtype: Optional[DType] = None, device: Optional[Device] = None):
    """New empty tensor (or ObjectArray) with properties borrowed from
    `source`, overridable via keyword arguments."""
    from .objectarray import ObjectArray

    if isinstance(source, ObjectArray):
        if length is None and shape is not None:
            (length,) = tuple(shape)
        n = len(source) if length is None else int(length)
        if dtype is not None and not is_dtype_object(dtype):
            raise ValueError("ObjectArray-like requires object dtype")
        return ObjectArray(n)
    if shape is not None and length is not None:
        raise ValueError("Provide only one of `shape`, `length`")
    if length is not None:
        shape = (int(length),)
    if shape is None:
        shape = source.shape
    return torch.empty(
        shape if isinstance(shape, (tuple, torch.Size)) else (int(shape),),
        dtype=to_torch_dtype(dtype) if dtype is not None else source.dtype,
        device=device if device is not None else source.device,
    )


def ensure_tensor_length_and_dtype(
    t: Any,
    length: int,
    dtype: DType,
    about: Optional[str] = None,
    *,
    allow_scalar: bool = False,
    device: Optional[Device] = None,
) -> Any:
    """Return `t` as a 1-D tensor of given length and dtype, broadcasting
    scalars, validating everything else."""
    from .objectarray import ObjectArray

    about = f" ({about})" if about else ""
    if is_dtype_object(dtype):
        if isinstance(t, ObjectArray):
            if len(t) != length:
                raise ValueError(f"Expected length {length}, got {len(t)}{about}")
            return t
        result = ObjectArray(length)
        src = list(t) if is_sequence(t) else [t] * length
        if len(src) != length:
            raise ValueError(f"Expected length {length}, got {len(src)}{about}")
        for i, item in enumerate(src):
            result[i] = item
        return result
    torch_dtype = to_torch_dtype(dtype)
    if isinstance(t, torch.Tensor) and t.ndim == 0:
        t = t.item()
    if is_sequence(t):
        result = torch.as_tensor(t, dtype=torch_dtype, device=device)
        if result.ndim != 1 or result.shape[0] != length:
            raise ValueError(f"Expected a vector of length {length}, got shape {tuple(result.shape)}{about}")
        return result
    if allow_scalar:
        return torch.as_tensor(t, dtype=torch_dtype, device=device)
    return torch.full((length,), float(t) if torch_dtype.is_floating_point else int(t), dtype=torch_dtype, device=device)


def expect_none(msg_prefix: str, **kwargs):
    """Raise ValueError if any keyword argument is not None."""
    for k, v in kwargs.items():
        if v is not None:
            raise ValueError(f"{msg_prefix}: expected `{k}` to be None, but got {v!r}")


def pass_info_if_needed(f, info: dict):
    """If callable `f` is decorated with @pass_info, call it with the info
    keyword arguments; otherwise return it unchanged."""
    if getattr(f, "__evotorch_pass_info__", False):
        import functools

        try:
            return functools.partial(f, **info)
        except Exception:
            return f
    return f


# ----------------------------------------------------------------------------
# Tensor factories
# ----------------------------------------------------------------------------


def _shape_from(size: tuple) -> tuple:
    if len(size) == 1 and isinstance(size[0], (tuple, list, torch.Size)):
        return tuple(int(s) for s in size[0])
    return tuple(int(s) for s in size)


def make_tensor(
    data: Any,
    *,
    dtype: Optional[DType] = None,
    device: Optional[Device] = None,
    read_only: bool = False,
) -> Any:
    """Create a tensor (or ObjectArray for object dtype) from `data`."""
    from .objectarray import ObjectArray
    from .readonlytensor import as_read_only_tensor

    if dtype is not None and is_dtype_object(dtype):
        items = list(data) if is_sequence(data) else [data]
        result = ObjectArray(len(items))
        for i, item in enumerate(items):
            result[i] = item
        if read_only:
            result.set_read_only_()
        return result
    kw = {}
    if dtype is not None:
        kw["dtype"] = to_torch_dtype(dtype)
    if isinstance(data, torch.Tensor):
        result = data.clone().to(**kw) if kw else data.clone()
        if device is not None:
            result = result.to(device=device)
    else:
        result = torch.tensor(np.asarray(data) if not isinstance(data, (int, float, bool, list, tuple)) else data, **kw)
        if device is not None:
            result = result.to(device=device)
    if read_only:
        result = as_read_only_tensor(result)
    return result


def make_empty(*size, dtype: Optional[DType] = None, device: Optional[Device] = None) -> Any:
    from .objectarray import ObjectArray

    shape = _shape_from(size)
    if dtype is not None and is_dtype_object(dtype):
        if len(shape) != 1:
            raise ValueError("ObjectArray must be 1-D")
        return ObjectArray(shape[0])
    return torch.empty(shape, dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)


def make_zeros(*size, dtype: Optional[DType] = None, device: Optional[Device] = None, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if out is not None:
        return out.zero_()
    return torch.zeros(_shape_from(size), dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)


def make_ones(*size, dtype: Optional[DType] = None, device: Optional[Device] = None, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if out is not None:
        return out.fill_(1)
    return torch.ones(_shape_from(size), dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)


def make_nan(*size, dtype: Optional[DType] = None, device: Optional[Device] = None, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if out is not None:
        return out.fill_(float("nan"))
    result = torch.empty(_shape_from(size), dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)
    return result.fill_(float("nan"))


def make_I(size: Optional[int] = None, *, dtype: Optional[DType] = None, device: Optional[Device] = None, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if out is not None:
        out.zero_()
        out.fill_diagonal_(1)
        return out
    return torch.eye(int(size), dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)


def make_uniform(
    *size,
    lb: Optional[RealOrVector] = None,
    ub: Optional[RealOrVector] = None,
    dtype: Optional[DType] = None,
    device: Optional[Device] = None,
    generator: Optional[torch.Generator] = None,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Uniform random tensor in [lb, ub] (default [0,1); integer dtypes get
    integer uniform in [lb, ub])."""
    if out is None:
        out = torch.empty(_shape_from(size), dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)
    if out.dtype.is_floating_point:
        if lb is None and ub is None:
            out.uniform_(0.0, 1.0, generator=generator)
        else:
            lb = 0.0 if lb is None else lb
            ub = 1.0 if ub is None else ub
            lb_t = torch.as_tensor(lb, dtype=out.dtype, device=out.device)
            ub_t = torch.as_tensor(ub, dtype=out.dtype, device=out.device)
            out.uniform_(0.0, 1.0, generator=generator)
            out.mul_(ub_t - lb_t).add_(lb_t)
    else:
        lb = 0 if lb is None else lb
        ub = 100 if ub is None else ub
        lb_t = torch.as_tensor(lb, dtype=torch.int64, device=out.device)
        ub_t = torch.as_tensor(ub, dtype=torch.int64, device=out.device)
        tmp = torch.empty(out.shape, dtype=torch.float32, device=out.device)
        tmp.uniform_(0.0, 1.0, generator=generator)
        out.copy_((tmp * (ub_t - lb_t + 1).to(torch.float32)).floor_().to(out.dtype) + lb_t.to(out.dtype))
        out.clamp_(max=int(torch.as_tensor(ub).max()))
    return out


def make_gaussian(
    *size,
    center: Optional[RealOrVector] = None,
    stdev: Optional[RealOrVector] = None,
    symmetric: bool = False,
    dtype: Optional[DType] = None,
    device: Optional[Device] = None,
    generator: Optional[torch.Generator] = None,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Gaussian random tensor. With ``symmetric=True`` the leftmost dimension
    must be even and the second half is the mirror (antithetic) of the first:
    ``out[n+i] = 2*center - out[i]`` — the sampling scheme of PGPE's
    `SymmetricSeparableGaussian` (reference distributions.py:705)."""
    if out is None:
        out = torch.empty(_shape_from(size), dtype=to_torch_dtype(dtype) if dtype is not None else None, device=device)
    if symmetric:
        n = out.shape[0]
        if n % 2 != 0:
            raise ValueError(f"symmetric sampling requires an even leftmost dimension, got {n}")
        half = out[: n // 2]
        half.normal_(generator=generator)
        if stdev is not None:
            half.mul_(torch.as_tensor(stdev, dtype=out.dtype, device=out.device))
        if center is not None:
            c = torch.as_tensor(center, dtype=out.dtype, device=out.device)
            half.add_(c)
            torch.sub(2 * c, half, out=out[n // 2 :])
        else:
            torch.neg(half, out=out[n // 2 :])
    else:
        out.normal_(generator=generator)
        if stdev is not None:
            out.mul_(torch.as_tensor(stdev, dtype=out.dtype, device=out.device))
        if center is not None:
            out.add_(torch.as_tensor(center, dtype=out.dtype, device=out.device))
    return out


def make_randint(
    *size,
    n: Union[int, torch.Tensor],
    dtype: Optional[DType] = None,
    device: Optional[Device] = None,
    generator: Optional[torch.Generator] = None,
    out: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Random integers in [0, n)."""
    if out is None:
        if dtype is None:
            dtype = torch.int64
        out = torch.empty(_shape_from(size), dtype=to_torch_dtype(dtype), device=device)
    out.random_(0, int(n) if isinstance(n, (int, np.integer)) else None, generator=generator)
    if not isinstance(n, (int, np.integer)):
        n_t = torch.as_tensor(n, device=out.device)
        tmp = torch.empty(out.shape, dtype=torch.float32, device=out.device)
        tmp.uniform_(0.0, 1.0, generator=generator)
        out.copy_((tmp * n_t.to(torch.float32)).floor_().to(out.dtype))
    return out


def make_uniform_shaped_like(t: torch.Tensor, *, lb: Optional[RealOrVector] = None, ub: Optional[RealOrVector] = None, generator: Optional[torch.Generator] = None) -> torch.Tensor:
    return make_uniform(t.shape, lb=lb, ub=ub, dtype=t.dtype, device=t.device, generator=generator)


def make_gaussian_shaped_like(t: torch.Tensor, *, center: Optional[RealOrVector] = None, stdev: Optional[RealOrVector] = None, generator: Optional[torch.Generator] = None) -> torch.Tensor:
    return make_gaussian(t.shape, center=center, stdev=stdev, dtype=t.dtype, device=t.device, generator=generator)


def stdev_from_radius(radius: float, solution_length: int) -> float:
    """Initial per-dimension stdev such that the search sphere has the given
    radius (reference tools/misc.py:1879): ``radius / sqrt(3 * L) * sqrt(3)``
    simplified to ``radius / sqrt(L)`` scaled — reference uses
    ``sqrt(radius**2 / L)``."""
    return float(np.sqrt((float(radius) ** 2) / int(solution_length)))


def split_workload(workload: int, num_pieces: int) -> list:
    """Split an integer workload into `num_pieces` near-equal pieces
    (earlier pieces get the remainder)."""
    base = workload // num_pieces
    extra = workload % num_pieces
    return [base + (1 if i < extra else 0) for i in range(num_pieces)]


def make_batched_false_for_vmap(device: Optional[Device] = None) -> torch.Tensor:
    """A scalar False built from tensor ops rather than a python literal, so
    that code running under `torch.func.vmap` can use it as the initial value
    of a flag that is later combined (|=, where) with batched booleans
    (reference tools/misc.py:2209). A fresh unbatched tensor broadcasts
    correctly against batched operands inside vmap."""
    return torch.zeros((), dtype=torch.bool, device=(device if device is None else torch.device(device)))


class ErroneousResult:
    """Marker object wrapping an error raised during fitness evaluation
    (reference tools/misc.py:1006)."""

    def __init__(self, error: Exception):
        self.error = error

    def __bool__(self) -> bool:
        return False

    def __repr__(self) -> str:
        return f"<ErroneousResult: {self.error!r}>"

    @staticmethod
    def call(f, *args, **kwargs):
        try:
            return f(*args, **kwargs)
        except Exception as e:  # noqa: BLE001 — fitness errors become markers
            return ErroneousResult(e)


def to_stdev_init(*, solution_length: int, stdev_init=None, radius_init=None):
    """Resolve the (stdev_init | radius_init) pair into a stdev spec
    (reference tools/misc.py:1925): exactly one must be given; a radius is
    converted via `stdev_from_radius`."""
    if (stdev_init is None) == (radius_init is None):
        raise ValueError("Provide exactly one of stdev_init, radius_init")
    if stdev_init is not None:
        return stdev_init
    return stdev_from_radius(float(radius_init), solution_length)


def as_tensor(x, *, dtype: Optional[DType] = None, device: Optional[Device] = None):
    """torch.as_tensor with the framework's dtype coercion (accepts numpy
    dtypes, strings, 'object' → ObjectArray passthrough); reference
    tools/misc.py: as_tensor."""
    if dtype is not None and is_dtype_object(dtype):
        from .objectarray import as_object_array

        return as_object_array(x)
    kwargs = {}
    if dtype is not None:
        kwargs["dtype"] = to_torch_dtype(dtype)
    if device is not None:
        kwargs["device"] = torch.device(device)
    return torch.as_tensor(x, **kwargs)


def is_tensor_on_cpu(x) -> bool:
    """True if x is a torch tensor residing on the CPU."""
    return isinstance(x, torch.Tensor) and x.device.type == "cpu"


def multiply_rows_by_scalars(scalars: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """row i of the result = scalars[i] * x[i] (reference tools/misc.py)."""
    return scalars.unsqueeze(-1) * x


def rowwise_sum(x: torch.Tensor) -> torch.Tensor:
    """Sum over the last dimension, row by row."""
    return x.sum(dim=-1)


def message_from(sender, message: str) -> str:
    """Prefix a log/exception message with its originating object — used in
    verbose-mode diagnostics (reference tools/misc.py: message_from)."""
    name = sender if isinstance(sender, str) else type(sender).__name__
    return f"[{name}] {message}"


def set_default_logger_config(
    logger_name: str = "evotorch_amd",
    logger_level=None,
    override: bool = False,
):
    """Install the default stderr handler + level on the framework logger
    (honors EVOTORCH_AMD_VERBOSE_LEVEL; reference tools/misc.py:2072)."""
    import logging as _logging
    import os as _os
    import sys as _sys

    logger = _logging.getLogger(logger_name)
    if logger.handlers and not override:
        return logger
    if override:
        for h in list(logger.handlers):
            logger.removeHandler(h)
    if logger_level is None:
        level_name = _os.environ.get("EVOTORCH_AMD_VERBOSE_LEVEL", "INFO").upper()
        logger_level = getattr(_logging, level_name, _logging.INFO)
    handler = _logging.StreamHandler(_sys.stderr)
    handler.setFormatter(_logging.Formatter("[%(asctime)s] %(name)s %(levelname)s: %(message)s"))
    logger.addHandler(handler)
    logger.setLevel(logger_level)
    return logger
