"""Assertion helpers for users and for the test-suite.

Reference parity: /root/reference/src/evotorch/testing.py:100-273.
"""

from typing import Iterable, Optional, Union

import numpy as np
import torch

__all__ = [
    "assert_eachclose","TestingError", "assert_allclose", "assert_almost_between", "assert_dtype_matches", "assert_shape_matches"]


class TestingError(AssertionError):
    pass


def _to_numpy(x) -> np.ndarray:
    if isinstance(x, torch.Tensor):
        return torch.Tensor.as_subclass(x, torch.Tensor).detach().cpu().to(torch.float64).numpy()
    return np.asarray(x, dtype=np.float64)


def assert_allclose(actual, desired, *, rtol: Optional[float] = None, atol: Optional[float] = None, equal_nan: bool = True):
    if rtol is None and atol is None:
        raise TestingError("Provide rtol and/or atol")
    a, d = _to_numpy(actual), _to_numpy(desired)
    kwargs = {}
    if rtol is not None:
        kwargs["rtol"] = rtol
    if atol is not None:
        kwargs["atol"] = atol
    if not np.allclose(a, d, equal_nan=equal_nan, **kwargs):
        max_abs = float(np.max(np.abs(a - d))) if a.shape == d.shape else float("nan")
        raise TestingError(f"assert_allclose failed (max abs diff {max_abs}); rtol={rtol} atol={atol}")


def assert_almost_between(x, lb: Union[float, Iterable], ub: Union[float, Iterable], *, atol: Optional[float] = None):
    arr = _to_numpy(x)
    lb_arr = _to_numpy(lb)
    ub_arr = _to_numpy(ub)
    tol = 0.0 if atol is None else float(atol)
    if not (np.all(arr >= lb_arr - tol) and np.all(arr <= ub_arr + tol)):
        raise TestingError(f"Values not (almost) within [{lb}, {ub}] with atol={atol}")


def assert_dtype_matches(x, dtype):
    from .utils.misc import is_dtype_object, to_torch_dtype

    if is_dtype_object(dtype):
        ok = getattr(x, "dtype", None) == object
    else:
        want = to_torch_dtype(dtype)
        got = getattr(x, "dtype", None)
        if isinstance(got, np.dtype):
            ok = np.dtype(str(want).replace("torch.", "")) == got
        else:
            ok = got == want
    if not ok:
        raise TestingError(f"dtype mismatch: expected {dtype}, got {getattr(x, 'dtype', None)}")


def assert_shape_matches(x, shape):
    want = tuple(int(s) if s is not None and s != "*" else None for s in (shape if isinstance(shape, (tuple, list)) else (shape,)))
    got = tuple(x.shape)
    if len(want) != len(got):
        raise TestingError(f"shape mismatch: expected {want}, got {got}")
    for w, g in zip(want, got):
        if w is not None and w != g:
            raise TestingError(f"shape mismatch: expected {want}, got {got}")


def assert_eachclose(x: Iterable, value, *, rtol: Optional[float] = None, atol: Optional[float] = None):
    """Assert that every element of `x` is close to the scalar `value`
    (reference testing.py: assert_eachclose)."""
    arr = np.asarray([float(v) for v in np.asarray(x).reshape(-1)])
    target = np.full_like(arr, float(value))
    kwargs = {}
    if rtol is not None:
        kwargs["rtol"] = rtol
    if atol is not None:
        kwargs["atol"] = atol
    if rtol is None and atol is None:
        kwargs["atol"] = 1e-8
    if not np.allclose(arr, target, **kwargs):
        raise TestingError(f"Elements {arr} are not all close to {value}")
