"""TensorMakerMixin: gives Problem / Distribution / SearchAlgorithm
`make_*` helper methods bound to the object's own dtype/device/generator.

Re-design of the reference's `tools/tensormaker.py`
(/root/reference/src/evotorch/tools/tensormaker.py:27).
"""

from typing import Any, Optional

import torch

from . import misc

__all__ = ["TensorMakerMixin"]


class TensorMakerMixin:
    def _make_kwargs(self, dtype=None, device=None, use_eval_dtype: bool = False) -> dict:
        if dtype is None:
            if use_eval_dtype and hasattr(self, "eval_dtype"):
                dtype = self.eval_dtype
            else:
                dtype = getattr(self, "dtype", None)
        if device is None:
            device = getattr(self, "device", None)
        return {"dtype": dtype, "device": device}

    @property
    def _maker_generator(self) -> Optional[torch.Generator]:
        return getattr(self, "generator", None)

    def make_tensor(self, data: Any, *, dtype=None, device=None, use_eval_dtype: bool = False, read_only: bool = False):
        kw = self._make_kwargs(dtype, device, use_eval_dtype)
        return misc.make_tensor(data, read_only=read_only, **kw)

    def make_empty(self, *size, dtype=None, device=None, use_eval_dtype: bool = False):
        return misc.make_empty(*size, **self._make_kwargs(dtype, device, use_eval_dtype))

    def make_zeros(self, *size, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        if out is not None:
            return misc.make_zeros(out=out)
        return misc.make_zeros(*size, **self._make_kwargs(dtype, device, use_eval_dtype))

    def make_ones(self, *size, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        if out is not None:
            return misc.make_ones(out=out)
        return misc.make_ones(*size, **self._make_kwargs(dtype, device, use_eval_dtype))

    def make_nan(self, *size, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        if out is not None:
            return misc.make_nan(out=out)
        return misc.make_nan(*size, **self._make_kwargs(dtype, device, use_eval_dtype))

    def make_I(self, size=None, *, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        if out is not None:
            return misc.make_I(out=out)
        return misc.make_I(size, **self._make_kwargs(dtype, device, use_eval_dtype))

    def make_uniform(self, *size, lb=None, ub=None, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        return misc.make_uniform(*size, lb=lb, ub=ub, generator=self._maker_generator, out=out, **({} if out is not None else self._make_kwargs(dtype, device, use_eval_dtype)))

    def make_gaussian(self, *size, center=None, stdev=None, symmetric: bool = False, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        return misc.make_gaussian(
            *size, center=center, stdev=stdev, symmetric=symmetric, generator=self._maker_generator, out=out, **({} if out is not None else self._make_kwargs(dtype, device, use_eval_dtype))
        )

    def make_randint(self, *size, n, dtype=None, device=None, use_eval_dtype: bool = False, out=None):
        kw = {} if out is not None else self._make_kwargs(dtype, device, use_eval_dtype)
        if out is None and (kw.get("dtype") is not None and misc.is_dtype_float(kw["dtype"])):
            kw["dtype"] = torch.int64
        return misc.make_randint(*size, n=n, generator=self._maker_generator, out=out, **kw)

    def make_uniform_shaped_like(self, t: torch.Tensor, *, lb=None, ub=None):
        return misc.make_uniform_shaped_like(t, lb=lb, ub=ub, generator=self._maker_generator)

    def make_gaussian_shaped_like(self, t: torch.Tensor, *, center=None, stdev=None):
        return misc.make_gaussian_shaped_like(t, center=center, stdev=stdev, generator=self._maker_generator)
