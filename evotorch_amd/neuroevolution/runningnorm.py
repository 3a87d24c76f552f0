"""Observation normalization: RunningNorm (on-device, mergeable).

Reference parity: /root/reference/src/evotorch/neuroevolution/net/
runningnorm.py:47-583 (RunningNorm) and runningstat.py:25-120
(RunningStat — here the same class serves both roles since everything is
torch; `to("cpu")` gives the numpy-era behavior).

The state is the associative triple (count, sum, sum_of_squares), so
merging across ranks is ONE fused RCCL all-reduce
instead of the reference's collect/merge/redistribute actor round-trip.
"""

from typing import Optional

import torch

__all__ = ["RunningNorm", "ObsNormLayer"]


class RunningNorm:
    def __init__(self, *, shape, dtype: torch.dtype = torch.float32, device="cpu", min_variance: float = 1e-2, clip: Optional[tuple] = None):
        self._shape = (int(shape),) if isinstance(shape, int) else tuple(shape)
        self._dtype = dtype
        self._device = torch.device(device)
        self._min_variance = float(min_variance)
        self._clip = clip
        self._count = torch.zeros((), dtype=torch.float64, device=self._device)
        self._sum = torch.zeros(self._shape, dtype=dtype, device=self._device)
        self._sum_sq = torch.zeros(self._shape, dtype=dtype, device=self._device)
        self._has_data = False

    @property
    def shape(self):
        return self._shape

    @property
    def dtype(self):
        return self._dtype

    @property
    def device(self):
        return self._device

    @property
    def min_variance(self) -> float:
        return self._min_variance

    @property
    def low(self):
        """Lower clip bound applied after normalization (or None)."""
        return None if self._clip is None else self._clip[0]

    @property
    def high(self):
        return None if self._clip is None else self._clip[1]

    @property
    def stats(self):
        """(count, sum, sum_of_squares) — reference runningnorm.py: stats."""
        return self.stats_triple()

    @property
    def count(self) -> float:
        return float(self._count)

    @property
    def has_data(self) -> bool:
        """Host-side flag: True once any observation was accumulated (no
        device sync, unlike `count`)."""
        return self._has_data

    @property
    def sum(self) -> torch.Tensor:
        return self._sum

    @property
    def sum_of_squares(self) -> torch.Tensor:
        return self._sum_sq

    @property
    def mean(self) -> torch.Tensor:
        c = torch.clamp(self._count, min=1.0).to(self._dtype)
        return self._sum / c

    @property
    def stdev(self) -> torch.Tensor:
        c = torch.clamp(self._count, min=1.0).to(self._dtype)
        var = self._sum_sq / c - self.mean**2
        return torch.sqrt(torch.clamp(var, min=self._min_variance))

    def update(self, x, mask: Optional[torch.Tensor] = None):
        """Accumulate observations. `x` may be a single observation, a
        2-D batch, another RunningNorm, or a raw (count, sum, sumsq)
        triple. `mask` selects batch rows."""
        if isinstance(x, RunningNorm):
            self._count += x._count.to(self._count.device)
            self._sum += x._sum.to(self._sum.device, self._dtype)
            self._sum_sq += x._sum_sq.to(self._sum_sq.device, self._dtype)
            self._has_data = self._has_data or x._has_data
            return
        if isinstance(x, tuple) and len(x) == 3:
            count, s, ss = x
            if isinstance(count, torch.Tensor):
                self._count += count.to(self._count.device, torch.float64)
            else:
                self._count += float(count)
            self._sum += torch.as_tensor(s, dtype=self._dtype, device=self._device)
            self._sum_sq += torch.as_tensor(ss, dtype=self._dtype, device=self._device)
            self._has_data = True
            return
        x = torch.as_tensor(x, dtype=self._dtype, device=self._device)
        if x.ndim == len(self._shape):
            x = x.unsqueeze(0)
        if mask is not None:
            if x.is_cuda:
                # Arithmetic masking: boolean indexing (x[mask]) calls
                # nonzero() under the hood, which host-syncs on GPU to size
                # the output — fatal in a per-env-step loop. Multiply-by-mask
                # keeps shapes static and the stream free. (If the batch is
                # nonempty but the mask all-False, has_data flips True with
                # zero counts; mean/stdev clamp the count so normalize stays
                # finite.)
                m = mask.to(self._device, self._dtype).reshape(x.shape[0], *([1] * (x.ndim - 1)))
                xm = x * m
                self._count += mask.sum()
                self._sum += xm.sum(dim=0)
                self._sum_sq += (xm * x).sum(dim=0)
                self._has_data = self._has_data or x.shape[0] > 0
                return
            x = x[mask]
        self._count += x.shape[0]
        self._sum += x.sum(dim=0)
        self._sum_sq += (x**2).sum(dim=0)
        self._has_data = self._has_data or x.shape[0] > 0

    def normalize(self, x: torch.Tensor) -> torch.Tensor:
        x = torch.as_tensor(x, dtype=self._dtype, device=self._device)
        if not self._has_data:
            result = x
        else:
            result = (x - self.mean) / self.stdev
        if self._clip is not None:
            lo, hi = self._clip
            result = torch.clamp(result, lo, hi)
        return result

    def update_and_normalize(self, x, mask: Optional[torch.Tensor] = None):
        self.update(x, mask=mask)
        return self.normalize(x)

    def stats_triple(self):
        """The associative (count, sum, sumsq) triple for all-reduce."""
        return self._count.clone(), self._sum.clone(), self._sum_sq.clone()

    def reset(self):
        self._count.zero_()
        self._sum.zero_()
        self._sum_sq.zero_()
        self._has_data = False

    def to(self, device) -> "RunningNorm":
        device = torch.device(device)
        if device == self._device:
            return self
        out = RunningNorm(shape=self._shape, dtype=self._dtype, device=device, min_variance=self._min_variance, clip=self._clip)
        out._count = self._count.to(device)
        out._sum = self._sum.to(device)
        out._sum_sq = self._sum_sq.to(device)
        out._has_data = self._has_data
        return out

    def to_layer(self) -> "ObsNormLayer":
        return ObsNormLayer(mean=self.mean.clone(), stdev=self.stdev.clone(), clip=self._clip)

    def __repr__(self):
        return f"<RunningNorm shape={self._shape} count={self.count}>"


class ObsNormLayer(torch.nn.Module):
    """Frozen normalization layer for exported policies (reference
    net/rl.py:166)."""

    def __init__(self, mean: torch.Tensor, stdev: torch.Tensor, clip: Optional[tuple] = None):
        super().__init__()
        self.register_buffer("mean", mean)
        self.register_buffer("stdev", stdev)
        self._clip = clip

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = (x - self.mean) / self.stdev
        if self._clip is not None:
            y = torch.clamp(y, self._clip[0], self._clip[1])
        return y


class RunningStat:
    """Numpy-side mergeable running mean/stdev (reference
    net/runningstat.py:25,90). `RunningNorm` is the on-device torch
    equivalent; this one serves CPU rollout paths (GymNE) where observations
    arrive as numpy arrays and stats are merged across worker processes."""

    def __init__(self):
        self.reset()

    def reset(self):
        self._count = 0
        self._sum = None
        self._sum_sq = None

    @property
    def count(self) -> int:
        return self._count

    @property
    def sum(self):
        return self._sum

    @property
    def sum_of_squares(self):
        return self._sum_sq

    @property
    def mean(self):
        return self._sum / self._count

    @property
    def stdev(self):
        import numpy as np

        var = self._sum_sq / self._count - self.mean**2
        return np.sqrt(np.clip(var, 1e-8, None))

    def update(self, x):
        """Accept a single observation vector, a 2-D batch, or another
        RunningStat (merge)."""
        import numpy as np

        if isinstance(x, RunningStat):
            if x._count == 0:
                return
            if self._count == 0:
                self._count = x._count
                self._sum = x._sum.copy()
                self._sum_sq = x._sum_sq.copy()
            else:
                self._count += x._count
                self._sum += x._sum
                self._sum_sq += x._sum_sq
            return
        arr = np.asarray(x, dtype=np.float64)
        if arr.ndim == 1:
            arr = arr[None, :]
        if self._count == 0:
            self._sum = arr.sum(axis=0)
            self._sum_sq = (arr**2).sum(axis=0)
        else:
            self._sum += arr.sum(axis=0)
            self._sum_sq += (arr**2).sum(axis=0)
        self._count += arr.shape[0]

    def normalize(self, x):
        import numpy as np

        arr = np.asarray(x, dtype=np.float64)
        return (arr - self.mean) / self.stdev

    def to_running_norm(self, rn: "RunningNorm"):
        """Pour these stats into a torch RunningNorm (device transfer)."""
        import torch as _torch

        if self._count == 0:
            return rn
        rn.update(
            (
                _torch.tensor(float(self._count)),
                _torch.as_tensor(self._sum, dtype=_torch.float32),
                _torch.as_tensor(self._sum_sq, dtype=_torch.float32),
            )
        )
        return rn

    def __repr__(self):
        return f"<RunningStat count={self._count}>"
