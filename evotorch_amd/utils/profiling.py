"""Profiling ranges: named phases (sample/eval/rank/grad/update) emitted as
roctx markers so rocprofv3 / omnitrace group kernel time by framework
phase. Green-field addition relative to the reference (SURVEY.md §5.1 —
the reference has no tracing surface).

Disabled (zero overhead beyond one env check at import) unless
EVOTORCH_AMD_PROFILE=1. torch.cuda.nvtx maps onto roctx on ROCm builds.
"""

import os
from contextlib import contextmanager

import torch

__all__ = ["profiling_enabled", "record_range"]

_ENABLED = os.environ.get("EVOTORCH_AMD_PROFILE", "0") == "1"


def profiling_enabled() -> bool:
    return _ENABLED


if _ENABLED and torch.cuda.is_available():

    @contextmanager
    def record_range(name: str):
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()

else:

    @contextmanager
    def record_range(name: str):
        yield
