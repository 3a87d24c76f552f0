"""RCCL-over-xGMI parallel substrate (replaces the reference's Ray layer)."""

from .comm import Comm, get_comm, init_comm
from .evalpool import EvalPool

__all__ = ["Comm", "EvalPool", "get_comm", "init_comm"]
