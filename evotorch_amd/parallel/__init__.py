"""RCCL-over-xGMI parallel substrate (replaces the reference's Ray layer)."""

from .comm import Comm, get_comm, init_comm

__all__ = ["Comm", "get_comm", "init_comm"]
