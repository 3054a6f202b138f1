"""Distributed (DP over RCCL/xGMI) collective helpers."""

from . import collectives

__all__ = ["collectives"]
