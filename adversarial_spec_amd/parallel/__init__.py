"""Distributed layer: RCCL-over-xGMI collectives for the debate."""

from .consensus import AsyncRoundGather, RankResult, gather_round

__all__ = ["AsyncRoundGather", "RankResult", "gather_round"]
