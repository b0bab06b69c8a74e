"""Allocation policy under saturated (over-capacity) conditions.

Parity with /root/reference/pkg/config/config.go.
"""

from __future__ import annotations

import enum


class SaturationPolicy(enum.Enum):
    NONE = "None"  # no allocation beyond satisfying SLOs
    PRIORITY_EXHAUSTIVE = "PriorityExhaustive"  # exhaustively, in priority order
    PRIORITY_ROUND_ROBIN = "PriorityRoundRobin"  # round-robin within priority groups
    ROUND_ROBIN = "RoundRobin"  # round-robin across all servers

    @classmethod
    def parse(cls, s: str) -> "SaturationPolicy":
        for p in cls:
            if p.value == s:
                return p
        return DEFAULT_SATURATION_POLICY

    def __str__(self) -> str:
        return self.value


DEFAULT_SATURATION_POLICY = SaturationPolicy.NONE
