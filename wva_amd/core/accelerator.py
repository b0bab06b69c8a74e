"""Accelerator: one allocation unit (full or multiple GPU cards).

Parity with /root/reference/pkg/core/accelerator.go including the
piecewise-linear power model with two slopes around the mid-utilization
inflection point.
"""

from __future__ import annotations

from ..config import AcceleratorSpec


class Accelerator:
    def __init__(self, spec: AcceleratorSpec) -> None:
        self.name = spec.name
        self.spec = spec
        self._slope_low = 0.0
        self._slope_high = 0.0

    def calculate(self) -> None:
        p = self.spec.power
        self._slope_low = (p.mid_power - p.idle) / p.mid_util if p.mid_util else 0.0
        self._slope_high = (
            (p.full - p.mid_power) / (1.0 - p.mid_util) if p.mid_util != 1.0 else 0.0
        )

    def power(self, util: float) -> float:
        """Power draw (W) at a given utilization in [0, 1]."""
        p = self.spec.power
        if util <= p.mid_util:
            return p.idle + self._slope_low * util
        return p.mid_power + self._slope_high * (util - p.mid_util)

    @property
    def type(self) -> str:
        return self.spec.type

    @property
    def cost(self) -> float:
        return self.spec.cost

    @property
    def multiplicity(self) -> int:
        return self.spec.multiplicity

    @property
    def mem_size(self) -> int:
        return self.spec.mem_size

    def __repr__(self) -> str:
        s = self.spec
        return (
            f"Accelerator: name={self.name}; type={s.type}; multiplicity={s.multiplicity}; "
            f"memSize={s.mem_size}; memBW={s.mem_bw}; cost={s.cost}"
        )
