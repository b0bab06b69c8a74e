"""Service class: named priority with per-model SLO targets.

Parity with /root/reference/pkg/core/serviceclass.go (priority clamped to
[1,100] with 100 the default / lowest).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

from ..config import (
    DEFAULT_HIGH_PRIORITY,
    DEFAULT_LOW_PRIORITY,
    DEFAULT_SERVICE_CLASS_PRIORITY,
    ModelTarget,
    ServiceClassSpec,
)


@dataclass
class Target:
    itl: float = 0.0  # ms
    ttft: float = 0.0  # ms (queueing + prefill)
    tps: float = 0.0  # tokens/s


class ServiceClass:
    def __init__(self, name: str, priority: int) -> None:
        if priority < DEFAULT_HIGH_PRIORITY or priority > DEFAULT_LOW_PRIORITY:
            priority = DEFAULT_SERVICE_CLASS_PRIORITY
        self.name = name
        self.priority = priority
        self.targets: Dict[str, Target] = {}

    @classmethod
    def from_spec(cls, spec: ServiceClassSpec) -> "ServiceClass":
        svc = cls(spec.name, spec.priority)
        for mt in spec.model_targets:
            svc.add_model_target(mt)
        return svc

    def model_target(self, model_name: str) -> Optional[Target]:
        return self.targets.get(model_name)

    def add_model_target(self, spec: ModelTarget) -> Target:
        t = Target(itl=spec.slo_itl, ttft=spec.slo_ttft, tps=spec.slo_tps)
        self.targets[spec.model] = t
        return t

    def update_model_targets(self, spec: ServiceClassSpec) -> bool:
        if spec.name != self.name or spec.priority != self.priority:
            return False
        for mt in spec.model_targets:
            self.add_model_target(mt)
        return True

    def remove_model_target(self, model_name: str) -> None:
        self.targets.pop(model_name, None)

    def spec(self) -> ServiceClassSpec:
        return ServiceClassSpec(
            name=self.name,
            priority=self.priority,
            model_targets=[
                ModelTarget(model=m, slo_itl=t.itl, slo_ttft=t.ttft, slo_tps=t.tps)
                for m, t in self.targets.items()
            ],
        )

    def __repr__(self) -> str:
        return f"ServiceClass: name={self.name}; priority={self.priority}; targets={self.targets}"
