"""Inference model: per-accelerator perf data and instance counts.

Parity with /root/reference/pkg/core/model.go — ``num_instances`` is the
number of accelerator units needed to fit the model (== accCount; a TP=8
Llama-70B variant on one 8xMI355X node is a profile row with acc_count=8).
"""

from __future__ import annotations

from typing import Dict, Optional

from ..config import ModelAcceleratorPerfData


class Model:
    def __init__(self, name: str) -> None:
        self.name = name
        self.perf_data: Dict[str, ModelAcceleratorPerfData] = {}
        self.num_instances: Dict[str, int] = {}

    def add_perf_data(self, spec: ModelAcceleratorPerfData) -> None:
        if spec.name != self.name:
            return
        self.perf_data[spec.acc] = spec
        self.num_instances[spec.acc] = spec.acc_count if spec.acc_count > 0 else 1

    def remove_perf_data(self, acc_name: str) -> None:
        self.perf_data.pop(acc_name, None)

    def get_perf_data(self, acc_name: str) -> Optional[ModelAcceleratorPerfData]:
        return self.perf_data.get(acc_name)

    def get_num_instances(self, acc_name: str) -> int:
        return self.num_instances.get(acc_name, 0)

    def calculate(self, accelerators) -> None:
        pass

    def __repr__(self) -> str:
        return f"Model: name={self.name}; numInstances={self.num_instances}"
