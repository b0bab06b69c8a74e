"""amd-smi / rocm-smi Prometheus exporter for MI355X GPU telemetry.

Exposes the ``amd_smi_*`` series the collector consumes as auxiliary
signals (wva_amd/controller/constants.py): GPU utilization, VRAM usage and
power draw per GPU.  There is no NVML/DCGM code path anywhere in this
framework — AMD SMI tooling is the only telemetry source.

Reads, in order of preference:
1. the ``amdsmi`` Python bindings (ROCm >= 6);
2. ``rocm-smi --json`` subprocess output;
3. ``--synthetic`` flag for CPU-only development.
"""

from __future__ import annotations

import argparse
import json
import subprocess
import time
from typing import List, Optional

from prometheus_client import CollectorRegistry, Gauge, start_http_server


class GpuSample:
    def __init__(self, gpu_id: str, utilization_pct: float, vram_used_bytes: float, power_watts: float):
        self.gpu_id = gpu_id
        self.utilization_pct = utilization_pct
        self.vram_used_bytes = vram_used_bytes
        self.power_watts = power_watts


def read_amdsmi() -> Optional[List[GpuSample]]:
    try:
        import amdsmi  # type: ignore
    except ImportError:
        return None
    try:
        amdsmi.amdsmi_init()
        out = []
        for i, handle in enumerate(amdsmi.amdsmi_get_processor_handles()):
            util = amdsmi.amdsmi_get_gpu_activity(handle)["gfx_activity"]
            vram = amdsmi.amdsmi_get_gpu_vram_usage(handle)["vram_used"] * 1024 * 1024
            power = amdsmi.amdsmi_get_power_info(handle)["average_socket_power"]
            out.append(GpuSample(str(i), float(util), float(vram), float(power)))
        amdsmi.amdsmi_shut_down()
        return out
    except Exception:
        return None


def read_rocm_smi() -> Optional[List[GpuSample]]:
    try:
        raw = subprocess.run(
            ["rocm-smi", "--showuse", "--showmemuse", "--showpower", "--json"],
            capture_output=True,
            text=True,
            timeout=10,
        )
        data = json.loads(raw.stdout)
    except Exception:
        return None
    out = []
    for card, fields in data.items():
        if not card.startswith("card"):
            continue
        util = float(fields.get("GPU use (%)", 0) or 0)
        vram_pct = float(fields.get("GPU Memory Allocated (VRAM%)", 0) or 0)
        vram = vram_pct / 100.0 * 288 * 1024**3  # MI355X 288 GB
        power = 0.0
        for key in ("Average Graphics Package Power (W)", "Current Socket Graphics Package Power (W)"):
            if fields.get(key):
                power = float(fields[key])
                break
        out.append(GpuSample(card, util, vram, power))
    return out or None


def read_synthetic() -> List[GpuSample]:
    return [GpuSample("0", 42.0, 128 * 1024**3, 750.0)]


class AmdSmiExporter:
    def __init__(self, registry: Optional[CollectorRegistry] = None, synthetic: bool = False):
        self.registry = registry or CollectorRegistry()
        self.synthetic = synthetic
        labels = ["gpu_id"]
        self.utilization = Gauge(
            "amd_smi_gpu_gfx_activity", "GPU gfx engine activity (%)", labels, registry=self.registry
        )
        self.vram = Gauge(
            "amd_smi_gpu_vram_used_bytes", "GPU VRAM used (bytes)", labels, registry=self.registry
        )
        self.power = Gauge(
            "amd_smi_gpu_power_watts", "GPU socket power (W)", labels, registry=self.registry
        )

    def sample(self) -> List[GpuSample]:
        if self.synthetic:
            return read_synthetic()
        return read_amdsmi() or read_rocm_smi() or []

    def collect_once(self) -> int:
        samples = self.sample()
        for s in samples:
            self.utilization.labels(gpu_id=s.gpu_id).set(s.utilization_pct)
            self.vram.labels(gpu_id=s.gpu_id).set(s.vram_used_bytes)
            self.power.labels(gpu_id=s.gpu_id).set(s.power_watts)
        return len(samples)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=9360)
    ap.add_argument("--interval", type=float, default=5.0)
    ap.add_argument("--synthetic", action="store_true")
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args()

    exporter = AmdSmiExporter(synthetic=args.synthetic)
    if args.once:
        n = exporter.collect_once()
        from prometheus_client import generate_latest

        print(generate_latest(exporter.registry).decode())
        print(f"# sampled {n} GPUs")
        return
    start_http_server(args.port, registry=exporter.registry)
    while True:
        exporter.collect_once()
        time.sleep(args.interval)


if __name__ == "__main__":
    main()
