"""MI355X-first accelerator catalog.

The reference ships NVIDIA-centric demo tables (A100/H100/G2 with MI300X as
an also-ran: /root/reference/docs/tutorials/demo.md:23-42,
test/utils/unitutils.go:71-83).  Here the AMD CDNA4 part is the first-class
default and the heterogeneous alternatives are *emulated* types used for
cost/SLO trade-off solves (BASELINE.json config #4).

MI355X (CDNA4, gfx950) datasheet points used below:
- 288 GB HBM3E per GPU, ~8 TB/s memory bandwidth
- OAM power envelope ~1400 W at full tilt; idle measured via amd-smi
- 8 GPUs per UBB node over xGMI (7 p2p links x ~153 GB/s per GPU)
"""

from __future__ import annotations

import json
from typing import Dict

from .types import AcceleratorSpec, PowerSpec

MI355X_ACCELERATOR = AcceleratorSpec(
    name="MI355X",
    type="AMD-MI355X-288GB",
    multiplicity=1,
    mem_size=288,  # GB HBM3E
    mem_bw=8000,  # GB/s
    power=PowerSpec(idle=140, full=1400, mid_power=900, mid_util=0.6),
    cost=85.0,  # cents/hr (unit-cost table entry; tune per deployment)
)

# Emulated heterogeneous pool for trade-off solves: a small, cheap part with
# much lower bandwidth, and a mid-range previous-gen AMD part.
MI355X_CATALOG: Dict[str, AcceleratorSpec] = {
    "MI355X": MI355X_ACCELERATOR,
    "MI300X": AcceleratorSpec(
        name="MI300X",
        type="AMD-MI300X-192GB",
        multiplicity=1,
        mem_size=192,
        mem_bw=5300,
        power=PowerSpec(idle=130, full=750, mid_power=520, mid_util=0.6),
        cost=65.0,
    ),
    "L40S": AcceleratorSpec(
        name="L40S",
        type="EMU-L40S-48GB",  # emulated comparison part, no vendor code path
        multiplicity=1,
        mem_size=48,
        mem_bw=864,
        power=PowerSpec(idle=30, full=350, mid_power=240, mid_util=0.6),
        cost=23.0,
    ),
}


def mi355x_accelerator_configmap() -> Dict[str, str]:
    """The ``accelerator-unit-costs`` ConfigMap payload, MI355X-first —
    shape-compatible with the reference's (JSON string per accelerator with
    ``device`` and ``cost`` keys), extended with memory/bandwidth fields the
    MI355X collector uses for KV-cache sizing.
    """
    out = {}
    for name, spec in MI355X_CATALOG.items():
        out[name] = json.dumps(
            {
                "device": spec.type,
                "cost": f"{spec.cost:.2f}",
                "memSize": str(spec.mem_size),
                "memBW": str(spec.mem_bw),
            }
        )
    return out
