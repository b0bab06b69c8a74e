"""wva_amd — MI355X-native workload-variant autoscaler.

A brand-new, built-from-scratch GPU-aware autoscaler for LLM inference
workloads, with the capabilities of llm-d-incubation/workload-variant-autoscaler
(the Go/Kubernetes reference, studied at /root/reference) re-designed for AMD
MI355X (CDNA4) deployments:

- ``wva_amd.analyzer``   — state-dependent M/M/1/K queueing analytics (L5)
- ``wva_amd.config``     — declarative system spec types (L4)
- ``wva_amd.core``       — domain model: System/Accelerator/Model/ServiceClass/
                           Server/Allocation with the sizing kernel (L4)
- ``wva_amd.solver``     — global min-cost assignment: unlimited + greedy
                           limited modes with saturation policies (L4)
- ``wva_amd.api``        — VariantAutoscaling CRD types + conditions (L1)
- ``wva_amd.kube``       — Kubernetes client protocol + in-memory fake (L0/L2)
- ``wva_amd.controller`` — reconciler, Prometheus collector, model analyzer,
                           optimizer engine, actuator, metrics emitter (L2/L3)
- ``wva_amd.promlib``    — offline mini-Prometheus (scraper + PromQL subset)
                           used by e2e tests and the bundled emulator stack
- ``wva_amd.ops``        — native batched queue-solver: HIP/gfx950 kernel and
                           C++(OpenMP) CPU path for the hot analytic kernel

No NVIDIA/NVML/DCGM assumptions anywhere; accelerator tables, perf profiles,
samples and the bundled vLLM emulator are CDNA4-first (MI355X: 288 GB HBM3E,
~8 TB/s, 1.4 kW OAM envelope).
"""

__version__ = "0.2.0"  # round 2
