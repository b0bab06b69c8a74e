"""Centralized metric and label names.

Parity with /root/reference/internal/constants/metrics.go — the vLLM input
metric names are identical because vLLM on ROCm emits the same series; the
additional ``amd_smi_*`` names cover the MI355X GPU-counter exporter
(utilization, VRAM, power) that replaces any NVML/DCGM path.
"""

# vLLM input metrics (scraped from Prometheus)
VLLM_REQUEST_SUCCESS_TOTAL = "vllm:request_success_total"
VLLM_REQUEST_PROMPT_TOKENS_SUM = "vllm:request_prompt_tokens_sum"
VLLM_REQUEST_PROMPT_TOKENS_COUNT = "vllm:request_prompt_tokens_count"
VLLM_REQUEST_GENERATION_TOKENS_SUM = "vllm:request_generation_tokens_sum"
VLLM_REQUEST_GENERATION_TOKENS_COUNT = "vllm:request_generation_tokens_count"
VLLM_TIME_TO_FIRST_TOKEN_SECONDS_SUM = "vllm:time_to_first_token_seconds_sum"
VLLM_TIME_TO_FIRST_TOKEN_SECONDS_COUNT = "vllm:time_to_first_token_seconds_count"
VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_SUM = "vllm:time_per_output_token_seconds_sum"
VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_COUNT = "vllm:time_per_output_token_seconds_count"

# amd-smi / rocm-smi exporter auxiliary input metrics (MI355X GPU counters)
AMD_SMI_GPU_UTILIZATION = "amd_smi_gpu_gfx_activity"  # percent
AMD_SMI_GPU_VRAM_USED = "amd_smi_gpu_vram_used_bytes"
AMD_SMI_GPU_POWER = "amd_smi_gpu_power_watts"

# Output metrics (emitted for HPA/KEDA) — inferno_* names kept byte-identical
# to the reference so existing HPA/KEDA integrations are drop-in.
INFERNO_REPLICA_SCALING_TOTAL = "inferno_replica_scaling_total"
INFERNO_DESIRED_REPLICAS = "inferno_desired_replicas"
INFERNO_CURRENT_REPLICAS = "inferno_current_replicas"
INFERNO_DESIRED_RATIO = "inferno_desired_ratio"

# First-class solver latency histogram (the reference keeps this in-process
# only; pkg/solver/optimizer.go:30-34)
WVA_SOLVER_DURATION_SECONDS = "wva_solver_duration_seconds"

# Per-phase reconcile-cycle timing (config/prepare/analyze/optimize/apply)
WVA_CYCLE_PHASE_DURATION_SECONDS = "wva_cycle_phase_duration_seconds"

# Label names
LABEL_MODEL_NAME = "model_name"
LABEL_NAMESPACE = "namespace"
LABEL_VARIANT_NAME = "variant_name"
LABEL_DIRECTION = "direction"
LABEL_REASON = "reason"
LABEL_ACCELERATOR_TYPE = "accelerator_type"

# controller ConfigMap coordinates (shared by reconciler and leader election)
CONTROLLER_NAMESPACE = "workload-variant-autoscaler-system"
