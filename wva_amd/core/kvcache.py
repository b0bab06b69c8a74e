"""KV-cache capacity model: what fits in an accelerator's HBM.

The reference carries ``memSize``/``memBW`` in the accelerator spec but
never consumes them; on MI355X the 288 GB of HBM3E is the reason large
models serve on a single GPU at all, so here the capacity becomes an
actual model: given a model's weight footprint and per-token KV size,
derive the maximum concurrent token budget and the max batch a context
length supports.  Used by ``tools/kv_plan.py`` for capacity planning and
by the batch-size validation helper (a *warning*, never a decision — the
reconcile path keeps reference semantics where ``maxBatchSize`` is
operator-declared).

Per-token KV bytes for a GQA transformer:
    2 (K and V) * layers * kv_heads * head_dim * dtype_bytes
e.g. Llama-3.1-8B (32 layers, 8 KV heads, dim 128, bf16) = 128 KiB/token,
so one MI355X holds ~2.1M concurrent tokens after 16 GB of weights —
batch ~500 at 4k context, which is exactly why the shipped sample pins
maxBatchSize 512 (deploy/samples/mi355x-variantautoscaling.yaml).
"""

from __future__ import annotations

from dataclasses import dataclass

__all__ = [
    "ModelMemoryProfile",
    "kv_bytes_per_token",
    "max_concurrent_tokens",
    "max_batch_for_context",
    "validate_max_batch",
]

GiB = 1024**3

# fraction of HBM reserved for activations, fragmentation, runtime pools
DEFAULT_OVERHEAD_FRACTION = 0.10


def kv_bytes_per_token(
    layers: int, kv_heads: int, head_dim: int, dtype_bytes: int = 2
) -> int:
    """Bytes of KV cache per token (K and V, all layers)."""
    if min(layers, kv_heads, head_dim, dtype_bytes) <= 0:
        raise ValueError("all dimensions must be positive")
    return 2 * layers * kv_heads * head_dim * dtype_bytes


@dataclass
class ModelMemoryProfile:
    """Weight footprint + per-token KV size of one model replica."""

    weight_bytes: float
    kv_bytes_per_token: float

    @classmethod
    def from_architecture(
        cls,
        params_billions: float,
        layers: int,
        kv_heads: int,
        head_dim: int,
        dtype_bytes: int = 2,
    ) -> "ModelMemoryProfile":
        return cls(
            weight_bytes=params_billions * 1e9 * dtype_bytes,
            kv_bytes_per_token=kv_bytes_per_token(layers, kv_heads, head_dim, dtype_bytes),
        )


def max_concurrent_tokens(
    mem_size_gb: int,
    profile: ModelMemoryProfile,
    overhead_fraction: float = DEFAULT_OVERHEAD_FRACTION,
) -> int:
    """Tokens of KV cache that fit beside the weights; 0 if weights don't fit."""
    if mem_size_gb <= 0:
        raise ValueError("mem_size_gb must be positive")
    if not 0.0 <= overhead_fraction < 1.0:
        raise ValueError("overhead_fraction must be in [0, 1)")
    usable = mem_size_gb * GiB * (1.0 - overhead_fraction) - profile.weight_bytes
    if usable <= 0:
        return 0
    return int(usable // profile.kv_bytes_per_token)


def max_batch_for_context(
    mem_size_gb: int,
    profile: ModelMemoryProfile,
    context_tokens: int,
    overhead_fraction: float = DEFAULT_OVERHEAD_FRACTION,
) -> int:
    """Max concurrent requests at an average context (prompt+generated)."""
    if context_tokens <= 0:
        raise ValueError("context_tokens must be positive")
    return max_concurrent_tokens(mem_size_gb, profile, overhead_fraction) // context_tokens


def validate_max_batch(
    declared_max_batch: int,
    mem_size_gb: int,
    profile: ModelMemoryProfile,
    context_tokens: int,
    overhead_fraction: float = DEFAULT_OVERHEAD_FRACTION,
) -> str:
    """'' when the declared batch fits; a human-readable warning otherwise.

    Advisory only: the analyzer keeps using the declared value (reference
    semantics), but an over-declared batch means vLLM will evict/preempt
    before the queueing model predicts it, so ITL/TTFT fits go stale.
    """
    fit = max_batch_for_context(mem_size_gb, profile, context_tokens, overhead_fraction)
    if declared_max_batch <= fit:
        return ""
    return (
        f"maxBatchSize {declared_max_batch} exceeds KV capacity: "
        f"{mem_size_gb} GB holds ~{fit} requests at {context_tokens} tokens "
        f"({profile.kv_bytes_per_token / 1024:.0f} KiB/token after "
        f"{profile.weight_bytes / GiB:.0f} GiB weights)"
    )
