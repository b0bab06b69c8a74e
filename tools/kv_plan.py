"""KV-cache capacity planner for MI355X (and any priced accelerator).

Answers "what maxBatchSize can this accelerator actually hold for this
model at this context length?" from the memory model in
wva_amd/core/kvcache.py — the number the VariantAutoscaling sample
pins as maxBatchSize should not exceed it.

    python tools/kv_plan.py --params-b 8 --layers 32 --kv-heads 8 \
        --head-dim 128 --context 4096
    python tools/kv_plan.py --params-b 70 --layers 80 --kv-heads 8 \
        --head-dim 128 --context 8192 --acc MI300X
"""

from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from wva_amd.config.mi355x import MI355X_CATALOG
from wva_amd.core.kvcache import (
    ModelMemoryProfile,
    max_batch_for_context,
    max_concurrent_tokens,
)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--acc", default="MI355X", choices=sorted(MI355X_CATALOG))
    ap.add_argument("--params-b", type=float, required=True, help="parameters, billions")
    ap.add_argument("--layers", type=int, required=True)
    ap.add_argument("--kv-heads", type=int, required=True)
    ap.add_argument("--head-dim", type=int, required=True)
    ap.add_argument("--dtype-bytes", type=int, default=2, help="2=bf16, 1=fp8")
    ap.add_argument("--context", type=int, default=4096, help="avg prompt+generated tokens")
    ap.add_argument("--overhead", type=float, default=0.10)
    args = ap.parse_args()

    acc = MI355X_CATALOG[args.acc]
    profile = ModelMemoryProfile.from_architecture(
        args.params_b, args.layers, args.kv_heads, args.head_dim, args.dtype_bytes
    )
    tokens = max_concurrent_tokens(acc.mem_size, profile, args.overhead)
    batch = max_batch_for_context(acc.mem_size, profile, args.context, args.overhead)
    print(
        json.dumps(
            {
                "accelerator": args.acc,
                "mem_size_gb": acc.mem_size,
                "weight_gib": round(profile.weight_bytes / 1024**3, 1),
                "kv_kib_per_token": round(profile.kv_bytes_per_token / 1024, 1),
                "max_concurrent_tokens": tokens,
                "context_tokens": args.context,
                "max_batch": batch,
            },
            indent=2,
        )
    )


if __name__ == "__main__":
    main()
