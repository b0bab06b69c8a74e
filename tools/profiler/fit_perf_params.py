"""Fit VariantAutoscaling perf parameters from measured MI355X curves.

The autoscaler's queue analyzer is parameterized by linear latency laws

    ITL  = alpha + beta  * batchSize            (decode step, ms)
    TTFT = gamma + delta * inTokens * batchSize (prefill, ms)

The reference fits these from guidellm runs against a live vLLM server
(/root/reference/docs/tutorials/parameter-estimation.md:24-265: a
synchronous run and a max-concurrency run, two-point fit).  This tool
measures them directly on the accelerator: it runs a random-init
transformer decoder (bf16, KV-cached decode / full prefill) across a
batch-size sweep and least-squares fits the laws.  Output is a ready
``perfParms`` block for the VA profile.

Usage (on an MI355X box):
    python tools/profiler/fit_perf_params.py --layers 8 --hidden 4096 \
        --batches 1 2 4 8 16 32 64 --out gpurun_out/perf_parms.json
"""

from __future__ import annotations

import argparse
import json
import time
from dataclasses import dataclass
from typing import List, Tuple

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F


class FP8Linear(nn.Module):
    """Linear layer computing in OCP e4m3fn via torch._scaled_mm (the
    hipBLASLt fp8 path on gfx950): weights quantized once column-wise,
    activations quantized row-wise per forward — the standard quantized-
    serving recipe, so the measured curves reflect fp8 deployment."""

    def __init__(self, in_features: int, out_features: int) -> None:
        super().__init__()
        w = torch.randn(out_features, in_features) * (in_features**-0.5)
        amax = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-6)
        scale = amax / 448.0  # e4m3fn max normal
        self.register_buffer("weight_fp8", (w / scale).to(torch.float8_e4m3fn))
        self.register_buffer("weight_scale", scale.T.float())  # [1, out]

    def forward(self, x):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        amax = x2.abs().amax(dim=1, keepdim=True).clamp(min=1e-6).float()
        scale_a = amax / 448.0
        x8 = (x2 / scale_a.to(x2.dtype)).to(torch.float8_e4m3fn)
        out = torch._scaled_mm(
            x8,
            self.weight_fp8.t(),
            scale_a=scale_a,
            scale_b=self.weight_scale,
            out_dtype=torch.bfloat16,
        )
        return out.reshape(*shape[:-1], -1)


class DecoderLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, ffn_mult: int = 4, linear_cls=nn.Linear) -> None:
        super().__init__()
        self.heads = heads
        self.head_dim = hidden // heads
        if linear_cls is nn.Linear:
            self.qkv = nn.Linear(hidden, 3 * hidden, bias=False)
            self.o = nn.Linear(hidden, hidden, bias=False)
            self.up = nn.Linear(hidden, ffn_mult * hidden, bias=False)
            self.down = nn.Linear(ffn_mult * hidden, hidden, bias=False)
        else:
            self.qkv = linear_cls(hidden, 3 * hidden)
            self.o = linear_cls(hidden, hidden)
            self.up = linear_cls(hidden, ffn_mult * hidden)
            self.down = linear_cls(ffn_mult * hidden, hidden)
        self.norm1 = nn.LayerNorm(hidden)
        self.norm2 = nn.LayerNorm(hidden)

    def forward(self, x, kv_cache=None):
        # x: [B, T, H]
        B, T, H = x.shape
        residual = x
        x = self.norm1(x)
        qkv = self.qkv(x).view(B, T, 3, self.heads, self.head_dim)
        q, k, v = qkv.unbind(2)  # [B, T, heads, hd]
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))  # [B, heads, T, hd]
        if kv_cache is not None:
            k = torch.cat([kv_cache[0], k], dim=2)
            v = torch.cat([kv_cache[1], v], dim=2)
        attn = F.scaled_dot_product_attention(q, k, v, is_causal=kv_cache is None)
        x = self.o(attn.transpose(1, 2).reshape(B, T, H))
        x = residual + x
        x = x + self.down(F.silu(self.up(self.norm2(x))))
        return x, (k, v)


class MoELayer(nn.Module):
    """Mixtral-style sparse FFN: top-2 of E experts per token, tokens
    grouped per expert (the serving-style dispatch, so measured decode
    curves reflect real MoE batching behavior on CDNA4)."""

    def __init__(self, hidden: int, experts: int = 8, top_k: int = 2, ffn_mult: int = 4) -> None:
        super().__init__()
        self.experts = experts
        self.top_k = top_k
        self.gate = nn.Linear(hidden, experts, bias=False)
        self.w_up = nn.Parameter(torch.randn(experts, hidden, ffn_mult * hidden) * hidden**-0.5)
        self.w_down = nn.Parameter(torch.randn(experts, ffn_mult * hidden, hidden) * (ffn_mult * hidden) ** -0.5)

    def forward(self, x):
        B, T, H = x.shape
        flat = x.reshape(-1, H)
        logits = self.gate(flat)
        weights, topk = logits.topk(self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1, dtype=torch.float32).to(x.dtype)
        out = torch.zeros_like(flat)
        for e in range(self.experts):
            mask = topk == e  # [N, top_k]
            token_idx, slot_idx = mask.nonzero(as_tuple=True)
            if token_idx.numel() == 0:
                continue
            tokens = flat[token_idx]
            h = F.silu(tokens @ self.w_up[e]) @ self.w_down[e]
            out.index_add_(0, token_idx, h * weights[token_idx, slot_idx, None])
        return out.reshape(B, T, H)


class MoEDecoderLayer(DecoderLayer):
    def __init__(self, hidden: int, heads: int, experts: int = 8) -> None:
        super().__init__(hidden, heads)
        self.moe = MoELayer(hidden, experts=experts)

    def forward(self, x, kv_cache=None):
        B, T, H = x.shape
        residual = x
        x = self.norm1(x)
        qkv = self.qkv(x).view(B, T, 3, self.heads, self.head_dim)
        q, k, v = qkv.unbind(2)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        if kv_cache is not None:
            k = torch.cat([kv_cache[0], k], dim=2)
            v = torch.cat([kv_cache[1], v], dim=2)
        attn = F.scaled_dot_product_attention(q, k, v, is_causal=kv_cache is None)
        x = self.o(attn.transpose(1, 2).reshape(B, T, H))
        x = residual + x
        x = x + self.moe(self.norm2(x))
        return x, (k, v)


class TinyDecoder(nn.Module):
    def __init__(
        self, layers: int, hidden: int, heads: int, vocab: int = 32000,
        linear_cls=nn.Linear, moe_experts: int = 0,
    ) -> None:
        super().__init__()
        self.embed = nn.Embedding(vocab, hidden)
        if moe_experts > 0:
            self.layers = nn.ModuleList(
                MoEDecoderLayer(hidden, heads, experts=moe_experts) for _ in range(layers)
            )
        else:
            self.layers = nn.ModuleList(
                DecoderLayer(hidden, heads, linear_cls=linear_cls) for _ in range(layers)
            )
        self.head = nn.Linear(hidden, vocab, bias=False)

    def prefill(self, tokens):
        x = self.embed(tokens)
        caches = []
        for layer in self.layers:
            x, kv = layer(x)
            caches.append(kv)
        return self.head(x[:, -1:]), caches

    def decode_step(self, tokens, caches):
        x = self.embed(tokens)
        new_caches = []
        for layer, kv in zip(self.layers, caches):
            x, new_kv = layer(x, kv_cache=kv)
            new_caches.append(new_kv)
        return self.head(x), new_caches


@dataclass
class FitResult:
    alpha: float
    beta: float
    gamma: float
    delta: float
    decode_points: List[Tuple[int, float]]
    prefill_points: List[Tuple[int, float]]
    r2_decode: float
    r2_prefill: float


def _linfit(x: np.ndarray, y: np.ndarray) -> Tuple[float, float, float]:
    A = np.stack([np.ones_like(x), x], axis=1)
    coef, *_ = np.linalg.lstsq(A, y, rcond=None)
    pred = A @ coef
    ss_res = float(((y - pred) ** 2).sum())
    ss_tot = float(((y - y.mean()) ** 2).sum())
    r2 = 1.0 - ss_res / ss_tot if ss_tot > 0 else 1.0
    return float(coef[0]), float(coef[1]), r2


def synthetic_caches(
    model: TinyDecoder, B: int, kv_len: int, device: str, dtype, paged: bool,
    page_size: int = 16,
):
    """Pre-filled KV caches of length ``kv_len`` (random-init — the
    decode step's memory traffic, not the cache contents, is what's
    measured).  With ``paged`` the cache is stored page-shuffled and
    gathered through a page table each step, so the reads follow the
    block-indirection pattern of paged attention rather than one
    contiguous stream."""
    heads = model.layers[0].heads
    hd = model.layers[0].head_dim
    caches = []
    n_pages = (kv_len + page_size - 1) // page_size
    perm = torch.randperm(n_pages * page_size, device=device)[:kv_len] if paged else None
    for _ in model.layers:
        k = torch.randn(B, heads, kv_len, hd, device=device, dtype=dtype)
        v = torch.randn(B, heads, kv_len, hd, device=device, dtype=dtype)
        caches.append((k, v))
    return caches, perm


def decode_step_kv(model: TinyDecoder, tokens, caches, perm):
    """One decode step over the synthetic caches; with a page table the
    K/V reads go through index_select (the gather a paged-KV kernel
    performs), without it they are the contiguous concat path."""
    import torch.nn.functional as F

    x = model.embed(tokens)
    for layer, (k_cache, v_cache) in zip(model.layers, caches):
        B, T, H = x.shape
        residual = x
        h = layer.norm1(x)
        qkv = layer.qkv(h).view(B, T, 3, layer.heads, layer.head_dim)
        q, k, v = qkv.unbind(2)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        if perm is not None:
            k_read = k_cache.index_select(2, perm)
            v_read = v_cache.index_select(2, perm)
        else:
            k_read, v_read = k_cache, v_cache
        k = torch.cat([k_read, k], dim=2)
        v = torch.cat([v_read, v], dim=2)
        attn = F.scaled_dot_product_attention(q, k, v)
        h = layer.o(attn.transpose(1, 2).reshape(B, T, H))
        x = residual + h
        if hasattr(layer, "moe"):
            x = x + layer.moe(layer.norm2(x))
        else:
            x = x + layer.down(F.silu(layer.up(layer.norm2(x))))
    return model.head(x)


def measure(
    model: TinyDecoder,
    device: str,
    batches: List[int],
    seq_len: int,
    decode_iters: int,
    warmup: int,
    kv_len: int = 0,
    paged: bool = False,
) -> FitResult:
    def sync():
        if device.startswith("cuda"):
            torch.cuda.synchronize()

    dtype = next(model.parameters()).dtype
    decode_points = []
    prefill_points = []
    with torch.no_grad():
        for B in batches:
            tokens = torch.randint(0, 31999, (B, seq_len), device=device)
            # prefill timing
            for _ in range(warmup):
                _, caches = model.prefill(tokens)
            sync()
            t0 = time.perf_counter()
            for _ in range(max(decode_iters // 4, 1)):
                _, caches = model.prefill(tokens)
            sync()
            prefill_ms = (time.perf_counter() - t0) / max(decode_iters // 4, 1) * 1000.0
            prefill_points.append((B * seq_len, prefill_ms))

            # decode timing with KV cache: either the caches the prefill
            # produced (short-context; weight-bound) or synthetic
            # long-context caches (kv_len > 0; KV-bandwidth-bound, the
            # regime real paged-attention decode lives in)
            perm = None
            if kv_len > 0:
                del caches
                caches, perm = synthetic_caches(model, B, kv_len, device, dtype, paged)
            step_tokens = torch.randint(0, 31999, (B, 1), device=device)

            def step():
                if kv_len > 0:
                    decode_step_kv(model, step_tokens, caches, perm)
                else:
                    model.decode_step(step_tokens, caches)

            for _ in range(warmup):
                step()
            sync()
            t0 = time.perf_counter()
            for _ in range(decode_iters):
                step()
            sync()
            decode_ms = (time.perf_counter() - t0) / decode_iters * 1000.0
            decode_points.append((B, decode_ms))
            if kv_len > 0:
                del caches  # free before the next batch size

    bx = np.array([p[0] for p in decode_points], dtype=np.float64)
    by = np.array([p[1] for p in decode_points], dtype=np.float64)
    alpha, beta, r2d = _linfit(bx, by)
    px = np.array([p[0] for p in prefill_points], dtype=np.float64)
    py = np.array([p[1] for p in prefill_points], dtype=np.float64)
    gamma, delta, r2p = _linfit(px, py)
    return FitResult(alpha, beta, gamma, delta, decode_points, prefill_points, r2d, r2p)


def fit(
    *,
    layers: int = 8,
    hidden: int = 2048,
    heads: int = 16,
    batches: List[int] = (1, 2, 4, 8, 16, 32),
    seq_len: int = 256,
    decode_iters: int = 20,
    warmup: int = 3,
    device: str = None,
    dtype=torch.bfloat16,
    fp8: bool = False,
    moe_experts: int = 0,
    kv_len: int = 0,
    paged: bool = False,
) -> FitResult:
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device == "cpu":
        dtype = torch.float32
    linear_cls = FP8Linear if fp8 else nn.Linear
    model = TinyDecoder(layers, hidden, heads, linear_cls=linear_cls, moe_experts=moe_experts)
    model = model.to(device=device, dtype=dtype).eval()
    if fp8:
        # .to(dtype) converted the fp8 buffers; restore their dtypes
        for module in model.modules():
            if isinstance(module, FP8Linear):
                module.weight_fp8 = module.weight_fp8.to(torch.float8_e4m3fn)
                module.weight_scale = module.weight_scale.float()
    return fit_with_model(
        model, device, list(batches), seq_len, decode_iters, warmup,
        kv_len=kv_len, paged=paged,
    )


def fit_with_model(
    model, device, batches, seq_len, decode_iters, warmup, kv_len=0, paged=False
) -> FitResult:
    return measure(
        model, device, batches, seq_len, decode_iters, warmup,
        kv_len=kv_len, paged=paged,
    )


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=8)
    ap.add_argument("--hidden", type=int, default=4096)
    ap.add_argument("--heads", type=int, default=32)
    ap.add_argument("--batches", type=int, nargs="+", default=[1, 2, 4, 8, 16, 32, 64])
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--decode-iters", type=int, default=40)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--acc", default="MI355X")
    ap.add_argument("--dtype", choices=["bf16", "fp8"], default="bf16")
    ap.add_argument("--moe-experts", type=int, default=0,
                    help="top-2 MoE with this many experts per layer (0 = dense)")
    ap.add_argument("--kv-len", type=int, default=0,
                    help="synthetic KV-cache length for decode timing (0 = "
                         "use the prefill's short cache); large values make "
                         "the decode step KV-bandwidth bound")
    ap.add_argument("--paged", action="store_true",
                    help="read the KV cache through a shuffled page table "
                         "(paged-attention gather pattern)")
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    result = fit(
        layers=args.layers,
        hidden=args.hidden,
        heads=args.heads,
        batches=args.batches,
        seq_len=args.seq_len,
        decode_iters=args.decode_iters,
        warmup=args.warmup,
        fp8=args.dtype == "fp8",
        moe_experts=args.moe_experts,
        kv_len=args.kv_len,
        paged=args.paged,
    )
    payload = {
        "acc": args.acc,
        "model": f"tiny-decoder-L{args.layers}-H{args.hidden}-{args.dtype}"
        + (f"-moe{args.moe_experts}x" if args.moe_experts else "")
        + (f"-kv{args.kv_len}" if args.kv_len else "")
        + ("-paged" if args.paged else ""),
        "perfParms": {
            "decodeParms": {"alpha": f"{result.alpha:.4f}", "beta": f"{result.beta:.6f}"},
            "prefillParms": {"gamma": f"{result.gamma:.4f}", "delta": f"{result.delta:.8f}"},
        },
        "fit": {
            "r2_decode": result.r2_decode,
            "r2_prefill": result.r2_prefill,
            "decode_points_ms": result.decode_points,
            "prefill_points_ms": result.prefill_points,
        },
    }
    text = json.dumps(payload, indent=2)
    print(text)
    if args.out:
        with open(args.out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
