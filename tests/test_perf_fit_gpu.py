"""GPU perf-fit tier: the sizing parameters must come
from measurements of the regime real serving lives in — a KV-cache-bound
decode read through a page table, in bf16 and fp8 — and the shipped
MI355X profile must pass a live drift check on this box
(`make test-gpu` runs this file)."""

import sys
from pathlib import Path

import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tools"))

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def torch_cuda():
    import torch

    assert torch.cuda.is_available()
    return torch


class TestKVBoundDecodeFit:
    def test_paged_kv_decode_bf16(self, torch_cuda):
        """Long-context paged-KV decode: the fitted law must be clean
        (R^2) and its beta must carry the KV-read term — strictly above
        the short-context (weight-bound) beta for the same model."""
        from profiler.fit_perf_params import fit

        kv = fit(
            layers=4, hidden=2048, heads=16,
            batches=[1, 2, 4, 8, 16], seq_len=128,
            decode_iters=10, warmup=2, kv_len=8192, paged=True,
        )
        short = fit(
            layers=4, hidden=2048, heads=16,
            batches=[1, 2, 4, 8, 16], seq_len=128,
            decode_iters=10, warmup=2,
        )
        assert kv.r2_decode > 0.95, kv
        assert kv.beta > 0 and kv.alpha > 0
        # per-step KV bytes scale with batch: at kv_len=8192 the decode
        # slope must be dominated by cache reads, not weights
        assert kv.beta > short.beta * 2.0, (kv.beta, short.beta)

    def test_paged_kv_decode_fp8(self, torch_cuda):
        from profiler.fit_perf_params import fit

        r = fit(
            layers=4, hidden=2048, heads=16,
            batches=[1, 2, 4, 8, 16], seq_len=128,
            decode_iters=10, warmup=2, kv_len=8192, paged=True, fp8=True,
        )
        assert r.r2_decode > 0.95, r
        assert r.beta > 0 and r.alpha > 0

    def test_paged_gather_costs_more_than_contiguous(self, torch_cuda):
        # the page-table indirection is a real read pattern, not a no-op
        from profiler.fit_perf_params import fit

        paged = fit(
            layers=2, hidden=2048, heads=16, batches=[8], seq_len=64,
            decode_iters=10, warmup=2, kv_len=16384, paged=True,
        )
        contiguous = fit(
            layers=2, hidden=2048, heads=16, batches=[8], seq_len=64,
            decode_iters=10, warmup=2, kv_len=16384, paged=False,
        )
        assert paged.decode_points[0][1] >= contiguous.decode_points[0][1] * 0.9


class TestShippedProfileDriftCheck:
    def test_sample_va_profile_passes_drift_check(self, torch_cuda):
        """The guidellm-style loop closed on hardware: re-measure the
        latency laws with the same geometry the shipped MI355X profile
        was fitted with and require the configured parameters to be
        within tolerance (drift_check exits nonzero in CI when a ROCm or
        kernel change moves the curves)."""
        from profiler.drift_check import check, load_profile
        from profiler.fit_perf_params import fit

        configured = load_profile(
            str(Path(__file__).resolve().parent.parent / "deploy" / "samples" /
                "mi355x-variantautoscaling.yaml"),
            "MI355X",
        )
        result = fit(
            layers=32, hidden=4096, heads=32,
            batches=[1, 2, 4, 8, 16, 32, 64], seq_len=512,
            decode_iters=12, warmup=2,
        )
        measured = {
            "alpha": result.alpha,
            "beta": result.beta,
            "gamma": result.gamma,
            "delta": result.delta,
        }
        report = check(configured, measured, tolerance=0.35)
        assert report["ok"], report
