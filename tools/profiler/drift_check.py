"""Perf-profile drift check: re-measure the latency laws and compare with
the parameters configured in a VariantAutoscaling profile.

Perf parameters go stale — ROCm upgrades, kernel changes, new attention
paths all move alpha/beta/gamma/delta, and a drifted profile silently
mis-sizes replicas (SURVEY.md §7 'a single slipped factor').  This tool
closes that gap: run it periodically (or per rollout) on an MI355X and
alert when the measured laws deviate beyond tolerance.

    python tools/profiler/drift_check.py deploy/samples/mi355x-variantautoscaling.yaml \
        --acc MI355X --layers 32 --hidden 4096 --tolerance 0.25
"""

from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent.parent))

import yaml

from profiler.fit_perf_params import fit


def load_profile(va_path: str, acc: str, doc_index: int = 0):
    docs = [d for d in yaml.safe_load_all(Path(va_path).read_text()) if d]
    doc = docs[doc_index]
    for profile in doc["spec"]["modelProfile"]["accelerators"]:
        if profile["acc"] == acc:
            pp = profile["perfParms"]
            return {
                "alpha": float(pp["decodeParms"]["alpha"]),
                "beta": float(pp["decodeParms"]["beta"]),
                "gamma": float(pp["prefillParms"]["gamma"]),
                "delta": float(pp["prefillParms"]["delta"]),
            }
    raise KeyError(f"no profile for accelerator {acc} in {va_path}")


def relative_drift(configured: float, measured: float) -> float:
    if configured == 0:
        return float("inf") if measured != 0 else 0.0
    return abs(measured - configured) / abs(configured)


def check(configured: dict, measured: dict, tolerance: float) -> dict:
    report = {"tolerance": tolerance, "parameters": {}, "drifted": []}
    for key in ("alpha", "beta", "gamma", "delta"):
        drift = relative_drift(configured[key], measured[key])
        report["parameters"][key] = {
            "configured": configured[key],
            "measured": measured[key],
            "relative_drift": drift,
        }
        if drift > tolerance:
            report["drifted"].append(key)
    report["ok"] = not report["drifted"]
    return report


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("va_yaml", help="VariantAutoscaling manifest to check")
    ap.add_argument("--acc", default="MI355X")
    ap.add_argument("--doc-index", type=int, default=0)
    ap.add_argument("--tolerance", type=float, default=0.25, help="relative drift alarm threshold")
    ap.add_argument("--layers", type=int, default=32)
    ap.add_argument("--hidden", type=int, default=4096)
    ap.add_argument("--heads", type=int, default=32)
    ap.add_argument("--batches", type=int, nargs="+", default=[1, 2, 4, 8, 16, 32, 64])
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--decode-iters", type=int, default=30)
    args = ap.parse_args()

    configured = load_profile(args.va_yaml, args.acc, args.doc_index)
    result = fit(
        layers=args.layers,
        hidden=args.hidden,
        heads=args.heads,
        batches=args.batches,
        seq_len=args.seq_len,
        decode_iters=args.decode_iters,
    )
    measured = {
        "alpha": result.alpha,
        "beta": result.beta,
        "gamma": result.gamma,
        "delta": result.delta,  # per-token slope
    }
    report = check(configured, measured, args.tolerance)
    print(json.dumps(report, indent=2))
    sys.exit(0 if report["ok"] else 2)


if __name__ == "__main__":
    main()
