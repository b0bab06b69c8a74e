"""Wide-range fuzz of the native batched solver against the CPU reference.

Draws problems across extreme parameter corners (tiny/huge latency
coefficients, decode-only shapes, batch limits beyond the LDS window,
TPS-target paths, zero-margin SLOs) and checks GPU↔CPU agreement: no
NaNs, bounded feasibility flips, replica counts within ±1 on agreement.

    python tools/fuzz_parity.py --n 20000 [--device cuda]
"""

from __future__ import annotations

import argparse
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

import numpy as np

from wva_amd.ops import solve_problems
from wva_amd.ops.batched import PROBLEM_FIELDS, R_FEASIBLE, R_REPLICAS, R_RATE_STAR


def draw(n, seed):
    rng = np.random.default_rng(seed)
    p = np.zeros((n, PROBLEM_FIELDS))
    p[:, 0] = 10 ** rng.uniform(-2, 3, n)  # alpha 0.01..1000 ms
    p[:, 1] = 10 ** rng.uniform(-4, 1, n)  # beta
    p[:, 2] = 10 ** rng.uniform(-2, 3, n)  # gamma
    p[:, 3] = 10 ** rng.uniform(-5, 0, n)  # delta
    p[:, 4] = np.where(rng.random(n) < 0.1, 0, rng.integers(1, 32768, n))  # in tokens
    p[:, 5] = np.where(rng.random(n) < 0.1, 1, rng.integers(1, 8192, n))  # out tokens
    p[:, 6] = np.where(rng.random(n) < 0.1, 1, rng.integers(1, 701, n))  # max batch
    # targets: mix of infeasible-tight, near-boundary and loose
    p[:, 7] = np.where(rng.random(n) < 0.2, 0, p[:, 2] * rng.uniform(0.5, 50, n) + rng.uniform(0, 5000, n))
    p[:, 8] = np.where(rng.random(n) < 0.2, 0, p[:, 0] * rng.uniform(0.8, 10, n))
    p[:, 9] = np.where(rng.random(n) < 0.8, 0, 10 ** rng.uniform(1, 4, n))  # tps
    p[:, 10] = 10 ** rng.uniform(-2, 3, n)  # total rate req/s
    p[:, 11] = rng.integers(0, 4, n)  # min replicas
    # avoid the all-targets-zero case only when TPS also zero (all free)
    return p


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=20000)
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    import torch

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    problems = draw(args.n, args.seed)
    a = solve_problems(problems, device=device)
    b = solve_problems(problems, device="cpu")

    assert np.isfinite(a).all(), "non-finite values in device results"
    assert np.isfinite(b).all(), "non-finite values in CPU results"

    flips = (a[:, R_FEASIBLE] != b[:, R_FEASIBLE]).sum()
    both = (a[:, R_FEASIBLE] == 1) & (b[:, R_FEASIBLE] == 1)
    rep_diff = np.abs(a[both, R_REPLICAS] - b[both, R_REPLICAS])
    big_rep = (rep_diff > 1).sum()
    # relative rate* agreement where replicas agree
    same = both.copy()
    same[both] = rep_diff == 0
    rel = np.abs(a[same, R_RATE_STAR] - b[same, R_RATE_STAR]) / np.maximum(b[same, R_RATE_STAR], 1e-12)
    report = {
        "n": args.n,
        "device": device,
        "feasible_frac": float(b[:, R_FEASIBLE].mean()),
        "feasibility_flips": int(flips),
        "flip_frac": float(flips / args.n),
        "replica_gt1_diffs": int(big_rep),
        "rate_star_rel_p99": float(np.percentile(rel, 99)) if rel.size else None,
        "rate_star_rel_max": float(rel.max()) if rel.size else None,
    }
    import json

    print(json.dumps(report))
    assert flips / args.n < 0.005, f"too many feasibility flips: {flips}"
    assert big_rep / max(both.sum(), 1) < 0.005, f"replica divergence: {big_rep}"
    assert rel.size and np.percentile(rel, 99) < 1e-3
    print("fuzz parity OK")


if __name__ == "__main__":
    main()
