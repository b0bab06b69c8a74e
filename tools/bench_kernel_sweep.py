"""Kernel geometry sweep: single-wave (64) vs 4-wave (256) workgroups
across max-batch sizes N up to the LDS-resident limit.

The fleet-size benches (tools/bench_queue_solver.py) showed the
single-wave kernel 20-31% faster at typical N (<=256).  Larger N means a
longer cumulative-table build (K = 11N states) and a wider significant
window per model evaluation, which could favor the extra parallelism of
4 waves — this sweep measures where (if anywhere) the crossover sits.

    python tools/bench_kernel_sweep.py --out gpurun_out/kernel_sweep.json
"""

import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))

import numpy as np

from wva_amd.ops import get_native, solve_problems
from wva_amd.ops.batched import P_MAX_BATCH, R_FEASIBLE
from test_ops import random_problems


def timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, nargs="+", default=[4096])
    ap.add_argument("--max-batches", type=int, nargs="+", default=[64, 128, 256, 512, 700])
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    import torch

    assert torch.cuda.is_available(), "sweep needs an MI355X"
    native = get_native()
    rows = []
    for B in args.batch:
        for n_hi in args.max_batches:
            problems = random_problems(B, max_batch_hi=n_hi, seed=31)
            problems[:, P_MAX_BATCH] = float(n_hi)  # pin N: uniform K = 11N chains
            t = torch.from_numpy(problems).cuda()
            entry = {"batch": B, "N": n_hi, "K": 11 * n_hi}
            results = {}
            for threads in (64, 128, 256, 384, 896, None):
                if threads == 896 and 11 * n_hi + 896 + 64 > 8192:
                    continue  # exceeds the 64 KiB LDS budget; launcher would substitute
                if threads is None:
                    os.environ.pop("WVA_GPU_THREADS", None)  # launcher auto-select
                else:
                    os.environ["WVA_GPU_THREADS"] = str(threads)

                def run():
                    native.solve_allocations(t)
                    torch.cuda.synchronize()

                entry[f"gpu{threads or 'auto'}_ms"] = timeit(run) * 1e3
                results[threads] = native.solve_allocations(t).cpu().numpy()
            # parity guards: geometries may differ at ulp level (separate
            # template instantiations contract FMAs differently), so the
            # gate is feasibility agreement + tight relative closeness of
            # the rate column on feasible rows
            cpu = solve_problems(problems, device="cpu")
            base = results[64]
            geoms = [g for g in (128, 256, 384, 896, None) if g in results]
            entry["geom_feas_agree"] = all(
                (results[g][:, R_FEASIBLE] == base[:, R_FEASIBLE]).all()
                for g in geoms
            )
            feas = base[:, R_FEASIBLE] == 1.0
            entry["max_rel_rate_diff"] = max(
                float(np.abs(
                    (results[g][feas, 2] - base[feas, 2]) / np.maximum(base[feas, 2], 1e-300)
                ).max()) if feas.any() else 0.0
                for g in geoms
            )
            entry["cpu_feas_flips"] = int((base[:, R_FEASIBLE] != cpu[:, R_FEASIBLE]).sum())
            entry["feasible_frac"] = float(base[:, R_FEASIBLE].mean())
            rows.append(entry)
            print(json.dumps(entry))

    if args.out:
        Path(args.out).parent.mkdir(parents=True, exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(rows, f, indent=2)


if __name__ == "__main__":
    main()
