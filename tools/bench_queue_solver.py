"""Micro-benchmark of the batched allocation-sizing solver.

Compares the pure-Python analyzer, the native CPU path, and (when a GPU is
present) the gfx950 HIP kernel on identical problem batches.
"""

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tests"))

from wva_amd.ops import native_available, solve_problems
from wva_amd.ops.batched import R_FEASIBLE, _solve_problems_python
from test_ops import random_problems  # reuse the generator


def timeit(fn, warmup=2, iters=5):
    for _ in range(warmup):
        fn()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes", type=int, nargs="+", default=[64, 256, 1024, 4096])
    ap.add_argument("--max-batch", type=int, default=256)
    ap.add_argument("--python-limit", type=int, default=256)
    args = ap.parse_args()

    import torch

    has_gpu = torch.cuda.is_available()
    rows = []
    for B in args.sizes:
        problems = random_problems(B, max_batch_hi=args.max_batch)
        entry = {"batch": B, "max_batch_hi": args.max_batch}
        if B <= args.python_limit:
            entry["python_ms"] = timeit(lambda: _solve_problems_python(problems), 0, 1) * 1e3
        if native_available():
            entry["native_cpu_ms"] = timeit(lambda: solve_problems(problems, "cpu")) * 1e3
        if has_gpu:
            t = torch.from_numpy(problems).cuda()
            from wva_amd.ops import get_native

            native = get_native()

            def gpu_run():
                native.solve_allocations(t)
                torch.cuda.synchronize()

            entry["gpu_ms"] = timeit(gpu_run) * 1e3
            feas = native.solve_allocations(t).cpu().numpy()[:, R_FEASIBLE].mean()
            entry["feasible_frac"] = float(feas)
        rows.append(entry)
        print(json.dumps(entry))


if __name__ == "__main__":
    main()
