"""Standalone solver CLI: one optimization pass over a SystemSpec JSON.

The reference's ``pkg/`` tree is a pure library usable outside the
controller (capacity planning against spec files; the ``sample-data``
submodule exists for exactly that).  Equivalent here:

    python -m wva_amd.solve system.json            # {"system": {...}} or bare spec
    python -m wva_amd.solve --device cuda < sys.json

Prints the allocation solution, the per-type aggregation and the solve
wall-clock as JSON.
"""

from __future__ import annotations

import argparse
import json
import sys
from typing import Optional

from .config import SystemData, SystemSpec
from .core import System
from .solver import Manager, Optimizer


def solve_spec(spec: SystemSpec, device: Optional[str] = None) -> dict:
    system = System()
    optimizer_spec = system.set_from_spec(spec)
    if device:
        from .ops import BatchedAllocationSolver

        BatchedAllocationSolver(device=device).calculate(system)
    else:
        system.calculate()
    optimizer = Optimizer(optimizer_spec)
    Manager(system, optimizer).optimize()
    solution = system.generate_solution()
    return {
        "solutionTimeMsec": optimizer.solution_time_msec,
        "allocations": {name: data.to_dict() for name, data in solution.spec.items()},
        "allocationByType": {
            t: {"count": a.count, "limit": a.limit, "cost": a.cost}
            for t, a in system.allocation_by_type.items()
        },
        "unallocated": sorted(
            name for name, server in system.servers.items() if server.allocation is None
        ),
    }


def load_spec(text: str) -> SystemSpec:
    raw = json.loads(text)
    if "system" in raw:
        return SystemData.from_dict(raw).spec
    return SystemSpec.from_dict(raw)


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="python -m wva_amd.solve", description=__doc__)
    ap.add_argument("spec", nargs="?", help="SystemSpec JSON file (default: stdin)")
    ap.add_argument("--device", default=None, help="batched sizing device (e.g. cuda)")
    ap.add_argument("--indent", type=int, default=2)
    args = ap.parse_args(argv)

    text = open(args.spec).read() if args.spec else sys.stdin.read()
    result = solve_spec(load_spec(text), device=args.device)
    print(json.dumps(result, indent=args.indent))
    return 0


if __name__ == "__main__":
    sys.exit(main())
