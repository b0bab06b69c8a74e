"""bench.py contract tests: JSON shape, weak-scaling distributed path
(gloo, world_size 2 — the driver launches the same way on GPU nodes)."""

import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def last_json_line(text):
    for line in reversed(text.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{text}")


def check_contract(out, n_gpus):
    assert out["metric"] == "solver_wall_clock_ms"
    assert out["unit"] == "ms"
    assert out["n_gpus"] == n_gpus
    assert out["higher_is_better"] is False
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["value"] > 0
    assert out["ms_per_step"] == out["value"]
    assert 0.0 <= out["slo_attainment_pct"] <= 100.0
    assert out["config"]["variants"] == n_gpus * 8


class TestBench:
    def test_single_process(self):
        proc = subprocess.run(
            [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
             "--variants-per-gpu", "8", "--no-observed-slo"],
            cwd=ROOT,
            capture_output=True,
            text=True,
            timeout=300,
        )
        assert proc.returncode == 0, proc.stderr
        out = last_json_line(proc.stdout)
        check_contract(out, n_gpus=1)
        # the headline is the capacity-constrained greedy solve; the
        # easier unlimited argmin rides along as a secondary series
        assert out["unlimited_ms_per_step"] > 0
        assert "greedy-limited" in out["config"]["solver_mode"]

    def test_two_rank_gloo(self):
        env = dict(os.environ, MASTER_ADDR="127.0.0.1")
        proc = subprocess.run(
            [
                sys.executable, "-m", "torch.distributed.run",
                "--nnodes=1", "--nproc-per-node", "2",
                "--master-addr", "127.0.0.1", "--master-port", "29517",
                "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
                "--variants-per-gpu", "8", "--backend", "gloo",
            ],
            cwd=ROOT,
            capture_output=True,
            text=True,
            timeout=600,
            env=env,
        )
        assert proc.returncode == 0, proc.stderr
        check_contract(last_json_line(proc.stdout), n_gpus=2)
