"""Native batched-solver tests: parity of the C++/HIP paths against the
pure-Python analyzer (the semantic reference)."""

import numpy as np
import pytest

from wva_amd.ops import BatchedAllocationSolver, native_available, solve_problems
from wva_amd.ops.batched import (
    PROBLEM_FIELDS,
    R_FEASIBLE,
    R_ITL,
    R_RATE_STAR,
    R_REPLICAS,
    R_RHO,
    R_TTFT,
    _solve_problems_python,
)
from fixtures import make_system, server_spec

def random_problems(n, max_batch_hi=256, seed=7):
    rng = np.random.default_rng(seed)
    rows = []
    for _ in range(n):
        alpha = rng.uniform(2.0, 25.0)
        beta = rng.uniform(0.005, 0.5)
        gamma = rng.uniform(2.0, 60.0)
        delta = rng.uniform(0.001, 0.2)
        in_tok = float(rng.integers(0, 2048))
        out_tok = float(rng.integers(1, 1024))
        n_batch = float(rng.integers(1, max_batch_hi))
        # targets comfortably above alpha/gamma so most problems are feasible
        t_itl = alpha + beta * 2 + rng.uniform(1.0, 30.0)
        t_ttft = gamma + delta * in_tok * 2 + rng.uniform(10.0, 5000.0)
        t_tps = 0.0 if rng.random() < 0.7 else rng.uniform(100.0, 5000.0)
        total_rate = rng.uniform(0.01, 500.0)
        min_rep = float(rng.integers(0, 3))
        rows.append(
            [alpha, beta, gamma, delta, in_tok, out_tok, n_batch, t_ttft, t_itl, t_tps, total_rate, min_rep]
        )
    return np.asarray(rows, dtype=np.float64)


def assert_results_close(a, b, rtol=1e-6, max_boundary_flips=0.03):
    """Compare two solver result sets.

    Feasibility is decided by comparisons at the bisection boundaries, so a
    target sitting exactly on the reachable-range edge can legitimately
    flip between implementations whose floating-point summation order
    differs.  A small fraction of such flips is tolerated; rows where both
    sides agree on feasibility must match closely.
    """
    assert a.shape == b.shape
    flips = a[:, R_FEASIBLE] != b[:, R_FEASIBLE]
    allowed = max(1, int(np.ceil(max_boundary_flips * len(flips))))
    assert flips.sum() <= allowed, f"{flips.sum()} / {len(flips)} feasibility mismatches"
    feas = (a[:, R_FEASIBLE] == 1.0) & (b[:, R_FEASIBLE] == 1.0)
    # replica counts may differ by one when rate* lands on a ceil boundary
    rep_diff = np.abs(a[feas, R_REPLICAS] - b[feas, R_REPLICAS])
    assert (rep_diff <= 1).all()
    assert (rep_diff != 0).sum() <= allowed
    same_rep = np.zeros(len(a), dtype=bool)
    same_rep[feas] = a[feas, R_REPLICAS] == b[feas, R_REPLICAS]
    for col in (R_RATE_STAR, R_ITL, R_TTFT, R_RHO):
        np.testing.assert_allclose(a[same_rep, col], b[same_rep, col], rtol=rtol, atol=1e-9)


@pytest.mark.skipif(not native_available(), reason="native extension not built")
class TestNativeCPUParity:
    def test_random_problem_parity(self):
        problems = random_problems(64, seed=11)
        got = solve_problems(problems, device="cpu")
        want = _solve_problems_python(problems)
        # bisection iterates on floats: tiny tolerance differences between
        # numpy-vectorized and scalar C++ sums can shift the found lambda
        assert_results_close(got, want, rtol=1e-4)
        assert got[:, R_FEASIBLE].sum() > 0  # exercise the feasible path

    def test_large_n_parity(self):
        # beyond GPU_MAX_BATCH_LIMIT: the CPU windowed sweep handles
        # K = 11*N state chains of ~16k states (the size the LDS-limit
        # fallback in batched.py routes to this path on GPU boxes)
        problems = random_problems(12, max_batch_hi=1500, seed=17)
        problems[0, 6] = 1500.0  # pin one at the top of the range
        got = solve_problems(problems, device="cpu")
        want = _solve_problems_python(problems)
        assert_results_close(got, want, rtol=1e-4)
        assert got[:, R_FEASIBLE].sum() > 0

    def test_infeasible_targets(self):
        problems = random_problems(4, seed=14)
        problems[:, 8] = 0.01  # ITL target below alpha: infeasible
        got = solve_problems(problems, device="cpu")
        assert (got[:, R_FEASIBLE] == 0.0).all()

    def test_empty_batch(self):
        out = solve_problems(np.zeros((0, PROBLEM_FIELDS)))
        assert out.shape == (0, 6)


class TestBatchedSystemParity:
    def test_matches_scalar_system_calculate(self):
        servers = [
            server_spec("a:ns", arrival_rate=600.0),
            server_spec("b:ns", model="llama-70b", arrival_rate=1200.0, max_batch=16),
            server_spec("c:ns", class_name="Freemium", arrival_rate=90.0),
            server_spec("z:ns", arrival_rate=0.0),  # zero-load path
            server_spec("k:ns", keep_accelerator=True, cur_accelerator="MI300X", cur_replicas=2),
        ]
        sys_scalar, _ = make_system(servers=servers)
        sys_batch, _ = make_system(servers=servers)
        sys_scalar.calculate()
        BatchedAllocationSolver().calculate(sys_batch)
        for name in sys_scalar.servers:
            sa = sys_scalar.server(name).all_allocations
            ba = sys_batch.server(name).all_allocations
            assert set(sa) == set(ba), name
            for acc in sa:
                x, y = sa[acc], ba[acc]
                assert x.num_replicas == y.num_replicas
                assert x.batch_size == y.batch_size
                np.testing.assert_allclose(x.cost, y.cost, rtol=1e-6)
                np.testing.assert_allclose(x.value, y.value, rtol=1e-6, atol=1e-9)
                np.testing.assert_allclose(x.itl, y.itl, rtol=1e-4)
                np.testing.assert_allclose(x.ttft, y.ttft, rtol=1e-4, atol=1e-6)
                np.testing.assert_allclose(
                    x.max_arrv_rate_per_replica, y.max_arrv_rate_per_replica, rtol=1e-4
                )


@pytest.mark.gpu
class TestGPU:
    def test_native_loaded_and_has_hip(self):
        import torch

        assert torch.cuda.is_available()
        from wva_amd.ops import get_native

        native = get_native()
        assert native is not None and native.HAS_HIP

    def test_gpu_matches_cpu(self):
        problems = random_problems(256, seed=12)
        cpu = solve_problems(problems, device="cpu")
        gpu = solve_problems(problems, device="cuda")
        assert_results_close(gpu, cpu, rtol=1e-4)
        assert cpu[:, R_FEASIBLE].sum() > 100

    def test_gpu_matches_python_reference(self):
        problems = random_problems(32, seed=13)
        gpu = solve_problems(problems, device="cuda")
        ref = _solve_problems_python(problems)
        assert_results_close(gpu, ref, rtol=1e-4)

    @pytest.mark.parametrize("threads", ["64", "128", "256"])
    def test_each_geometry_matches_cpu(self, threads, monkeypatch):
        # pin each kernel geometry explicitly (the launcher's auto pick
        # would otherwise leave two of the three unexercised)
        monkeypatch.setenv("WVA_GPU_THREADS", threads)
        problems = random_problems(128, seed=19)
        gpu = solve_problems(problems, device="cuda")
        cpu = solve_problems(problems, device="cpu")
        assert_results_close(gpu, cpu, rtol=1e-4)
        assert gpu[:, R_FEASIBLE].sum() > 50

    def test_hipgraph_replay_matches_plain(self, monkeypatch):
        # same-shape solves replay a captured graph; fresh data must flow
        # through the static buffers on every replay (stale-buffer check)
        import wva_amd.ops.batched as B

        monkeypatch.setattr(B, "_graph_cache", {})
        monkeypatch.setattr(B, "_graph_disabled", False)
        monkeypatch.setenv("WVA_GPU_GRAPH", "1")
        for seed in (41, 42, 43):  # capture once, replay twice
            problems = random_problems(96, seed=seed)
            got = solve_problems(problems, device="cuda")
            monkeypatch.setenv("WVA_GPU_GRAPH", "0")
            plain = solve_problems(problems, device="cuda")
            monkeypatch.setenv("WVA_GPU_GRAPH", "1")
            np.testing.assert_array_equal(got, plain)
        assert not B._graph_disabled  # capture really worked, no fallback

    def test_large_batch_limit_falls_back(self):
        problems = random_problems(8, seed=15)
        problems[0, 6] = 1024.0  # beyond the LDS-resident limit
        out = solve_problems(problems, device="cuda")
        ref = solve_problems(problems, device="cpu")
        assert_results_close(out, ref, rtol=1e-4)


@pytest.mark.gpu
def test_fuzz_parity_gpu():
    """Wide-range fuzz (subset of tools/fuzz_parity.py) as a regression
    gate: zero feasibility flips, zero replica differences."""
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tools"))
    from fuzz_parity import draw

    problems = draw(4000, seed=99)
    gpu = solve_problems(problems, device="cuda")
    cpu = solve_problems(problems, device="cpu")
    assert np.isfinite(gpu).all() and np.isfinite(cpu).all()
    assert (gpu[:, R_FEASIBLE] == cpu[:, R_FEASIBLE]).all()
    both = gpu[:, R_FEASIBLE] == 1
    assert (gpu[both, R_REPLICAS] == cpu[both, R_REPLICAS]).all()


@pytest.mark.gpu
def test_concurrent_stream_submissions():
    """Two host threads driving the kernel on separate HIP streams must
    produce the same results as serial execution (extension stream-safety)."""
    import concurrent.futures as cf

    import torch

    from wva_amd.ops import get_native

    native = get_native()
    problems = random_problems(512, seed=21)
    t = torch.from_numpy(problems).cuda()
    serial = native.solve_allocations(t).cpu().numpy()

    def run_on_stream(_):
        stream = torch.cuda.Stream()
        with torch.cuda.stream(stream):
            out = native.solve_allocations(t)
        stream.synchronize()
        return out.cpu().numpy()

    with cf.ThreadPoolExecutor(max_workers=4) as pool:
        outs = list(pool.map(run_on_stream, range(8)))
    for out in outs:
        np.testing.assert_array_equal(out, serial)


class TestTorchFreeCPUBinding:
    """_queue_native_cpu (pybind11/numpy/OpenMP): the slim controller
    image's native sizing path; shares csrc/queue_host.h with the torch
    binding so results must be bit-identical to it."""

    @pytest.fixture(autouse=True)
    def _need(self):
        from wva_amd.ops import native_cpu_available

        if not native_cpu_available():
            pytest.skip("torch-free extension not built")

    def test_identical_to_torch_binding(self):
        if not native_available():
            pytest.skip("torch extension not built")
        import torch

        from wva_amd.ops import get_native, get_native_cpu

        problems = random_problems(128, seed=23)
        a = get_native_cpu().solve_allocations(problems)
        b = get_native().solve_allocations(torch.from_numpy(problems)).numpy()
        np.testing.assert_array_equal(a, b)

    def test_parity_vs_python_reference(self):
        from wva_amd.ops import get_native_cpu

        problems = random_problems(64, seed=29)
        got = get_native_cpu().solve_allocations(problems)
        want = _solve_problems_python(problems)
        assert_results_close(got, want, rtol=1e-4)
        assert got[:, R_FEASIBLE].sum() > 0

    def test_shape_validation_and_empty(self):
        from wva_amd.ops import get_native_cpu

        cpu = get_native_cpu()
        with pytest.raises(Exception):
            cpu.solve_allocations(np.zeros((3, 5)))
        assert cpu.solve_allocations(np.zeros((0, PROBLEM_FIELDS))).shape == (0, 6)

    def test_used_when_torch_absent(self):
        # the container scenario: torch unimportable -> solve_problems
        # must route to the torch-free binding, not pure Python
        import subprocess
        import sys
        from pathlib import Path

        repo = Path(__file__).resolve().parent.parent
        code = (
            "import sys\n"
            "class B:\n"
            "    # PEP 451 finder: find_spec raises, which aborts the import\n"
            "    # on every supported Python (the legacy find_module hook is\n"
            "    # gone in 3.12+)\n"
            "    def find_spec(self, n, path=None, target=None):\n"
            "        if n == 'torch' or n.startswith('torch.'):\n"
            "            raise ImportError('blocked')\n"
            "sys.meta_path.insert(0, B())\n"
            "import numpy as np\n"
            "import wva_amd.ops as ops\n"
            "assert not ops.native_available()\n"
            "assert ops.native_cpu_available()\n"
            "from unittest import mock\n"
            "p = np.zeros((2, ops.batched.PROBLEM_FIELDS)); p[:, 5] = 4; p[:, 6] = 2\n"
            "p[:, 8] = 50.0; p[:, 10] = 1.0; p[:, 11] = 1\n"
            "with mock.patch.object(ops.batched, '_solve_problems_python',\n"
            "                       side_effect=AssertionError('python fallback used')):\n"
            "    out = ops.solve_problems(p)\n"
            "assert out.shape == (2, 6) and (out[:, 0] == 1.0).all()\n"
            "print('torch-free routing OK')\n"
        )
        r = subprocess.run(
            [sys.executable, "-c", code], cwd=repo, capture_output=True, text=True, timeout=120
        )
        assert r.returncode == 0, r.stdout + r.stderr
        assert "torch-free routing OK" in r.stdout
