"""Batched allocation sizing over a whole System.

The reference sizes one (server, accelerator) pair at a time inside the
reconcile loop (hot loop #1: /root/reference/pkg/core/server.go:55-67 ->
allocation.go:27-163).  Here the analyze phase is *batched*: all pairs of
the fleet are packed into one [B, 12] float64 matrix and solved either by
the native extension (the torch-free C++/OpenMP binding on CPU, or the
gfx950 HIP kernel — one workgroup per pair) or by the pure-Python
analyzer fallback; mg1 mode routes through the scalar analyzer with
per-server cs^2 support.

Field layouts mirror wva_amd/csrc/queue_core.h.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..analyzer import (
    AnalyzerError,
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from ..config import MAX_QUEUE_TO_BATCH_RATIO
from ..core import Allocation, System
from ..core.allocation import _zero_load_allocation, energy_value_term

# problem columns
P_ALPHA, P_BETA, P_GAMMA, P_DELTA = 0, 1, 2, 3
P_IN_TOKENS, P_OUT_TOKENS, P_MAX_BATCH = 4, 5, 6
P_TARGET_TTFT, P_TARGET_ITL, P_TARGET_TPS = 7, 8, 9
P_TOTAL_RATE, P_MIN_REPLICAS = 10, 11
PROBLEM_FIELDS = 12

# result columns
R_FEASIBLE, R_REPLICAS, R_RATE_STAR, R_ITL, R_TTFT, R_RHO = range(6)
RESULT_FIELDS = 6

# the GPU path keeps the cumulative table in LDS: N <= ~700
GPU_MAX_BATCH_LIMIT = 700

# hipGraph dispatch cache: repeated fleet solves of the same shape replay
# a captured graph instead of re-launching.  Keyed by (B, max_k); replays
# copy fresh problem data into the captured input buffer.  Opt-in via
# WVA_GPU_GRAPH=1: the fleet solve is ONE kernel, so launch overhead is
# ~1% of the dispatch and replay measured no win (0.96 vs 0.94 ms
# headline cycle) — the path exists for launch-bound configurations
# (many small heterogeneous dispatches), not the default.  Any capture
# failure falls back to plain dispatch permanently (logged once).
_graph_cache: dict = {}
_graph_lock = None
_graph_disabled = False
_mg1_warned = False


def _graph_solve(native, problems_t, max_k):
    """Solve via a cached hipGraph; None -> caller uses plain dispatch."""
    global _graph_lock, _graph_disabled
    import os
    import threading

    if _graph_disabled or os.environ.get("WVA_GPU_GRAPH", "0") != "1":
        return None
    import torch

    if _graph_lock is None:
        _graph_lock = threading.Lock()
    key = (problems_t.shape[0], max_k)
    with _graph_lock:
        entry = _graph_cache.get(key)
        try:
            if entry is None:
                if len(_graph_cache) >= 16:
                    # bound the private graph memory pools when fleet
                    # shapes churn; oldest capture goes first
                    _graph_cache.pop(next(iter(_graph_cache)))
                static_in = problems_t.clone()
                # warm up on a side stream (capture requires a clean stream)
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    native.solve_allocations(static_in, max_k)
                torch.cuda.current_stream().wait_stream(side)
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    static_out = native.solve_allocations(static_in, max_k)
                _graph_cache[key] = (graph, static_in, static_out)
                # capture only RECORDS the work — replay to actually run
                # it for this call's data
                graph.replay()
                return static_out.cpu().numpy()
            graph, static_in, static_out = entry
            static_in.copy_(problems_t)
            graph.replay()
            return static_out.cpu().numpy()
        except Exception as e:  # pragma: no cover - depends on ROCm graph support
            _graph_disabled = True
            import logging

            logging.getLogger("wva").warning(
                "hipGraph dispatch unavailable; using plain launches: %s", e
            )
            return None


def _solve_problems_python(problems: np.ndarray, scv=1.0) -> np.ndarray:
    """Reference-semantics scalar fallback via the Python analyzer.

    ``scv`` is the service-time cs^2 — a scalar for the whole batch or a
    per-row array (mg1 auto mode derives one per server)."""
    scv_arr = np.broadcast_to(np.asarray(scv, dtype=np.float64), (problems.shape[0],))
    out = np.zeros((problems.shape[0], RESULT_FIELDS), dtype=np.float64)
    for i, row in enumerate(problems):
        N = int(row[P_MAX_BATCH])
        config = Configuration(
            max_batch_size=N,
            max_queue_size=N * MAX_QUEUE_TO_BATCH_RATIO,
            service_parms=ServiceParms(
                prefill=PrefillParms(gamma=row[P_GAMMA], delta=row[P_DELTA]),
                decode=DecodeParms(alpha=row[P_ALPHA], beta=row[P_BETA]),
            ),
        )
        try:
            qa = QueueAnalyzer(
                config,
                RequestSize(
                    avg_input_tokens=int(row[P_IN_TOKENS]),
                    avg_output_tokens=int(row[P_OUT_TOKENS]),
                ),
                scv=float(scv_arr[i]),
            )
            _, metrics, _ = qa.size(
                TargetPerf(
                    target_ttft=row[P_TARGET_TTFT],
                    target_itl=row[P_TARGET_ITL],
                    target_tps=row[P_TARGET_TPS],
                )
            )
            rate_star = metrics.throughput
            total_rate = row[P_TOTAL_RATE]
            n_rep = max(math.ceil(total_rate / rate_star), int(row[P_MIN_REPLICAS]))
            final = qa.analyze(total_rate / n_rep)
        except AnalyzerError:
            continue
        out[i, R_FEASIBLE] = 1.0
        out[i, R_REPLICAS] = n_rep
        out[i, R_RATE_STAR] = rate_star
        out[i, R_ITL] = final.avg_token_time
        out[i, R_TTFT] = final.avg_wait_time + final.avg_prefill_time
        out[i, R_RHO] = final.rho
    return out


def solve_problems(problems: np.ndarray, device: Optional[str] = None) -> np.ndarray:
    """Solve a [B, 12] problem matrix -> [B, 6] results.

    device: None for automatic (native CPU if built, else Python), "cpu" to
    force the native/Python CPU path, or a torch device string like "cuda"
    to run the gfx950 kernel.
    """
    from . import get_native, native_available

    problems = np.ascontiguousarray(problems, dtype=np.float64)
    if problems.ndim != 2 or problems.shape[1] != PROBLEM_FIELDS:
        raise ValueError(f"problems must be [B, {PROBLEM_FIELDS}]")
    if problems.shape[0] == 0:
        return np.zeros((0, RESULT_FIELDS), dtype=np.float64)

    from ..analyzer.mg1 import configured_scv

    scv = configured_scv()
    if scv != 1.0:
        # WVA_ANALYZER=mg1: the Allen-Cunneen wait scaling lives in the
        # Python analyzer; the native kernels implement the Markovian
        # contract only, so mg1 mode routes through the scalar path
        # (opt-in capacity-planning mode, not the hot default)
        global _mg1_warned
        if not _mg1_warned:
            _mg1_warned = True
            import logging

            logging.getLogger("wva_amd.ops").info(
                "WVA_ANALYZER=mg1 (cs^2=%.3g): sizing via the Python "
                "M/G/1-corrected analyzer instead of the native kernel", scv,
            )
        return _solve_problems_python(problems, scv=scv)

    want_gpu = device is not None and str(device).startswith("cuda")
    if want_gpu:
        native = get_native()  # raises loudly if missing on a GPU box
        import torch

        max_batch = problems[:, P_MAX_BATCH].max()
        if max_batch > GPU_MAX_BATCH_LIMIT:
            # LDS-resident limit: solve oversized problems on CPU
            big = problems[:, P_MAX_BATCH] > GPU_MAX_BATCH_LIMIT
            out = np.empty((problems.shape[0], RESULT_FIELDS), dtype=np.float64)
            out[~big] = solve_problems(problems[~big], device=device)
            out[big] = solve_problems(problems[big], device="cpu")
            return out
        # chain length computed host-side (also feeds hipGraph capture,
        # where the binding's device-sync fallback is illegal)
        max_k = int(max_batch) * (1 + MAX_QUEUE_TO_BATCH_RATIO)
        t = torch.from_numpy(problems).to(device)
        res = _graph_solve(native, t, max_k)
        if res is not None:
            return res
        return native.solve_allocations(t, max_k).cpu().numpy()

    from . import get_native_cpu, native_cpu_available

    if native_cpu_available():
        # torch-free binding: same host solver as the torch binding
        # (bit-identical; csrc/queue_host.h) but measured ~6x faster on
        # CPU — plain OpenMP beats at::parallel_for's per-call overhead
        # on this workload (profiles/r01_torchfree_cpu.json) — and it is
        # the only native path in the slim controller image.
        return get_native_cpu().solve_allocations(problems)
    if native_available():
        import torch

        native = get_native()
        return native.solve_allocations(torch.from_numpy(problems)).numpy()
    from . import warn_if_pure_python_fallback

    warn_if_pure_python_fallback()
    return _solve_problems_python(problems)


class BatchedAllocationSolver:
    """Batched analyze phase: computes ``server.all_allocations`` for every
    server in a System in one native call (semantically equal to
    ``System.calculate()``)."""

    def __init__(self, device: Optional[str] = None) -> None:
        self.device = device

    def calculate(self, system: System) -> None:
        for g in system.accelerators.values():
            g.calculate()

        # pre-resolve the per-(model, accelerator) constants once — fleets
        # repeat model/accelerator combinations heavily
        pair_cache: Dict[Tuple[str, str], Optional[Tuple[float, float, float, float, int, int, float]]] = {}

        def pair_info(model, acc):
            key = (model.name, acc.name)
            hit = pair_cache.get(key, False)
            if hit is not False:
                return hit
            perf = model.get_perf_data(acc.name)
            if perf is None:
                pair_cache[key] = None
                return None
            info = (
                perf.decode_parms.alpha,
                perf.decode_parms.beta,
                perf.prefill_parms.gamma,
                perf.prefill_parms.delta,
                perf.max_batch_size,
                perf.at_tokens,
                acc.cost * model.get_num_instances(acc.name),
            )
            pair_cache[key] = info
            return info

        from ..core.allocation import sizing_headroom

        headroom_factor = 1.0 + sizing_headroom()
        rows: List[Tuple[float, ...]] = []
        keys: List[Tuple[str, str, int]] = []  # (server, acc, N)
        zero_load: Dict[str, Dict[str, Allocation]] = {}
        costs: List[float] = []
        scv_rows: List[float] = []  # per-server cs^2 (negative = use global)
        rows_append, keys_append, costs_append = rows.append, keys.append, costs.append
        scv_append = scv_rows.append

        for server in system.servers.values():
            server.all_allocations = {}
            load = server.load
            if (
                load is None
                or load.arrival_rate < 0
                or load.avg_in_tokens < 0
                or load.avg_out_tokens < 0
            ):
                continue
            model = system.model(server.model_name)
            if model is None:
                continue
            svc = system.service_class(server.service_class_name)
            if svc is None:
                continue
            target = svc.model_target(server.model_name)
            if target is None:
                continue
            target_ttft, target_itl, target_tps = target.ttft, target.itl, target.tps
            arrival, in_tok, out_tok = load.arrival_rate, load.avg_in_tokens, load.avg_out_tokens
            server_name = server.name
            server_max_batch = server.max_batch_size
            min_replicas = float(server.min_num_replicas)
            candidates = server.get_candidate_accelerators(system.accelerators)
            for acc in candidates.values():
                info = pair_info(model, acc)
                if info is None:
                    continue
                if arrival == 0 or out_tok == 0:
                    perf = model.get_perf_data(acc.name)
                    alloc = _zero_load_allocation(server, model, acc, perf)
                    zero_load.setdefault(server_name, {})[acc.name] = alloc
                    continue
                alpha, beta, gamma, delta, perf_max_batch, at_tokens, cost_per_rep = info
                K = int(out_tok)
                N = server_max_batch if server_max_batch > 0 else max(perf_max_batch * at_tokens // K, 1)
                total_rate = (
                    arrival / 60.0 if target_tps == 0 else target_tps / float(K)
                ) * headroom_factor
                rows_append(
                    (alpha, beta, gamma, delta, float(int(in_tok)), float(K), float(N),
                     target_ttft, target_itl, target_tps, total_rate, min_replicas)
                )
                keys_append((server_name, acc.name, N))
                costs_append(cost_per_rep)
                scv_append(getattr(server, "service_scv", -1.0))

        if rows:
            arr = np.array(rows, dtype=np.float64)
            from ..analyzer.mg1 import configured_scv

            global_scv = configured_scv()
            scv_arr = np.array(
                [s if s >= 0 else global_scv for s in scv_rows], dtype=np.float64
            )
            if (scv_arr != 1.0).any():
                # M/G/1-corrected sizing (global mg1 mode and/or
                # per-server measured cs^2): the scalar analyzer path
                results = _solve_problems_python(arr, scv=scv_arr)
            else:
                results = solve_problems(arr, self.device)
        else:
            results = np.zeros((0, RESULT_FIELDS))

        opt_spec = system.optimizer_spec
        energy_enabled = (
            opt_spec is not None
            and opt_spec.objective == "cost+energy"
            and opt_spec.energy_cost_per_kwh > 0
        )
        servers = system.servers
        results_list = results.tolist()  # plain floats beat numpy scalar access
        for (server_name, acc_name, N), res, cost_per_replica in zip(keys, results_list, costs):
            if res[R_FEASIBLE] != 1.0:
                continue
            server = servers[server_name]
            num_replicas = int(res[R_REPLICAS])
            alloc = Allocation(
                accelerator=acc_name,
                num_replicas=num_replicas,
                batch_size=N,
                cost=cost_per_replica * num_replicas,
                itl=res[R_ITL],
                ttft=res[R_TTFT],
                rho=res[R_RHO],
                max_arrv_rate_per_replica=res[R_RATE_STAR] / 1000.0,
            )
            cur = server.cur_allocation
            alloc.value = cur.transition_penalty(alloc) if cur is not None else alloc.cost
            if energy_enabled:
                alloc.value += energy_value_term(system, server, alloc)
            server.all_allocations[acc_name] = alloc

        for server_name, accs in zero_load.items():
            server = servers[server_name]
            for acc_name, alloc in accs.items():
                if server.cur_allocation is not None:
                    alloc.set_value(server.cur_allocation.transition_penalty(alloc))
                if energy_enabled:
                    alloc.set_value(alloc.value + energy_value_term(system, server, alloc))
                server.all_allocations[acc_name] = alloc
