"""Concurrency safety: the reference relies on single-threaded execution
(package-global analyzer eval state + the TheSystem singleton;
SURVEY.md §5 'Race detection').  This rebuild eliminated both, so
concurrent optimizations over independent systems must be safe — exercised here
with threads (the closest Python analog of running `go test -race`)."""

import concurrent.futures as cf

import numpy as np

from wva_amd.core import System
from wva_amd.solver import Manager, Optimizer
from wva_amd.ops import BatchedAllocationSolver
from fixtures import make_spec, server_spec


def run_one(seed: int):
    rng = np.random.default_rng(seed)
    servers = [
        server_spec(f"s{i}:ns", arrival_rate=float(rng.uniform(30, 6000)))
        for i in range(8)
    ]
    spec = make_spec(servers=servers)
    system = System()
    opt_spec = system.set_from_spec(spec)
    BatchedAllocationSolver().calculate(system)
    Manager(system, Optimizer(opt_spec)).optimize()
    solution = system.generate_solution()
    return {name: (d.accelerator, d.num_replicas) for name, d in solution.spec.items()}


class TestConcurrentOptimization:
    def test_parallel_solves_match_serial(self):
        seeds = list(range(12))
        serial = [run_one(s) for s in seeds]
        with cf.ThreadPoolExecutor(max_workers=6) as pool:
            parallel = list(pool.map(run_one, seeds))
        assert parallel == serial

    def test_analyzer_is_reentrant(self):
        # many concurrent QueueAnalyzer.size() calls over shared-nothing
        # instances (the reference's analyzer uses package globals and
        # would race here)
        from wva_amd.analyzer import (
            Configuration,
            DecodeParms,
            PrefillParms,
            QueueAnalyzer,
            RequestSize,
            ServiceParms,
            TargetPerf,
        )

        def size_one(i: int) -> float:
            qa = QueueAnalyzer(
                Configuration(
                    max_batch_size=8 + (i % 8),
                    max_queue_size=80,
                    service_parms=ServiceParms(
                        prefill=PrefillParms(gamma=5.0, delta=0.01),
                        decode=DecodeParms(alpha=4.0 + i * 0.1, beta=0.05),
                    ),
                ),
                RequestSize(avg_input_tokens=128, avg_output_tokens=64),
            )
            _, metrics, _ = qa.size(TargetPerf(target_itl=30.0, target_ttft=2000.0))
            return metrics.throughput

        serial = [size_one(i) for i in range(16)]
        with cf.ThreadPoolExecutor(max_workers=8) as pool:
            parallel = list(pool.map(size_one, range(16)))
        assert parallel == serial

    def test_concurrent_kube_clients(self):
        from wva_amd.api import v1alpha1
        from wva_amd.api.v1alpha1.types import ObjectMeta
        from wva_amd.kube import InMemoryKubeClient

        client = InMemoryKubeClient()

        def churn(i: int) -> int:
            name = f"va-{i}"
            client.create(
                v1alpha1.VariantAutoscaling(metadata=ObjectMeta(name=name, namespace="ns"))
            )
            for _ in range(20):
                va = client.get(v1alpha1.VariantAutoscaling, name, "ns")
                va.status.desired_optimized_alloc.num_replicas += 1
                client.update_status(va)
            return client.get(
                v1alpha1.VariantAutoscaling, name, "ns"
            ).status.desired_optimized_alloc.num_replicas

        with cf.ThreadPoolExecutor(max_workers=8) as pool:
            results = list(pool.map(churn, range(16)))
        assert results == [20] * 16
