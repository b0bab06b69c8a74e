"""The VariantAutoscaling reconciler and its manager runtime.

Parity with /root/reference/internal/controller/variantautoscaling_controller.go:

- cycle: read 3 ConfigMaps -> list & filter VAs -> build SystemSpec ->
  prepare (SLO lookup, profiles, ownerRef, metrics gate, collect) ->
  analyze -> optimize -> apply statuses + emit metrics -> requeue
  (Reconcile :86-202);
- ConfigMap names/namespace and GLOBAL_OPT_INTERVAL cadence (:74-77,94-101);
- graceful degradation: metrics unavailable skips that VA only; optimizer
  failure sets OptimizationReady=False on all prepared VAs and requeues
  (:168-186);
- Create-only event filter semantics: steady-state reconciles are purely
  timer-driven; only VA/watched-ConfigMap creation enqueues extra
  reconciles (:456-487) — implemented in ManagerRuntime;
- Prometheus bootstrap: env > ConfigMap config, mandatory HTTPS validation,
  fatal 'up' probe with long backoff (:410-452).

MI355X-native addition: the analyze phase can run through the batched
native solver (one gfx950 dispatch per cycle) via ``batched_analyzer``.
"""

from __future__ import annotations

import json
import os
import re
import threading
import time
from typing import Dict, Optional

from ..api import v1alpha1
from ..core import System
from ..kube import ConfigMap, KubeClient, NotFoundError
from ..solver import Manager, Optimizer
from . import collector
from . import metrics as ctrl_metrics
from .actuator import Actuator
from .engine import OptimizationError, VariantAutoscalingsEngine
from .interfaces import ModelAnalyzeResponse, PrometheusConfig
from .logger import log
from .modelanalyzer import ModelAnalyzer
from .promclient import HTTPPromAPI, PromAPI, parse_prometheus_config_from_env, validate_tls_config
from .utils import (
    add_model_accelerator_profile_to_system_data,
    add_server_info_to_system_data,
    create_system_data,
    find_model_slo,
    full_name,
    get_configmap_with_backoff,
    get_deployment_with_backoff,
    get_variant_autoscaling_with_backoff,
    scale_to_zero_enabled,
    update_status_with_backoff,
    validate_prometheus_api,
)

CONFIG_MAP_NAME = "workload-variant-autoscaler-variantautoscaling-config"
CONFIG_MAP_NAMESPACE = "workload-variant-autoscaler-system"
ACCELERATOR_COSTS_CM = "accelerator-unit-costs"
SERVICE_CLASSES_CM = "service-classes-config"

DEFAULT_REQUEUE_SECONDS = 60.0

_DURATION_RE = re.compile(r"(\d+(?:\.\d+)?)(ns|us|µs|ms|s|m|h)")
_DURATION_UNITS = {"ns": 1e-9, "us": 1e-6, "µs": 1e-6, "ms": 1e-3, "s": 1.0, "m": 60.0, "h": 3600.0}


def parse_go_duration(s: str) -> float:
    """Parse a Go time.ParseDuration string ('60s', '1m30s') to seconds."""
    s = s.strip()
    if not s:
        raise ValueError("empty duration")
    pos = 0
    total = 0.0
    for m in _DURATION_RE.finditer(s):
        if m.start() != pos:
            raise ValueError(f"invalid duration {s!r}")
        total += float(m.group(1)) * _DURATION_UNITS[m.group(2)]
        pos = m.end()
    if pos != len(s):
        raise ValueError(f"invalid duration {s!r}")
    return total


class ReconcileResult:
    def __init__(self, requeue_after: Optional[float] = None) -> None:
        self.requeue_after = requeue_after


class VariantAutoscalingReconciler:
    def __init__(
        self,
        client: KubeClient,
        prom_api: Optional[PromAPI] = None,
        *,
        batched_analyzer: Optional[bool] = None,
        analyzer_device: Optional[str] = None,
    ) -> None:
        self.client = client
        self.prom_api = prom_api
        if batched_analyzer is None:
            # WVA_BATCHED_ANALYZER: "1"/"0" force; unset = auto (batched
            # when a native sizing binding is importable — the deployed
            # controller then runs one native dispatch per cycle instead
            # of the scalar Python loop; semantics are identical, see
            # tests/test_controller.py::TestBatchedReconcile)
            env = os.environ.get("WVA_BATCHED_ANALYZER", "")
            if env in ("1", "true"):
                batched_analyzer = True
            elif env in ("0", "false"):
                batched_analyzer = False
            else:
                from ..ops import native_available, native_cpu_available

                batched_analyzer = native_available() or native_cpu_available()
        if batched_analyzer and analyzer_device is None:
            # WVA_ANALYZER_DEVICE: explicit "cuda"/"cpu"; unset = cuda
            # when a GPU is visible and the torch extension is built
            env_dev = os.environ.get("WVA_ANALYZER_DEVICE", "")
            if env_dev:
                analyzer_device = env_dev
            else:
                from ..ops import native_available

                if native_available():
                    import torch

                    if torch.cuda.is_available():
                        analyzer_device = "cuda"
        self.batched_analyzer = batched_analyzer
        self.analyzer_device = analyzer_device
        self.last_gpu_telemetry = {}
        self._warm_analyzer()

    def _warm_analyzer(self) -> None:
        """Prime the sizing path (imports, native extension load, first
        GPU dispatch/JIT) so the first reconcile is not 20x the steady
        state — round-1 soaks opened at 180-242 ms before settling to
        5-13 ms."""
        import numpy as np

        try:
            from ..ops import solve_problems

            # one tiny representative problem through the configured path
            row = np.array(
                [[10.0, 0.1, 20.0, 0.01, 128, 64, 8, 500.0, 24.0, 0.0, 1.0, 1.0]]
            )
            solve_problems(row, self.analyzer_device if self.batched_analyzer else None)
        except Exception as e:  # pragma: no cover - never fatal
            log.debug("analyzer warmup skipped", error=str(e))

    # ------------------------------------------------------------- config IO
    def _read_optimization_config(self) -> Dict[str, str]:
        cm = get_configmap_with_backoff(self.client, CONFIG_MAP_NAME, CONFIG_MAP_NAMESPACE)
        return cm.data

    def _read_accelerator_config(self) -> Dict[str, Dict[str, str]]:
        cm = get_configmap_with_backoff(self.client, ACCELERATOR_COSTS_CM, CONFIG_MAP_NAMESPACE)
        out: Dict[str, Dict[str, str]] = {}
        for acc, text in cm.data.items():
            try:
                out[acc] = json.loads(text)
            except json.JSONDecodeError as e:
                raise ValueError(
                    f"failed to read entry {acc} in ConfigMap "
                    f"{CONFIG_MAP_NAMESPACE}/{ACCELERATOR_COSTS_CM}: {e}"
                ) from e
        return out

    def _read_service_class_config(self) -> Dict[str, str]:
        cm = get_configmap_with_backoff(self.client, SERVICE_CLASSES_CM, CONFIG_MAP_NAMESPACE)
        return cm.data

    # -------------------------------------------------------------- reconcile
    def reconcile(self) -> ReconcileResult:
        phase_t0 = time.perf_counter()

        def mark_phase(name: str) -> None:
            nonlocal phase_t0
            now = time.perf_counter()
            ctrl_metrics.observe_cycle_phase(name, now - phase_t0)
            phase_t0 = now

        optimization_cm = self._read_optimization_config()
        interval = optimization_cm.get("GLOBAL_OPT_INTERVAL", "")
        requeue = DEFAULT_REQUEUE_SECONDS
        if interval:
            requeue = parse_go_duration(interval)

        if scale_to_zero_enabled():
            log.info("Scaling to zero is enabled!")

        accelerator_cm = self._read_accelerator_config()
        service_class_cm = self._read_service_class_config()

        va_list = self.client.list(v1alpha1.VariantAutoscaling)
        active = [va for va in va_list if va.metadata.deletion_timestamp is None]
        for va in va_list:
            if va.metadata.deletion_timestamp is not None:
                log.info("skipping deleted variantAutoscaling", name=va.name)
        if not active:
            log.info("No active VariantAutoscalings found, skipping optimization")
            return ReconcileResult(None)

        mark_phase("config")
        # live GPU inventory: in limited mode with WVA_INVENTORY=k8s the
        # capacity pool comes from the cluster's Nodes (amd.com/gpu
        # allocatable + product labels) instead of the static CM field —
        # a working version of the inventory the reference stubs
        # (collector.go:37-42); inventory failures degrade to the static
        # pool, never fail the cycle
        inventory = None
        if (
            optimization_cm.get("WVA_OPTIMIZER_MODE", "").lower() == "limited"
            and optimization_cm.get("WVA_INVENTORY", "").lower() == "k8s"
        ):
            inventory = collector.collect_inventory_k8s(self.client)
            if inventory:
                log.info(
                    "node inventory collected",
                    products={k: v["count"] for k, v in inventory.items()},
                )
        system_data = create_system_data(
            accelerator_cm, service_class_cm, optimization_cm, inventory=inventory
        )
        update_list, va_map, responses = self._prepare_variant_autoscalings(
            active, accelerator_cm, service_class_cm, system_data
        )

        # auxiliary MI355X telemetry (amd-smi exporter) per namespace —
        # best-effort observability alongside the vLLM signals; absent
        # exporters cost one query and change nothing
        self.last_gpu_telemetry = {}
        for namespace in sorted({va.namespace for va in active}):
            telemetry = collector.collect_gpu_telemetry(self.prom_api, namespace)
            if telemetry is not None and (
                telemetry.utilization_pct or telemetry.vram_used_bytes or telemetry.power_watts
            ):
                self.last_gpu_telemetry[namespace] = telemetry
                log.debug(
                    "GPU telemetry",
                    namespace=namespace,
                    gfx_pct=telemetry.utilization_pct,
                    vram_gib=telemetry.vram_used_bytes / 2**30,
                    power_w=telemetry.power_watts,
                )
        mark_phase("prepare")

        system = System()
        optimizer_spec = system.set_from_spec(system_data.spec)
        optimizer = Optimizer(optimizer_spec)
        manager = Manager(system, optimizer)

        analyzer = ModelAnalyzer(
            system, batched=self.batched_analyzer, device=self.analyzer_device
        )
        if not self.batched_analyzer:
            for g in system.accelerators.values():
                g.calculate()
        for name, server in system.servers.items():
            va = va_map.get(name)
            if va is None:
                continue
            response = analyzer.analyze_model(va)
            if not response.allocations:
                log.info("No potential allocations found for server", serverName=name)
                continue
            responses[name] = response

        mark_phase("analyze")
        engine = VariantAutoscalingsEngine(manager, system)
        try:
            optimized = engine.optimize(
                v1alpha1.VariantAutoscalingList(items=update_list), responses
            )
        except OptimizationError as e:
            log.error("unable to perform model optimization, skipping this iteration", error=str(e))
            for va in update_list:
                v1alpha1.set_condition(
                    va,
                    v1alpha1.TYPE_OPTIMIZATION_READY,
                    "False",
                    v1alpha1.REASON_OPTIMIZATION_FAILED,
                    f"Optimization failed: {e}",
                )
                try:
                    self.client.update_status(va)
                except Exception as status_err:
                    log.error(
                        "failed to update status condition after optimization failure",
                        variantAutoscaling=va.name,
                        error=str(status_err),
                    )
            return ReconcileResult(requeue)

        mark_phase("optimize")
        self._apply_optimized_allocations(update_list, optimized)
        mark_phase("apply")
        return ReconcileResult(requeue)

    # ---------------------------------------------------------------- prepare
    def _prepare_variant_autoscalings(
        self, active, accelerator_cm, service_class_cm, system_data
    ):
        update_list = []
        va_map: Dict[str, v1alpha1.VariantAutoscaling] = {}
        responses: Dict[str, ModelAnalyzeResponse] = {}

        for va in active:
            model_name = va.spec.model_id
            if not model_name:
                log.info("variantAutoscaling missing modelID, skipping", name=va.name)
                continue

            try:
                entry, class_name = find_model_slo(service_class_cm, model_name)
            except (KeyError, ValueError) as e:
                log.error("failed to locate SLO for model", name=va.name, model=model_name, error=str(e))
                continue
            log.info(
                "Found SLO for model",
                model=model_name,
                cls=class_name,
                slo_tpot=entry.slo_tpot,
                slo_ttft=entry.slo_ttft,
            )

            for profile in va.spec.model_profile.accelerators:
                try:
                    add_model_accelerator_profile_to_system_data(system_data, model_name, profile)
                except ValueError:
                    # skip the bad profile, keep the VA (controller.go:243-248)
                    log.error("variantAutoscaling bad model accelerator profile data", name=va.name)
                    continue

            acc_name = va.metadata.labels.get(collector.ACCELERATOR_LABEL, "")
            cost_str = accelerator_cm.get(acc_name, {}).get("cost")
            if cost_str is None:
                log.error("variantAutoscaling missing accelerator cost in configMap, skipping", name=va.name)
                continue
            try:
                cost = float(cost_str)
            except ValueError:
                log.error("variantAutoscaling unable to parse accelerator cost, skipping", name=va.name)
                continue

            try:
                deploy = get_deployment_with_backoff(self.client, va.name, va.namespace)
            except Exception as e:
                log.error("failed to get Deployment after retries", name=va.name, error=str(e))
                continue

            try:
                update_va = get_variant_autoscaling_with_backoff(
                    self.client, deploy.name, deploy.namespace
                )
            except Exception as e:
                log.error("unable to get variantAutoscaling for deployment", name=deploy.name, error=str(e))
                continue

            # ownerReference before the metrics gate so GC works even when
            # metrics never arrive (controller.go:276-293)
            if not any(
                ref.kind == "Deployment" and ref.name == deploy.name and ref.controller
                for ref in update_va.metadata.owner_references
            ):
                update_va.metadata.owner_references.append(
                    v1alpha1.types.OwnerReference(
                        apiVersion="apps/v1",
                        kind="Deployment",
                        name=deploy.name,
                        uid=deploy.metadata.uid,
                        controller=True,
                        blockOwnerDeletion=False,
                    )
                )
                try:
                    update_va = self.client.patch_metadata(update_va)
                except Exception as e:
                    log.error("failed to patch ownerReference", name=update_va.name, error=str(e))
                    continue
                log.info("Set ownerReference on VariantAutoscaling", name=update_va.name, owner=deploy.name)

            validation = collector.validate_metrics_availability(
                self.prom_api, model_name, deploy.namespace
            )
            if validation.available:
                v1alpha1.set_condition(
                    update_va,
                    v1alpha1.TYPE_METRICS_AVAILABLE,
                    "True",
                    validation.reason,
                    validation.message,
                )
            else:
                log.warn(
                    "Metrics unavailable, skipping optimization for variant",
                    variant=update_va.name,
                    namespace=update_va.namespace,
                    model=model_name,
                    reason=validation.reason,
                    troubleshooting=validation.message,
                )
                continue

            try:
                current_alloc = collector.add_metrics_to_opt_status(
                    update_va, deploy, cost, self.prom_api
                )
            except Exception as e:
                log.error("unable to fetch metrics, skipping this variantAutoscaling loop", error=str(e))
                continue
            update_va.status.current_alloc = current_alloc

            try:
                add_server_info_to_system_data(system_data, update_va, class_name)
            except Exception:
                log.info("variantAutoscaling bad deployment server data, skipping", name=update_va.name)
                continue

            # mg1 auto mode: per-model cs^2 from the measured generation-
            # token histogram (never fatal; absent histogram -> the
            # fleet-wide WVA_SERVICE_SCV fallback applies)
            from ..analyzer.mg1 import auto_scv_enabled, recommended_service_scv

            if auto_scv_enabled():
                try:
                    token_scv = collector.estimate_token_scv(
                        self.prom_api, model_name, update_va.namespace
                    )
                    if token_scv is not None:
                        profile = update_va.spec.model_profile.accelerators[0]
                        pp = profile.perf_parms
                        scv = recommended_service_scv(
                            mean_out_tokens=float(
                                update_va.status.current_alloc.load.avg_output_tokens or 0
                            ),
                            token_scv=token_scv,
                            alpha=float(pp.decode_parms.get("alpha", "0")),
                            beta=float(pp.decode_parms.get("beta", "0")),
                            gamma=float(pp.prefill_parms.get("gamma", "0")),
                            delta=float(pp.prefill_parms.get("delta", "0")),
                            in_tokens=float(
                                update_va.status.current_alloc.load.avg_input_tokens or 0
                            ),
                        )
                        system_data.spec.servers.spec[-1].service_scv = scv
                        log.info(
                            "auto service cs^2 from token histogram",
                            variant=update_va.name,
                            token_scv=round(token_scv, 4),
                            service_scv=round(scv, 4),
                        )
                except Exception as e:
                    log.warn("token-scv estimation failed; using fallback", error=str(e))

            update_list.append(update_va)
            va_map[full_name(va.name, va.namespace)] = va
        return update_list, va_map, responses

    # ------------------------------------------------------------------ apply
    def _apply_optimized_allocations(self, update_list, optimized) -> None:
        for va in update_list:
            if va.name not in optimized:
                log.debug("No optimized allocation found for variant", name=va.name)
                continue
            try:
                update_va = get_variant_autoscaling_with_backoff(
                    self.client, va.name, va.namespace
                )
            except Exception as e:
                log.error("failed to get latest VariantAutoscaling", name=va.name, error=str(e))
                continue

            update_va.status.current_alloc = va.status.current_alloc
            update_va.status.desired_optimized_alloc = optimized[va.name]
            update_va.status.actuation.applied = False
            # preserve conditions set during preparation
            update_va.status.conditions = va.status.conditions

            v1alpha1.set_condition(
                update_va,
                v1alpha1.TYPE_OPTIMIZATION_READY,
                "True",
                v1alpha1.REASON_OPTIMIZATION_SUCCEEDED,
                f"Optimization completed: {update_va.status.desired_optimized_alloc.num_replicas} "
                f"replicas on {update_va.status.desired_optimized_alloc.accelerator}",
            )

            actuator = Actuator(self.client)
            try:
                actuator.emit_metrics(update_va)
                update_va.status.actuation.applied = True
            except Exception as e:
                log.error("failed to emit optimization signals", variant=update_va.name, error=str(e))

            try:
                update_status_with_backoff(self.client, update_va, "VariantAutoscaling")
            except Exception as e:
                log.error("failed to patch status after retries", name=update_va.name, error=str(e))
                continue

        if update_list:
            log.info(
                "Reconciliation completed",
                variants_processed=len(update_list),
                optimization_successful=True,
            )


class ManagerRuntime:
    """controller-runtime manager analog: Prometheus bootstrap + the
    requeue-driven reconcile loop with Create-only event triggers."""

    def __init__(
        self,
        client: KubeClient,
        prom_api: Optional[PromAPI] = None,
        prom_config: Optional[PrometheusConfig] = None,
        **reconciler_kw,
    ) -> None:
        self.client = client
        if prom_api is None:
            config = prom_config or self._get_prometheus_config()
            validate_tls_config(config)
            log.info("Initializing Prometheus client", address=config.base_url, tls_enabled=True)
            prom_api = HTTPPromAPI(config)
            validate_prometheus_api(prom_api)
            log.info("Prometheus client and API wrapper initialized and validated successfully")
        self.reconciler = VariantAutoscalingReconciler(client, prom_api, **reconciler_kw)
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._watch_threads: list = []
        if hasattr(client, "on_create"):
            client.on_create(self._on_create)
        elif hasattr(client, "watch_events"):
            # HTTP tier: Create-event wakeups via resumable watch
            # sessions — resourceVersion continuity across windows, 410
            # re-list recovery, exponential backoff
            from ..kube.http_client import CreateWatchSession

            self._watch_sessions = []
            for cls, namespace in (
                (v1alpha1.VariantAutoscaling, None),
                (ConfigMap, CONFIG_MAP_NAMESPACE),
            ):
                session = CreateWatchSession(
                    client, cls, namespace, window_seconds=5, stop_event=self._stop
                )
                self._watch_sessions.append(session)
                t = threading.Thread(
                    target=session.run, args=(self._on_create,), daemon=True
                )
                t.start()
                self._watch_threads.append(t)

    def _get_prometheus_config(self) -> PrometheusConfig:
        config = parse_prometheus_config_from_env()
        if config.base_url:
            log.info("Using Prometheus configuration from environment variables", address=config.base_url)
            return config
        try:
            cm = self.client.get(ConfigMap, CONFIG_MAP_NAME, CONFIG_MAP_NAMESPACE)
        except NotFoundError:
            cm = None
        if cm is not None and cm.data.get("PROMETHEUS_BASE_URL"):
            d = cm.data
            log.info("Using Prometheus configuration from ConfigMap", address=d["PROMETHEUS_BASE_URL"])
            return PrometheusConfig(
                base_url=d["PROMETHEUS_BASE_URL"],
                insecure_skip_verify=d.get("PROMETHEUS_TLS_INSECURE_SKIP_VERIFY", "") == "true",
                ca_cert_path=d.get("PROMETHEUS_CA_CERT_PATH", ""),
                client_cert_path=d.get("PROMETHEUS_CLIENT_CERT_PATH", ""),
                client_key_path=d.get("PROMETHEUS_CLIENT_KEY_PATH", ""),
                server_name=d.get("PROMETHEUS_SERVER_NAME", ""),
                bearer_token=d.get("PROMETHEUS_BEARER_TOKEN", ""),
            )
        raise RuntimeError(
            "no Prometheus configuration found. Please set PROMETHEUS_BASE_URL "
            "environment variable or configure via ConfigMap"
        )

    def _on_create(self, obj) -> None:
        # Create-only event filter: VAs and the watched ConfigMap enqueue
        if isinstance(obj, v1alpha1.VariantAutoscaling):
            self._wake.set()
        elif (
            isinstance(obj, ConfigMap)
            and obj.name == CONFIG_MAP_NAME
            and obj.namespace == CONFIG_MAP_NAMESPACE
        ):
            self._wake.set()

    def run_once(self) -> ReconcileResult:
        return self.reconciler.reconcile()

    def run(self, max_cycles: Optional[int] = None) -> None:
        cycles = 0
        while not self._stop.is_set():
            try:
                result = self.reconciler.reconcile()
            except Exception as e:
                log.error("reconcile failed", error=str(e))
                result = ReconcileResult(DEFAULT_REQUEUE_SECONDS)
            cycles += 1
            if max_cycles is not None and cycles >= max_cycles:
                return
            timeout = result.requeue_after
            self._wake.wait(timeout=timeout)
            self._wake.clear()

    def stop(self) -> None:
        self._stop.set()
        self._wake.set()
