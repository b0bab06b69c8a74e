"""Controller process entry point (L0).

Counterpart of /root/reference/cmd/main.go: flag surface
(--metrics-bind-address, --health-probe-bind-address, --leader-elect,
--metrics-secure + cert paths, --enable-http2), /healthz and /readyz
probes, the controller metrics endpoint, custom metric registration, and
the manager start.  Differences, deliberate:

- leader election: with ``--kube-backend in-cluster`` it is the
  reference's protocol — a coordination/v1 Lease named
  ``72dd1cf1.llm-d.ai`` renewed by the active manager
  (wva_amd/controller/leader.py; lost leadership terminates the
  process); the memory backend falls back to an exclusive flock on a
  lock file with the same id (no API server to hold a lease);
- HTTP/2 stays disabled by default (same CVE rationale,
  cmd/main.go:107-120) — uvicorn serves HTTP/1.1;
- the Kubernetes backend is pluggable: ``--kube-backend memory`` runs
  against the in-memory fake (dev mode; the default here since no real
  cluster exists in this environment).
"""

from __future__ import annotations

import argparse
import fcntl
import os
import sys
import threading

from prometheus_client import CollectorRegistry, generate_latest

from .controller import metrics as ctrl_metrics
from .controller.logger import log
from .controller.reconciler import ManagerRuntime
from .kube import InMemoryKubeClient

LEADER_LOCK_ID = "72dd1cf1.llm-d.ai"


def parse_args(argv=None):
    ap = argparse.ArgumentParser(prog="wva-amd-controller")
    ap.add_argument("--metrics-bind-address", default="0",
                    help="metrics endpoint bind address ('0' disables; the "
                         "deploy manifests pass :8443)")
    ap.add_argument("--health-probe-bind-address", default=":8081")
    ap.add_argument("--leader-elect", action="store_true",
                    help="enable leader election for controller manager")
    ap.add_argument("--leader-lock-path", default=f"/tmp/{LEADER_LOCK_ID}.lock")
    ap.add_argument("--metrics-secure", action="store_true", default=True)
    ap.add_argument("--no-metrics-secure", dest="metrics_secure", action="store_false")
    ap.add_argument("--metrics-cert-path", default="")
    ap.add_argument("--metrics-cert-name", default="tls.crt")
    ap.add_argument("--metrics-cert-key", default="tls.key")
    # webhook certificate flags accepted for CLI parity; no admission
    # webhook is served (the reference wires them into its webhook server)
    ap.add_argument("--webhook-cert-path", default="")
    ap.add_argument("--webhook-cert-name", default="tls.crt")
    ap.add_argument("--webhook-cert-key", default="tls.key")
    ap.add_argument("--enable-http2", action="store_true", default=False)
    ap.add_argument(
        "--kube-backend",
        choices=["memory", "in-cluster", "auto"],
        default="auto",
        help="'in-cluster' uses the mounted service account against the real "
             "API server; 'auto' (default) resolves to in-cluster when "
             "KUBERNETES_SERVICE_HOST is set (i.e. running in a pod) and to "
             "the in-memory dev backend otherwise",
    )
    ap.add_argument("--max-cycles", type=int, default=None, help="exit after N reconcile cycles")
    return ap.parse_args(argv)


def acquire_leader_lock(path: str):
    """Exclusive-flock leader election; blocks until leadership."""
    fd = os.open(path, os.O_CREAT | os.O_RDWR, 0o644)
    log.info("attempting to acquire leader lease", lock=path, id=LEADER_LOCK_ID)
    fcntl.flock(fd, fcntl.LOCK_EX)
    os.ftruncate(fd, 0)
    os.write(fd, str(os.getpid()).encode())
    log.info("successfully acquired lease", id=LEADER_LOCK_ID)
    return fd


def _split_bind(addr: str, default_port: int):
    host, _, port = addr.rpartition(":")
    return host or "0.0.0.0", int(port) if port else default_port


def serve_http(args, registry: CollectorRegistry, ready_fn):
    """Health probes + metrics endpoints (separate thread)."""
    from fastapi import FastAPI, Response
    import uvicorn

    app = FastAPI(title="wva-amd-controller")

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok"}

    @app.get("/readyz")
    async def readyz():
        return {"status": "ok" if ready_fn() else "not-ready"}

    if args.metrics_bind_address != "0":
        @app.get("/metrics")
        async def metrics_endpoint():
            return Response(generate_latest(registry), media_type="text/plain; version=0.0.4")

    host, port = _split_bind(args.health_probe_bind_address, 8081)
    kwargs = {}
    if args.metrics_secure and args.metrics_cert_path:
        kwargs["ssl_certfile"] = os.path.join(args.metrics_cert_path, args.metrics_cert_name)
        kwargs["ssl_keyfile"] = os.path.join(args.metrics_cert_path, args.metrics_cert_key)
    config = uvicorn.Config(app, host=host, port=port, log_level="warning", **kwargs)
    server = uvicorn.Server(config)
    thread = threading.Thread(target=server.run, daemon=True)
    thread.start()
    return server


def main(argv=None) -> int:
    args = parse_args(argv)
    log.init()
    if not args.enable_http2:
        log.info("disabling http/2")

    registry = CollectorRegistry()
    ctrl_metrics.init_metrics(registry)

    if args.kube_backend == "auto":
        # In a pod the kubelet exports KUBERNETES_SERVICE_HOST; a helm /
        # kustomize install must never silently reconcile the in-memory
        # fake (advisor r01: released image defaulted to 'memory').
        args.kube_backend = (
            "in-cluster" if os.environ.get("KUBERNETES_SERVICE_HOST") else "memory"
        )
        log.info("kube backend auto-resolved", backend=args.kube_backend)

    if args.kube_backend == "in-cluster":
        from .kube import HTTPKubeClient

        client = HTTPKubeClient()  # service-account config
    else:
        client = InMemoryKubeClient()

    elector = None
    if args.leader_elect:
        if args.kube_backend == "in-cluster":
            # the reference's protocol: a coordination/v1 Lease named
            # 72dd1cf1.llm-d.ai, renewed by the active manager
            from .controller.constants import CONTROLLER_NAMESPACE
            from .controller.leader import LeaseElector

            elector = LeaseElector(
                client,
                namespace=os.environ.get("POD_NAMESPACE", CONTROLLER_NAMESPACE),
            )

            def _lost() -> None:
                log.error("leader lease lost; terminating")
                os._exit(1)

            elector.on_lost = _lost
            elector.acquire()
        else:
            # single-node dev backend: exclusive flock stands in
            acquire_leader_lock(args.leader_lock_path)

    ready = {"ok": False}
    server = serve_http(args, registry, lambda: ready["ok"])

    try:
        runtime = ManagerRuntime(client)
    except Exception as e:
        log.error("unable to start manager", error=str(e))
        server.should_exit = True
        return 1
    ready["ok"] = True
    log.info("starting manager")
    try:
        runtime.run(max_cycles=args.max_cycles)
    except KeyboardInterrupt:
        pass
    finally:
        runtime.stop()
        if elector is not None:
            elector.release()
        server.should_exit = True
    return 0


if __name__ == "__main__":
    sys.exit(main())
