"""Stub Kubernetes API server: the REST surface the controller uses,
backed by an InMemoryKubeClient — the envtest analog for exercising
HTTPKubeClient (and, through it, the whole controller) over real HTTP.

List routes accept ``?watch=true`` and then stream newline-delimited
``{"type": "ADDED", "object": ...}`` events for objects created while
the connection is open (one bounded watch window of ``timeoutSeconds``).
Divergence from the real API server, documented on purpose: no initial
ADDED replay of existing objects — the consumer is the reconciler's
Create-only event filter, where replay on every reconnect would fire a
spurious wakeup per window.  Only ADDED is emitted; the reference
ignores update/delete events anyway
(variantautoscaling_controller.go:473-486).
"""

import json as _json
import queue as _queue
import time
from typing import Dict, Optional, Type

from ..api.v1alpha1.types import VariantAutoscaling
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from .client import InMemoryKubeClient
from .errors import ConflictError, NotFoundError
from .objects import ConfigMap, Deployment, Lease

_ROUTES: Dict[str, Type] = {
    "apis/llmd.ai/v1alpha1": VariantAutoscaling,
    "api/v1": ConfigMap,
    "apis/apps/v1": Deployment,
    "apis/coordination.k8s.io/v1": Lease,
}
_PLURALS = {
    VariantAutoscaling: "variantautoscalings",
    ConfigMap: "configmaps",
    Deployment: "deployments",
    Lease: "leases",
}


def create_stub_api_server(store: Optional[InMemoryKubeClient] = None):
    """Returns (FastAPI app, backing InMemoryKubeClient)."""
    store = store or InMemoryKubeClient()
    app = FastAPI(title="stub kube-apiserver")

    def dump(obj) -> dict:
        d = obj.model_dump(by_alias=True, exclude_none=True, mode="json")
        rv = d.get("metadata", {}).get("resourceVersion")
        if rv is not None:
            d["metadata"]["resourceVersion"] = str(rv)
        return d

    def handle(fn):
        try:
            return fn()
        except NotFoundError as e:
            return JSONResponse({"message": str(e)}, status_code=404)
        except ConflictError as e:
            return JSONResponse({"message": str(e)}, status_code=409)
        except Exception as e:  # validation and the rest
            return JSONResponse({"message": str(e)}, status_code=422)

    def watch_stream(cls, namespace: Optional[str], timeout_s: int) -> StreamingResponse:
        q: "_queue.Queue" = _queue.Queue()

        def hook(obj):
            if isinstance(obj, cls) and (
                namespace is None or obj.metadata.namespace == namespace
            ):
                q.put(obj)

        store.on_create(hook)

        def gen():
            try:
                deadline = time.monotonic() + timeout_s
                while True:
                    remaining = deadline - time.monotonic()
                    if remaining <= 0:
                        return
                    try:
                        obj = q.get(timeout=min(remaining, 0.25))
                    except _queue.Empty:
                        continue
                    yield _json.dumps({"type": "ADDED", "object": dump(obj)}) + "\n"
            finally:
                store.remove_create_hook(hook)

        return StreamingResponse(gen(), media_type="application/json")

    for prefix, cls in _ROUTES.items():
        plural = _PLURALS[cls]

        def make_routes(prefix=prefix, cls=cls, plural=plural):
            base = f"/{prefix}/namespaces/{{namespace}}/{plural}"

            @app.get(f"/{prefix}/{plural}")
            async def list_all(watch: bool = False, timeoutSeconds: int = 30):
                if watch:
                    return watch_stream(cls, None, timeoutSeconds)
                return {"items": [dump(o) for o in store.list(cls)]}

            @app.get(base)
            async def list_ns(namespace: str, watch: bool = False, timeoutSeconds: int = 30):
                if watch:
                    return watch_stream(cls, namespace, timeoutSeconds)
                return {"items": [dump(o) for o in store.list(cls, namespace)]}

            @app.get(base + "/{name}")
            async def get_one(namespace: str, name: str):
                return handle(lambda: dump(store.get(cls, name, namespace)))

            @app.post(base)
            async def create(namespace: str, request: Request):
                body = await request.json()
                obj = cls.model_validate(body)
                obj.metadata.namespace = namespace
                return handle(lambda: dump(store.create(obj)))

            @app.put(base + "/{name}")
            async def update(namespace: str, name: str, request: Request):
                body = await request.json()
                obj = cls.model_validate(body)
                return handle(lambda: dump(store.update(obj)))

            @app.patch(base + "/{name}")
            async def patch(namespace: str, name: str, request: Request):
                body = await request.json()

                def do():
                    cur = store.get(cls, name, namespace)
                    meta = body.get("metadata", {})
                    merged = cur.model_dump(by_alias=True, exclude_none=True, mode="json")
                    merged["metadata"].update(meta)
                    obj = cls.model_validate(merged)
                    return dump(store.update(obj))

                return handle(do)

            @app.put(base + "/{name}/status")
            async def update_status(namespace: str, name: str, request: Request):
                body = await request.json()
                obj = cls.model_validate(body)
                return handle(lambda: dump(store.update_status(obj)))

            @app.delete(base + "/{name}")
            async def delete(namespace: str, name: str):
                return handle(lambda: store.delete(cls, name, namespace) or {"status": "Success"})

        make_routes()

    return app, store
