"""Stub Kubernetes API server: the REST surface the controller uses,
backed by an InMemoryKubeClient — the envtest analog for exercising
HTTPKubeClient (and, through it, the whole controller) over real HTTP.

Protocol conformance (round 2; the round-1 stub only streamed ADDED over
bounded windows):

- **server-side schema validation**: VariantAutoscaling writes are
  validated against the shipped CRD YAML's openAPIV3Schema (422 on
  violation), so specs the real apiserver rejects are rejected here;
- **chunked lists**: ``?limit=`` + ``?continue=`` with a list
  ``metadata.resourceVersion`` and continue token per chunk;
- **watch resume**: ``?watch=true&resourceVersion=N`` replays
  ADDED/MODIFIED/DELETED history after N from the store's bounded event
  log, then streams live; an N older than the retained log yields the
  apiserver's mid-stream ``ERROR`` event with a 410 Status object;
  ``resourceVersion=0``/unset synthesizes initial ADDED events for
  existing objects (the legacy list+watch form);
- **bookmarks**: ``allowWatchBookmarks=true`` emits a BOOKMARK carrying
  the current collection resourceVersion at window close, so idle
  watchers advance their resume point;
- **status-subresource conflicts**: stale resourceVersion on
  PUT .../status returns 409 (store semantics), exercised under
  concurrent writers in tests/test_kube_conformance.py.
"""

import json as _json
import time
from typing import Dict, Optional, Type

from ..api.v1alpha1.types import VariantAutoscaling
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from .client import InMemoryKubeClient
from .errors import ConflictError, GoneError, NotFoundError
from .objects import ConfigMap, Deployment, Lease, Node
from .schema import CRDValidator, SchemaValidationError

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

# how often the watch generator polls the event log (seconds)
_WATCH_POLL_S = 0.05


def create_stub_api_server(store: Optional[InMemoryKubeClient] = None):
    """Returns (FastAPI app, backing InMemoryKubeClient)."""
    store = store or InMemoryKubeClient()
    app = FastAPI(title="stub kube-apiserver")
    validator = CRDValidator()

    def dump(obj) -> dict:
        d = obj.model_dump(by_alias=True, exclude_none=True, mode="json")
        rv = d.get("metadata", {}).get("resourceVersion")
        if rv is not None:
            d["metadata"]["resourceVersion"] = str(rv)
        return d

    def status_payload(code: int, reason: str, message: str) -> dict:
        return {
            "kind": "Status",
            "apiVersion": "v1",
            "status": "Failure",
            "message": message,
            "reason": reason,
            "code": code,
        }

    def handle(fn, *, cls=None, body=None, subresource: str = ""):
        try:
            if cls is VariantAutoscaling and body is not None:
                if subresource == "status":
                    validator.validate(body, subresource="status")
                else:
                    # the status subresource strips .status from main-
                    # resource writes before validation (real apiserver
                    # semantics), so a fresh create's zero-valued status
                    # never trips minLength/pattern rules
                    body = {k: v for k, v in body.items() if k != "status"}
                    validator.validate(body)
            return fn()
        except SchemaValidationError as e:
            return JSONResponse(
                status_payload(422, "Invalid", "; ".join(e.causes)), status_code=422
            )
        except NotFoundError as e:
            return JSONResponse(
                status_payload(404, "NotFound", str(e)), status_code=404
            )
        except ConflictError as e:
            return JSONResponse(
                status_payload(409, "Conflict", str(e)), status_code=409
            )
        except GoneError as e:
            return JSONResponse(
                status_payload(410, "Expired", str(e)), status_code=410
            )
        except Exception as e:  # model validation and the rest
            return JSONResponse(
                status_payload(422, "Invalid", str(e)), status_code=422
            )

    def watch_stream(
        cls,
        namespace: Optional[str],
        timeout_s: int,
        resource_version: str,
        bookmarks: bool,
    ) -> StreamingResponse:
        kind = store._kind(cls)

        def want(obj) -> bool:
            return store._kind(obj) == kind and (
                namespace is None or obj.metadata.namespace == namespace
            )

        def event_line(etype: str, payload: dict) -> str:
            return _json.dumps({"type": etype, "object": payload}) + "\n"

        def gen():
            cursor = 0
            if resource_version:
                try:
                    cursor = int(resource_version)
                except ValueError:
                    yield event_line(
                        "ERROR",
                        status_payload(
                            410, "Expired", f"invalid resourceVersion {resource_version!r}"
                        ),
                    )
                    return
            if cursor == 0:
                # legacy list+watch form: synthesize ADDED for current
                # state, then stream from the current collection rv
                items, rv, _ = store.list_meta(cls, namespace)
                for obj in items:
                    yield event_line("ADDED", dump(obj))
                cursor = rv
            deadline = time.monotonic() + timeout_s
            while True:
                try:
                    events = store.events_since(cursor)
                except GoneError as e:
                    yield event_line("ERROR", status_payload(410, "Expired", str(e)))
                    return
                for erv, etype, obj in events:
                    cursor = erv
                    if want(obj):
                        yield event_line(etype, dump(obj))
                if time.monotonic() >= deadline:
                    if bookmarks:
                        yield event_line(
                            "BOOKMARK",
                            {"metadata": {"resourceVersion": str(cursor)}},
                        )
                    return
                time.sleep(_WATCH_POLL_S)

        return StreamingResponse(gen(), media_type="application/json")

    def list_response(cls, namespace, limit: int, continue_token: str):
        def do():
            items, rv, next_token = store.list_meta(
                cls, namespace, limit=limit, continue_token=continue_token
            )
            meta = {"resourceVersion": str(rv)}
            if next_token:
                meta["continue"] = next_token
            return {"metadata": meta, "items": [dump(o) for o in items]}

        return handle(do)

    for prefix, cls in _ROUTES.items():
        plural = _PLURALS[cls]

        def make_routes(prefix=prefix, cls=cls, plural=plural):
            base = f"/{prefix}/namespaces/{{namespace}}/{plural}"

            @app.get(f"/{prefix}/{plural}")
            async def list_all(
                watch: bool = False,
                timeoutSeconds: int = 30,
                resourceVersion: str = "",
                allowWatchBookmarks: bool = False,
                limit: int = 0,
                request: Request = None,
            ):
                if watch:
                    return watch_stream(
                        cls, None, timeoutSeconds, resourceVersion, allowWatchBookmarks
                    )
                cont = request.query_params.get("continue", "") if request else ""
                return list_response(cls, None, limit, cont)

            @app.get(base)
            async def list_ns(
                namespace: str,
                watch: bool = False,
                timeoutSeconds: int = 30,
                resourceVersion: str = "",
                allowWatchBookmarks: bool = False,
                limit: int = 0,
                request: Request = None,
            ):
                if watch:
                    return watch_stream(
                        cls, namespace, timeoutSeconds, resourceVersion, allowWatchBookmarks
                    )
                cont = request.query_params.get("continue", "") if request else ""
                return list_response(cls, namespace, limit, cont)

            @app.get(base + "/{name}")
            async def get_one(namespace: str, name: str):
                return handle(lambda: dump(store.get(cls, name, namespace)))

            @app.post(base)
            async def create(namespace: str, request: Request):
                body = await request.json()

                def do():
                    obj = cls.model_validate(body)
                    obj.metadata.namespace = namespace
                    if cls is VariantAutoscaling:
                        # status subresource: stripped on create
                        obj.status = type(obj.status)()
                    return dump(store.create(obj))

                return handle(do, cls=cls, body=body)

            @app.put(base + "/{name}")
            async def update(namespace: str, name: str, request: Request):
                body = await request.json()

                def do():
                    obj = cls.model_validate(body)
                    if cls is VariantAutoscaling:
                        # status subresource: main-resource updates keep
                        # the stored status
                        obj.status = store.get(cls, name, namespace).status
                    return dump(store.update(obj))

                return handle(do, cls=cls, body=body)

            @app.patch(base + "/{name}")
            async def patch(namespace: str, name: str, request: Request):
                body = await request.json()

                def do():
                    cur = store.get(cls, name, namespace)
                    meta = body.get("metadata", {})
                    merged = cur.model_dump(by_alias=True, exclude_none=True, mode="json")
                    merged["metadata"].update(meta)
                    if cls is VariantAutoscaling:
                        validator.validate(
                            {k: v for k, v in merged.items() if k != "status"}
                        )
                    obj = cls.model_validate(merged)
                    return dump(store.update(obj))

                return handle(do)

            @app.put(base + "/{name}/status")
            async def update_status(namespace: str, name: str, request: Request):
                body = await request.json()

                def do():
                    obj = cls.model_validate(body)
                    return dump(store.update_status(obj))

                return handle(do, cls=cls, body=body, subresource="status")

            @app.delete(base + "/{name}")
            async def delete(namespace: str, name: str):
                return handle(lambda: store.delete(cls, name, namespace) or {"status": "Success"})

        make_routes()

    # cluster-scoped core/v1 Nodes (GPU inventory for limited mode)
    @app.get("/api/v1/nodes")
    async def list_nodes(
        watch: bool = False,
        timeoutSeconds: int = 30,
        resourceVersion: str = "",
        allowWatchBookmarks: bool = False,
        limit: int = 0,
        request: Request = None,
    ):
        if watch:
            return watch_stream(
                Node, None, timeoutSeconds, resourceVersion, allowWatchBookmarks
            )
        cont = request.query_params.get("continue", "") if request else ""
        return list_response(Node, None, limit, cont)

    @app.get("/api/v1/nodes/{name}")
    async def get_node(name: str):
        return handle(lambda: dump(store.get(Node, name, "")))

    @app.post("/api/v1/nodes")
    async def create_node(request: Request):
        body = await request.json()

        def do():
            obj = Node.model_validate(body)
            obj.metadata.namespace = ""
            return dump(store.create(obj))

        return handle(do)

    @app.put("/api/v1/nodes/{name}")
    async def update_node(name: str, request: Request):
        body = await request.json()
        return handle(lambda: dump(store.update(Node.model_validate(body))))

    @app.delete("/api/v1/nodes/{name}")
    async def delete_node(name: str):
        return handle(lambda: store.delete(Node, name, "") or {"status": "Success"})

    return app, store
