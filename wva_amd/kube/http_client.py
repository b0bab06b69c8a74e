"""HTTP Kubernetes client — the real-cluster implementation of the
KubeClient protocol.

Speaks the API-server REST conventions directly over httpx (no external
kubernetes package exists in this environment):

- core/v1 ConfigMaps, apps/v1 Deployments, llmd.ai/v1alpha1
  VariantAutoscalings (+ the /status subresource);
- in-cluster configuration from the mounted service account
  (KUBERNETES_SERVICE_HOST/PORT + token + CA), or explicit base_url/token;
- error mapping onto the package's retryability classes: 404 NotFound,
  403 Forbidden, 409 Conflict, 400/422 Invalid.

The controller is agnostic: it only sees the protocol
(wva_amd/kube/client.py), so the in-memory fake and this client are
interchangeable (exercised by tests/test_kube_http.py against a stub API
server).

Note on events: ``watch_create`` streams ADDED events over bounded watch
windows (``?watch=true&timeoutSeconds=N``), which ManagerRuntime uses
for Create-event wakeups — the same Create-only event filter the
reference applies (variantautoscaling_controller.go:473-486; everything
else remains RequeueAfter-driven).
"""

from __future__ import annotations

import os
import ssl
from typing import List, Optional, Type, TypeVar

from ..api.v1alpha1.types import VariantAutoscaling
from .errors import (
    ConflictError,
    ForbiddenError,
    GoneError,
    InvalidError,
    KubeError,
    NotFoundError,
)
from .objects import ConfigMap, Deployment, Lease, Node

T = TypeVar("T")

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

_RESOURCES = {
    # (prefix, plural, namespaced)
    VariantAutoscaling: ("apis/llmd.ai/v1alpha1", "variantautoscalings", True),
    ConfigMap: ("api/v1", "configmaps", True),
    Deployment: ("apis/apps/v1", "deployments", True),
    Lease: ("apis/coordination.k8s.io/v1", "leases", True),
    Node: ("api/v1", "nodes", False),
}


def in_cluster_config() -> dict:
    host = os.environ.get("KUBERNETES_SERVICE_HOST")
    port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
    if not host:
        raise RuntimeError("not running in a cluster (KUBERNETES_SERVICE_HOST unset)")
    with open(f"{SA_DIR}/token") as f:
        token = f.read().strip()
    return {
        "base_url": f"https://{host}:{port}",
        "token": token,
        "ca_cert_path": f"{SA_DIR}/ca.crt",
    }


class HTTPKubeClient:
    def __init__(
        self,
        base_url: Optional[str] = None,
        token: str = "",
        ca_cert_path: str = "",
        verify: bool = True,
    ) -> None:
        import httpx

        if base_url is None:
            config = in_cluster_config()
            base_url, token, ca_cert_path = (
                config["base_url"],
                config["token"],
                config["ca_cert_path"],
            )
        headers = {"Authorization": f"Bearer {token}"} if token else {}
        if base_url.startswith("https"):
            if verify and ca_cert_path:
                ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
                ctx.minimum_version = ssl.TLSVersion.TLSv1_2
                ctx.load_verify_locations(cafile=ca_cert_path)
                verify_arg = ctx
            elif verify:
                verify_arg = True
            else:
                verify_arg = False
        else:
            verify_arg = False
        self._client = httpx.Client(
            base_url=base_url, headers=headers, verify=verify_arg, timeout=10.0
        )

    # ------------------------------------------------------------------ paths
    @staticmethod
    def _resource(cls_or_obj):
        cls = cls_or_obj if isinstance(cls_or_obj, type) else type(cls_or_obj)
        try:
            return _RESOURCES[cls]
        except KeyError:
            raise TypeError(f"unregistered kind {cls.__name__}") from None

    def _path(self, cls_or_obj, namespace: str, name: str = "", subresource: str = "") -> str:
        prefix, plural, namespaced = self._resource(cls_or_obj)
        if namespaced:
            path = f"/{prefix}/namespaces/{namespace}/{plural}"
        else:  # cluster-scoped (e.g. Node)
            path = f"/{prefix}/{plural}"
        if name:
            path += f"/{name}"
        if subresource:
            path += f"/{subresource}"
        return path

    @staticmethod
    def _raise_for(resp, what: str) -> None:
        if resp.status_code < 400:
            return
        msg = f"{what}: {resp.status_code} {resp.text[:300]}"
        if resp.status_code == 404:
            raise NotFoundError(msg)
        if resp.status_code == 403:
            raise ForbiddenError(msg)
        if resp.status_code == 409:
            raise ConflictError(msg)
        if resp.status_code == 410:
            raise GoneError(msg)
        if resp.status_code in (400, 422):
            raise InvalidError(msg)
        raise KubeError(msg)

    @staticmethod
    def _dump(obj) -> dict:
        d = obj.model_dump(by_alias=True, exclude_none=True, mode="json")
        rv = d.get("metadata", {}).get("resourceVersion")
        if rv is not None:
            # the API server requires resourceVersion as a string
            d["metadata"]["resourceVersion"] = str(rv)
        return d

    # ------------------------------------------------------------------- CRUD
    def get(self, cls: Type[T], name: str, namespace: str) -> T:
        resp = self._client.get(self._path(cls, namespace, name))
        self._raise_for(resp, f"get {cls.__name__} {namespace}/{name}")
        return cls.model_validate(resp.json())

    # controller-runtime's default list chunk size
    LIST_PAGE_SIZE = 500

    def list(self, cls: Type[T], namespace: Optional[str] = None) -> List[T]:
        items, _ = self.list_with_rv(cls, namespace)
        return items

    def list_with_rv(
        self, cls: Type[T], namespace: Optional[str] = None, limit: int = 0
    ) -> tuple:
        """Chunked list following ``continue`` tokens; returns
        (items, collection resourceVersion) — the rv a watch should
        resume from."""
        prefix, plural, namespaced = self._resource(cls)
        if namespace is None:
            path = f"/{prefix}/{plural}"
        else:
            path = f"/{prefix}/namespaces/{namespace}/{plural}"
        limit = limit or self.LIST_PAGE_SIZE
        items: List[T] = []
        rv = 0
        params = {"limit": limit}
        while True:
            resp = self._client.get(path, params=params)
            self._raise_for(resp, f"list {cls.__name__}")
            body = resp.json()
            items.extend(cls.model_validate(i) for i in body.get("items", []))
            meta = body.get("metadata", {})
            if meta.get("resourceVersion") is not None:
                rv = int(meta["resourceVersion"])
            token = meta.get("continue", "")
            if not token:
                return items, rv
            params = {"limit": limit, "continue": token}

    def create(self, obj: T) -> T:
        resp = self._client.post(
            self._path(obj, obj.metadata.namespace), json=self._dump(obj)
        )
        self._raise_for(resp, f"create {type(obj).__name__}")
        return type(obj).model_validate(resp.json())

    def update(self, obj: T) -> T:
        resp = self._client.put(
            self._path(obj, obj.metadata.namespace, obj.metadata.name),
            json=self._dump(obj),
        )
        self._raise_for(resp, f"update {type(obj).__name__}")
        return type(obj).model_validate(resp.json())

    def patch_metadata(self, obj: T) -> T:
        patch = {"metadata": self._dump(obj)["metadata"]}
        resp = self._client.patch(
            self._path(obj, obj.metadata.namespace, obj.metadata.name),
            json=patch,
            headers={"Content-Type": "application/merge-patch+json"},
        )
        self._raise_for(resp, f"patch {type(obj).__name__}")
        return type(obj).model_validate(resp.json())

    def update_status(self, obj: T) -> T:
        resp = self._client.put(
            self._path(obj, obj.metadata.namespace, obj.metadata.name, "status"),
            json=self._dump(obj),
        )
        self._raise_for(resp, f"update status {type(obj).__name__}")
        return type(obj).model_validate(resp.json())

    def delete(self, cls: Type[T], name: str, namespace: str) -> None:
        resp = self._client.delete(self._path(cls, namespace, name))
        self._raise_for(resp, f"delete {cls.__name__} {namespace}/{name}")

    # ------------------------------------------------------------------ watch
    def watch_events(
        self,
        cls: Type[T],
        namespace: Optional[str] = None,
        timeout_seconds: int = 30,
        resource_version: int = 0,
        allow_bookmarks: bool = True,
    ):
        """Yield (event_type, object_or_none, resourceVersion) for one
        bounded watch window starting after ``resource_version``.

        BOOKMARK events yield (type, None, rv).  A server-side 410
        ('Expired' ERROR event or HTTP status) raises GoneError — the
        caller must re-list and resume from the fresh list rv.
        """
        import httpx
        import json

        prefix, plural, _ = self._resource(cls)
        if namespace is None:
            path = f"/{prefix}/{plural}"
        else:
            path = f"/{prefix}/namespaces/{namespace}/{plural}"
        params = {"watch": "true", "timeoutSeconds": timeout_seconds}
        if resource_version:
            params["resourceVersion"] = str(resource_version)
        if allow_bookmarks:
            params["allowWatchBookmarks"] = "true"
        with self._client.stream(
            "GET",
            path,
            params=params,
            timeout=httpx.Timeout(10.0, read=timeout_seconds + 10.0),
        ) as resp:
            if resp.status_code >= 400:
                resp.read()
                self._raise_for(resp, f"watch {cls.__name__}")
            for line in resp.iter_lines():
                if not line:
                    continue
                event = json.loads(line)
                etype = event.get("type")
                obj = event.get("object", {})
                if etype == "ERROR":
                    if obj.get("code") == 410:
                        raise GoneError(obj.get("message", "resourceVersion expired"))
                    raise KubeError(f"watch {cls.__name__}: {obj}")
                if etype == "BOOKMARK":
                    yield etype, None, int(obj["metadata"]["resourceVersion"])
                    continue
                rv = int(obj.get("metadata", {}).get("resourceVersion", 0))
                yield etype, cls.model_validate(obj), rv

    def watch_create(
        self, cls: Type[T], namespace: Optional[str] = None, timeout_seconds: int = 30,
        resource_version: int = 0,
    ):
        """Back-compat shim: ADDED objects from one watch window."""
        for etype, obj, _ in self.watch_events(
            cls, namespace, timeout_seconds, resource_version
        ):
            if etype == "ADDED":
                yield obj


class CreateWatchSession:
    """Continuity layer over bounded watch windows (reference analog:
    controller-runtime's reflector ListAndWatch).

    Round-1 weakness: windows streamed only live ADDED
    events with no resourceVersion carry-over, so creates landing between
    windows were missed until the next timer tick.  This session:

    - primes with a paged list and resumes every window from the last
      seen rv (bookmarks advance it while idle);
    - on 410 Gone re-lists and surfaces objects whose uid was not seen
      before (the creates the expired window dropped);
    - applies exponential backoff (1 s .. 30 s) to connection errors,
      resetting after any healthy window.

    ``run(callback)`` invokes callback(obj) for every create until
    ``stop()``.
    """

    BACKOFF_BASE_S = 1.0
    BACKOFF_MAX_S = 30.0

    def __init__(
        self,
        client: HTTPKubeClient,
        cls,
        namespace: Optional[str] = None,
        window_seconds: int = 30,
        stop_event=None,
    ) -> None:
        import threading

        self.client = client
        self.cls = cls
        self.namespace = namespace
        self.window_seconds = window_seconds
        self._stop = stop_event if stop_event is not None else threading.Event()
        self._rv = 0
        self._seen_uids: set = set()
        self.backoff_s = self.BACKOFF_BASE_S  # exposed for tests

    def stop(self) -> None:
        self._stop.set()

    def _prime(self):
        """(Re-)list: refresh rv and return creates missed while blind.

        On first prime this surfaces every existing object — the same
        initial-sync Add events a controller-runtime informer delivers.
        A prime also compacts the seen-uid set to the live objects, so
        long sessions on churny clusters stay bounded (deleted objects'
        uids would otherwise accumulate forever).
        """
        items, rv = self.client.list_with_rv(self.cls, self.namespace)
        self._rv = rv
        fresh = [o for o in items if o.metadata.uid not in self._seen_uids]
        self._seen_uids = {o.metadata.uid for o in items}
        return fresh

    def run(self, callback) -> None:
        primed = False
        while not self._stop.is_set():
            try:
                if not primed:
                    for obj in self._prime():
                        callback(obj)
                    primed = True
                for etype, obj, rv in self.client.watch_events(
                    self.cls,
                    self.namespace,
                    timeout_seconds=self.window_seconds,
                    resource_version=self._rv,
                ):
                    self._rv = max(self._rv, rv)
                    if self._stop.is_set():
                        return
                    if etype == "ADDED" and obj.metadata.uid not in self._seen_uids:
                        self._seen_uids.add(obj.metadata.uid)
                        callback(obj)
                self.backoff_s = self.BACKOFF_BASE_S
                if len(self._seen_uids) > self.SEEN_UIDS_LIMIT:
                    primed = False  # force a compacting re-list
            except GoneError:
                # watch history compacted: re-list immediately, resume
                # from the fresh collection rv
                primed = False
            except Exception:
                self._stop.wait(self.backoff_s)
                self.backoff_s = min(self.backoff_s * 2, self.BACKOFF_MAX_S)
