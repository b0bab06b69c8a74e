"""KubeClient protocol and the in-memory (envtest-analog) implementation.

The fake implements the API-machinery behaviors the controller depends on:
typed CRUD, a separate status subresource, resourceVersion bumping with
conflict detection on stale updates, deep-copy isolation (readers never
alias store state), ownerReference cascade deletion (GC), and create-event
hooks for the reconciler's Create-only event filter
(/root/reference/internal/controller/variantautoscaling_controller.go:473-486).

Protocol-conformance surface (exercised through the HTTP stub server,
tests/test_kube_conformance.py):

- a bounded **watch event log** (ADDED/MODIFIED/DELETED, monotone
  resourceVersions).  ``events_since(rv)`` replays history; an rv older
  than the retained window reports expired — the API server's
  ``410 Gone`` that forces clients to re-list;
- **paged lists**: ``list_meta`` returns (items, collection rv,
  continue-token) honoring ``limit``/``continue`` like the real
  chunked-list API.
"""

from __future__ import annotations

import copy
import datetime
import threading
from typing import Callable, Dict, List, Optional, Protocol, Tuple, Type, TypeVar

from ..api.v1alpha1.types import VariantAutoscaling
from .errors import ConflictError, GoneError, NotFoundError
from .objects import ConfigMap, Deployment, Lease, Node

T = TypeVar("T")

_KINDS = {
    VariantAutoscaling: "VariantAutoscaling",
    ConfigMap: "ConfigMap",
    Deployment: "Deployment",
    Lease: "Lease",
    Node: "Node",  # cluster-scoped: stored under namespace ""
}

# Watch history bound: events older than the newest EVENT_LOG_LIMIT are
# compacted away and watches resuming from before them get 410 Gone
# (etcd's compaction analog; small by default so tests exercise it).
EVENT_LOG_LIMIT = 1024


class KubeClient(Protocol):
    def get(self, cls: Type[T], name: str, namespace: str) -> T: ...

    def list(self, cls: Type[T], namespace: Optional[str] = None) -> List[T]: ...

    def create(self, obj: T) -> T: ...

    def update(self, obj: T) -> T: ...

    def patch_metadata(self, obj: T) -> T: ...

    def update_status(self, obj: T) -> T: ...

    def delete(self, cls: Type[T], name: str, namespace: str) -> None: ...


class InMemoryKubeClient:
    def __init__(self, event_log_limit: int = EVENT_LOG_LIMIT) -> None:
        self._lock = threading.RLock()
        self._store: Dict[Tuple[str, str, str], object] = {}
        self._rv = 0
        self._uid_n = 0
        self._create_hooks: List[Callable[[object], None]] = []
        # (rv, "ADDED"|"MODIFIED"|"DELETED", deep-copied object)
        self._events: List[Tuple[int, str, object]] = []
        self._event_log_limit = event_log_limit
        self._compacted_to = 0  # rvs <= this may be gone from the log

    # -- helpers ------------------------------------------------------------
    @staticmethod
    def _kind(obj_or_cls) -> str:
        cls = obj_or_cls if isinstance(obj_or_cls, type) else type(obj_or_cls)
        try:
            return _KINDS[cls]
        except KeyError:
            raise TypeError(f"unregistered kind {cls.__name__}") from None

    def _key(self, obj) -> Tuple[str, str, str]:
        return (self._kind(obj), obj.metadata.namespace, obj.metadata.name)

    def _next_rv(self) -> int:
        self._rv += 1
        return self._rv

    def _record(self, event_type: str, stored) -> None:
        """Append to the watch log (caller holds the lock; ``stored`` is
        already a private copy)."""
        self._events.append((stored.metadata.resource_version, event_type, stored))
        if len(self._events) > self._event_log_limit:
            drop = len(self._events) - self._event_log_limit
            self._compacted_to = self._events[drop - 1][0]
            del self._events[:drop]

    def on_create(self, hook: Callable[[object], None]) -> None:
        """Register a create-event hook (the only event type the reference's
        reconciler reacts to)."""
        self._create_hooks.append(hook)

    def remove_create_hook(self, hook: Callable[[object], None]) -> None:
        """Unsubscribe a hook (watch connections detach on close)."""
        try:
            self._create_hooks.remove(hook)
        except ValueError:
            pass

    # -- watch/list protocol surface ----------------------------------------
    @property
    def resource_version(self) -> int:
        """Current collection resourceVersion (what a fresh list returns)."""
        with self._lock:
            return self._rv

    def events_since(self, rv: int) -> List[Tuple[int, str, object]]:
        """Events with resourceVersion > rv, oldest first.

        Raises GoneError when rv predates the retained log — the caller
        must re-list (apiserver watch-cache/etcd-compaction semantics).
        """
        with self._lock:
            if rv < self._compacted_to:
                raise GoneError(
                    f"too old resource version: {rv} ({self._compacted_to})"
                )
            return [
                (erv, et, copy.deepcopy(obj))
                for (erv, et, obj) in self._events
                if erv > rv
            ]

    def list_meta(
        self,
        cls: Type[T],
        namespace: Optional[str] = None,
        limit: int = 0,
        continue_token: str = "",
    ) -> Tuple[List[T], int, str]:
        """Chunked list: (items, collection resourceVersion, continue).

        The continue token is "<offset>:<rv>" over the stable
        (namespace, name) ordering; like the real API server, every chunk
        of one logical list reports the resourceVersion the list started
        at.
        """
        kind = self._kind(cls)
        with self._lock:
            all_items = [
                o
                for (k, ns, _), o in sorted(self._store.items())
                if k == kind and (namespace is None or ns == namespace)
            ]
            offset = 0
            rv = self._rv
            if continue_token:
                try:
                    off_s, rv_s = continue_token.split(":", 1)
                    offset, rv = int(off_s), int(rv_s)
                except ValueError:
                    raise GoneError(f"invalid continue token {continue_token!r}")
            if limit and offset + limit < len(all_items):
                chunk = all_items[offset : offset + limit]
                next_token = f"{offset + limit}:{rv}"
            else:
                chunk = all_items[offset:]
                next_token = ""
            return [copy.deepcopy(o) for o in chunk], rv, next_token

    # -- CRUD ---------------------------------------------------------------
    def get(self, cls: Type[T], name: str, namespace: str) -> T:
        with self._lock:
            obj = self._store.get((self._kind(cls), namespace, name))
            if obj is None:
                raise NotFoundError(f"{cls.__name__} {namespace}/{name} not found")
            return copy.deepcopy(obj)  # type: ignore[return-value]

    def list(self, cls: Type[T], namespace: Optional[str] = None) -> List[T]:
        items, _, _ = self.list_meta(cls, namespace)
        return items

    def create(self, obj: T) -> T:
        with self._lock:
            key = self._key(obj)
            if key in self._store:
                # AlreadyExists is a Conflict in this package's error
                # taxonomy (the API server's 409) — leader election's
                # create race depends on it
                raise ConflictError(f"{key} already exists")
            stored = copy.deepcopy(obj)
            stored.metadata.resource_version = self._next_rv()
            self._uid_n += 1
            stored.metadata.uid = stored.metadata.uid or f"uid-{self._uid_n}"
            stored.metadata.creation_timestamp = datetime.datetime.now(
                datetime.timezone.utc
            )
            self._store[key] = stored
            self._record("ADDED", copy.deepcopy(stored))
            result = copy.deepcopy(stored)
        for hook in self._create_hooks:
            hook(result)
        return result  # type: ignore[return-value]

    def _update(self, obj: T, *, status_only: bool, metadata_only: bool = False) -> T:
        with self._lock:
            key = self._key(obj)
            cur = self._store.get(key)
            if cur is None:
                raise NotFoundError(f"{key} not found")
            if (
                obj.metadata.resource_version
                and obj.metadata.resource_version != cur.metadata.resource_version
            ):
                raise ConflictError(
                    f"{key}: resourceVersion {obj.metadata.resource_version} "
                    f"!= {cur.metadata.resource_version}"
                )
            stored = copy.deepcopy(cur)
            if status_only:
                stored.status = copy.deepcopy(obj.status)
            elif metadata_only:
                stored.metadata = copy.deepcopy(obj.metadata)
            else:
                stored = copy.deepcopy(obj)
            stored.metadata.resource_version = self._next_rv()
            self._store[key] = stored
            self._record("MODIFIED", copy.deepcopy(stored))
            return copy.deepcopy(stored)  # type: ignore[return-value]

    def update(self, obj: T) -> T:
        return self._update(obj, status_only=False)

    def patch_metadata(self, obj: T) -> T:
        return self._update(obj, status_only=False, metadata_only=True)

    def update_status(self, obj: T) -> T:
        return self._update(obj, status_only=True)

    def delete(self, cls: Type[T], name: str, namespace: str) -> None:
        kind = self._kind(cls)
        with self._lock:
            if (kind, namespace, name) not in self._store:
                raise NotFoundError(f"{cls.__name__} {namespace}/{name} not found")
            deleted = self._store.pop((kind, namespace, name))
            deleted.metadata.resource_version = self._next_rv()
            self._record("DELETED", copy.deepcopy(deleted))
            self._garbage_collect(deleted)

    def _garbage_collect(self, owner) -> None:
        """Cascade-delete objects owned (via ownerReferences) by ``owner``."""
        owner_uid = owner.metadata.uid
        doomed = []
        for key, obj in self._store.items():
            refs = getattr(obj.metadata, "owner_references", None) or []
            for ref in refs:
                if ref.uid == owner_uid or (
                    ref.kind == self._kind(owner)
                    and ref.name == owner.metadata.name
                    and obj.metadata.namespace == owner.metadata.namespace
                ):
                    doomed.append(key)
                    break
        for key in doomed:
            child = self._store.pop(key)
            child.metadata.resource_version = self._next_rv()
            self._record("DELETED", copy.deepcopy(child))
            self._garbage_collect(child)
