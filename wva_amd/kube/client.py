"""KubeClient protocol and the in-memory (envtest-analog) implementation.

The fake implements the API-machinery behaviors the controller depends on:
typed CRUD, a separate status subresource, resourceVersion bumping with
conflict detection on stale updates, deep-copy isolation (readers never
alias store state), ownerReference cascade deletion (GC), and create-event
hooks for the reconciler's Create-only event filter
(/root/reference/internal/controller/variantautoscaling_controller.go:473-486).
"""

from __future__ import annotations

import copy
import datetime
import itertools
import threading
from typing import Callable, Dict, List, Optional, Protocol, Tuple, Type, TypeVar

from ..api.v1alpha1.types import VariantAutoscaling
from .errors import ConflictError, NotFoundError
from .objects import ConfigMap, Deployment, Lease

T = TypeVar("T")

_KINDS = {
    VariantAutoscaling: "VariantAutoscaling",
    ConfigMap: "ConfigMap",
    Deployment: "Deployment",
    Lease: "Lease",
}


class KubeClient(Protocol):
    def get(self, cls: Type[T], name: str, namespace: str) -> T: ...

    def list(self, cls: Type[T], namespace: Optional[str] = None) -> List[T]: ...

    def create(self, obj: T) -> T: ...

    def update(self, obj: T) -> T: ...

    def patch_metadata(self, obj: T) -> T: ...

    def update_status(self, obj: T) -> T: ...

    def delete(self, cls: Type[T], name: str, namespace: str) -> None: ...


class InMemoryKubeClient:
    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._store: Dict[Tuple[str, str, str], object] = {}
        self._rv = itertools.count(1)
        self._uid = itertools.count(1)
        self._create_hooks: List[Callable[[object], None]] = []

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

    # -- CRUD ---------------------------------------------------------------
    def get(self, cls: Type[T], name: str, namespace: str) -> T:
        with self._lock:
            obj = self._store.get((self._kind(cls), namespace, name))
            if obj is None:
                raise NotFoundError(f"{cls.__name__} {namespace}/{name} not found")
            return copy.deepcopy(obj)  # type: ignore[return-value]

    def list(self, cls: Type[T], namespace: Optional[str] = None) -> List[T]:
        kind = self._kind(cls)
        with self._lock:
            out = [
                copy.deepcopy(o)
                for (k, ns, _), o in sorted(self._store.items())
                if k == kind and (namespace is None or ns == namespace)
            ]
        return out  # type: ignore[return-value]

    def create(self, obj: T) -> T:
        with self._lock:
            key = self._key(obj)
            if key in self._store:
                # AlreadyExists is a Conflict in this package's error
                # taxonomy (the API server's 409) — leader election's
                # create race depends on it
                raise ConflictError(f"{key} already exists")
            stored = copy.deepcopy(obj)
            stored.metadata.resource_version = next(self._rv)
            stored.metadata.uid = stored.metadata.uid or f"uid-{next(self._uid)}"
            stored.metadata.creation_timestamp = datetime.datetime.now(
                datetime.timezone.utc
            )
            self._store[key] = stored
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
            stored.metadata.resource_version = next(self._rv)
            self._store[key] = stored
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
            self._garbage_collect(child)
