"""Lease-based leader election over the Kubernetes API.

The reference elects through controller-runtime with
``LeaderElectionID: "72dd1cf1.llm-d.ai"`` (cmd/main.go) — a
coordination.k8s.io/v1 Lease object renewed by the active manager.
This is the same protocol, implemented directly against the KubeClient
(so it runs identically over the in-memory fake, the stub API server,
and a real cluster):

- acquire: create the Lease (AlreadyExists = somebody won the race), or
  take over a lease whose ``renewTime`` is older than
  ``leaseDurationSeconds`` (bumping ``leaseTransitions``);
- renew: re-update ``renewTime`` every ``retry_period`` while leading;
  any failed renewal (conflict = stolen, transport error) drops
  leadership — the caller decides whether to exit (the reference's
  manager terminates the process, __main__ does the same);
- release: on graceful stop the holder is cleared so a successor
  acquires immediately instead of waiting out the lease.

The flock-based election in __main__ remains for the in-memory backend
(single-node dev), Lease election is used with ``--kube-backend
in-cluster``.
"""

from __future__ import annotations

import datetime
import socket
import threading
import uuid
from typing import Callable, Optional

from ..api.v1alpha1.types import ObjectMeta
from ..kube import ConflictError, KubeClient, Lease, LeaseSpec, NotFoundError
from .constants import CONTROLLER_NAMESPACE
from .logger import log

# same election id as the reference's manager (cmd/main.go)
LEASE_NAME = "72dd1cf1.llm-d.ai"


def _utcnow() -> datetime.datetime:
    return datetime.datetime.now(datetime.timezone.utc)


class LeaseElector:
    """client-go style leader election on a coordination/v1 Lease."""

    def __init__(
        self,
        client: KubeClient,
        namespace: str = CONTROLLER_NAMESPACE,
        name: str = LEASE_NAME,
        identity: Optional[str] = None,
        lease_duration: float = 15.0,
        retry_period: float = 2.0,
        clock: Callable[[], datetime.datetime] = _utcnow,
    ) -> None:
        self.client = client
        self.namespace = namespace
        self.name = name
        self.identity = identity or f"{socket.gethostname()}_{uuid.uuid4().hex[:8]}"
        self.lease_duration = lease_duration
        self.retry_period = retry_period
        self._clock = clock
        self._stop = threading.Event()
        self._leading = threading.Event()
        self._renew_thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------ state
    def is_leader(self) -> bool:
        return self._leading.is_set()

    # ---------------------------------------------------------------- attempt
    def try_acquire_or_renew(self) -> bool:
        """One acquisition/renewal attempt; True while this identity holds."""
        now = self._clock()
        try:
            lease = self.client.get(Lease, self.name, self.namespace)
        except NotFoundError:
            fresh = Lease(
                metadata=ObjectMeta(name=self.name, namespace=self.namespace),
                spec=LeaseSpec(
                    holder_identity=self.identity,
                    lease_duration_seconds=int(self.lease_duration),
                    acquire_time=now,
                    renew_time=now,
                    lease_transitions=0,
                ),
            )
            try:
                self.client.create(fresh)
                return True
            except ConflictError:
                return False  # lost the creation race
        holder = lease.spec.holder_identity
        if holder and holder != self.identity:
            renew = lease.spec.renew_time
            duration = lease.spec.lease_duration_seconds or int(self.lease_duration)
            if renew is not None and (now - renew).total_seconds() < duration:
                return False  # held by a live leader
            lease.spec.lease_transitions += 1
            lease.spec.acquire_time = now
            log.info(
                "taking over expired leader lease",
                lease=self.name,
                previous_holder=holder,
            )
        elif not holder:
            lease.spec.lease_transitions += 1
            lease.spec.acquire_time = now
        lease.spec.holder_identity = self.identity
        lease.spec.renew_time = now
        lease.spec.lease_duration_seconds = int(self.lease_duration)
        try:
            self.client.update(lease)
            return True
        except ConflictError:
            return False  # concurrent writer won

    # ---------------------------------------------------------------- acquire
    def acquire(self, timeout: Optional[float] = None) -> bool:
        """Block until this identity leads (starting the renew loop) or
        timeout/stop; False when leadership was not obtained."""
        deadline = None if timeout is None else self._clock().timestamp() + timeout
        while not self._stop.is_set():
            try:
                won = self.try_acquire_or_renew()
            except Exception as e:
                log.error("leader election attempt failed", error=str(e))
                won = False
            if won:
                self._leading.set()
                log.info("acquired leader lease", lease=self.name, identity=self.identity)
                self._renew_thread = threading.Thread(
                    target=self._renew_loop, daemon=True
                )
                self._renew_thread.start()
                return True
            if deadline is not None and self._clock().timestamp() >= deadline:
                return False
            self._stop.wait(self.retry_period)
        return False

    def _renew_loop(self) -> None:
        while not self._stop.is_set():
            self._stop.wait(self.retry_period)
            if self._stop.is_set():
                return
            try:
                still = self.try_acquire_or_renew()
            except Exception as e:
                log.error("lease renewal failed", error=str(e))
                still = False
            if not still:
                self._leading.clear()
                log.error(
                    "lost leader lease", lease=self.name, identity=self.identity
                )
                if self.on_lost is not None:
                    self.on_lost()
                return

    # callback invoked from the renew thread when leadership is lost;
    # __main__ terminates the process here, mirroring controller-runtime
    on_lost: Optional[Callable[[], None]] = None

    # ---------------------------------------------------------------- release
    def release(self) -> None:
        """Stop renewing and clear the holder for instant succession."""
        self._stop.set()
        if self._renew_thread is not None:
            self._renew_thread.join(timeout=self.retry_period + 1.0)
        if not self._leading.is_set():
            return
        self._leading.clear()
        try:
            lease = self.client.get(Lease, self.name, self.namespace)
            if lease.spec.holder_identity == self.identity:
                lease.spec.holder_identity = ""
                self.client.update(lease)
        except Exception as e:  # best effort — expiry recovers anyway
            log.debug("lease release failed", error=str(e))
