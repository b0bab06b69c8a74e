"""Lease-based leader election (wva_amd/controller/leader.py): acquire,
contention, expiry takeover, renewal loss, graceful release — against
the in-memory client with a controllable clock, plus one pass over the
HTTP client + stub API server."""

import datetime
import threading
import time

import pytest

from wva_amd.controller.leader import LEASE_NAME, LeaseElector
from wva_amd.kube import InMemoryKubeClient, Lease


class FakeClock:
    def __init__(self):
        self.now = datetime.datetime(2026, 1, 1, tzinfo=datetime.timezone.utc)

    def __call__(self):
        return self.now

    def advance(self, seconds):
        self.now += datetime.timedelta(seconds=seconds)


def make_elector(client, clock, identity, **kw):
    kw.setdefault("lease_duration", 15.0)
    kw.setdefault("retry_period", 0.05)
    return LeaseElector(client, identity=identity, clock=clock, **kw)


class TestLeaseElection:
    def test_first_acquire_creates_lease(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        assert a.try_acquire_or_renew()
        lease = client.get(Lease, LEASE_NAME, a.namespace)
        assert lease.spec.holder_identity == "a"
        assert lease.spec.lease_duration_seconds == 15
        assert lease.spec.lease_transitions == 0

    def test_contender_blocked_while_fresh(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        b = make_elector(client, clock, "b")
        assert a.try_acquire_or_renew()
        clock.advance(14.0)  # still inside the 15 s lease
        assert not b.try_acquire_or_renew()
        lease = client.get(Lease, LEASE_NAME, a.namespace)
        assert lease.spec.holder_identity == "a"

    def test_takeover_after_expiry(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        b = make_elector(client, clock, "b")
        assert a.try_acquire_or_renew()
        clock.advance(16.0)  # leader died: no renewals
        assert b.try_acquire_or_renew()
        lease = client.get(Lease, LEASE_NAME, b.namespace)
        assert lease.spec.holder_identity == "b"
        assert lease.spec.lease_transitions == 1
        # and the old leader's next renewal must FAIL
        assert not a.try_acquire_or_renew()

    def test_renew_keeps_leadership(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        b = make_elector(client, clock, "b")
        assert a.try_acquire_or_renew()
        for _ in range(5):
            clock.advance(10.0)
            assert a.try_acquire_or_renew()  # renewal refreshes renewTime
            assert not b.try_acquire_or_renew()

    def test_acquire_blocks_then_wins_and_detects_loss(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        assert a.acquire(timeout=1.0)
        assert a.is_leader()
        lost = threading.Event()
        a.on_lost = lost.set
        # steal the lease out from under the renew loop
        clock.advance(16.0)
        b = make_elector(client, clock, "b")
        assert b.try_acquire_or_renew()
        assert lost.wait(timeout=3.0)
        assert not a.is_leader()
        a.release()
        b.release()

    def test_acquire_timeout_when_held(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        assert a.try_acquire_or_renew()
        b = make_elector(client, clock, "b")
        # FakeClock never moves during acquire, so pin the deadline by
        # advancing it from a side thread
        t = threading.Thread(target=lambda: (time.sleep(0.2), clock.advance(10.0)), daemon=True)
        t.start()
        assert not b.acquire(timeout=5.0)

    def test_graceful_release_enables_instant_succession(self):
        client, clock = InMemoryKubeClient(), FakeClock()
        a = make_elector(client, clock, "a")
        assert a.acquire(timeout=1.0)
        a.release()
        b = make_elector(client, clock, "b")
        assert b.try_acquire_or_renew()  # no 15 s wait
        lease = client.get(Lease, LEASE_NAME, b.namespace)
        assert lease.spec.holder_identity == "b"
        assert lease.spec.lease_transitions == 1


class TestLeaseOverHTTP:
    def test_lease_crud_and_election_roundtrip(self):
        import uvicorn

        from wva_amd.kube.http_client import HTTPKubeClient
        from wva_amd.kube.stub_server import create_stub_api_server

        app, _ = create_stub_api_server()
        server = uvicorn.Server(
            uvicorn.Config(app, host="127.0.0.1", port=0, log_level="error")
        )
        thread = threading.Thread(target=server.run, daemon=True)
        thread.start()
        for _ in range(200):
            if server.started:
                break
            time.sleep(0.05)
        assert server.started
        port = server.servers[0].sockets[0].getsockname()[1]
        try:
            client = HTTPKubeClient(base_url=f"http://127.0.0.1:{port}")
            clock = FakeClock()
            a = make_elector(client, clock, "a")
            b = make_elector(client, clock, "b")
            assert a.try_acquire_or_renew()
            assert not b.try_acquire_or_renew()  # AlreadyExists -> 409 -> Conflict
            clock.advance(16.0)
            assert b.try_acquire_or_renew()
            lease = client.get(Lease, LEASE_NAME, b.namespace)
            assert lease.spec.holder_identity == "b"
        finally:
            server.should_exit = True
            thread.join(timeout=5.0)
