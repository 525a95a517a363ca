"""Leader election: exactly one leader; takeover after the holder stops
renewing; graceful release hands over immediately."""
from __future__ import annotations

import threading
import time

from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
from elastic_gpu_scheduler_amd.k8s.leader import LeaderElector


def run_elector(elector, events, label):
    t = threading.Thread(
        target=elector.run,
        args=(lambda: events.append(f"{label}-started"),
              lambda: events.append(f"{label}-stopped")),
        daemon=True)
    t.start()
    return t


def test_single_leader_and_takeover():
    client = FakeKubeClient()
    events = []
    # generous lease vs renew ratio so a loaded CI machine cannot make the
    # holder miss renewals spuriously
    a = LeaderElector(client, "egs", "a", lease_duration=3.0,
                      renew_period=0.1, retry_period=0.1)
    b = LeaderElector(client, "egs", "b", lease_duration=3.0,
                      renew_period=0.1, retry_period=0.1)
    ta = run_elector(a, events, "a")
    deadline = time.time() + 5
    while time.time() < deadline and not a.is_leader:
        time.sleep(0.02)
    tb = run_elector(b, events, "b")
    time.sleep(0.5)
    assert a.is_leader and not b.is_leader
    assert events == ["a-started"]

    # a dies abruptly (no release): b takes over after the lease expires
    a._stop.set()
    ta.join(timeout=5)
    # a's graceful release may or may not have happened depending on timing;
    # either way b must eventually lead (after <= lease_duration)
    deadline = time.time() + 15
    while time.time() < deadline and not b.is_leader:
        time.sleep(0.05)
    assert b.is_leader
    assert "b-started" in events
    b.stop()
    tb.join(timeout=2)


def test_graceful_release_hands_over_fast():
    client = FakeKubeClient()
    events = []
    a = LeaderElector(client, "egs", "a", lease_duration=5.0,
                      renew_period=0.1, retry_period=0.05)
    ta = run_elector(a, events, "a")
    deadline = time.time() + 2
    while time.time() < deadline and not a.is_leader:
        time.sleep(0.02)
    assert a.is_leader
    a.stop()
    ta.join(timeout=2)

    # despite the 5 s lease, a released -> b acquires immediately
    b = LeaderElector(client, "egs", "b", lease_duration=5.0,
                      renew_period=0.1, retry_period=0.05)
    tb = run_elector(b, events, "b")
    deadline = time.time() + 2
    while time.time() < deadline and not b.is_leader:
        time.sleep(0.02)
    assert b.is_leader
    b.stop()
    tb.join(timeout=2)


def test_lease_transitions_counted():
    client = FakeKubeClient()
    a = LeaderElector(client, "egs", "a", lease_duration=0.3,
                      renew_period=0.05, retry_period=0.05)
    assert a._try_acquire_or_renew()
    time.sleep(0.4)  # expire
    b = LeaderElector(client, "egs", "b", lease_duration=0.3,
                      renew_period=0.05, retry_period=0.05)
    assert b._try_acquire_or_renew()
    lease = client.get_lease("kube-system", "egs")
    assert lease["spec"]["holderIdentity"] == "b"
    assert lease["spec"]["leaseTransitions"] == 1
