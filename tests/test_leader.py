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


def test_lease_wire_format_is_microtime():
    """spec.renewTime/acquireTime must be RFC3339
    MicroTime STRINGS — a real apiserver rejects a unix-float write, and a
    kubelet-written lease must parse."""
    from elastic_gpu_scheduler_amd.k8s import leader as lmod

    client = FakeKubeClient()
    el = LeaderElector(client, "egs", "replica-a", namespace="ns")
    assert el._try_acquire_or_renew()
    lease = client.get_lease("ns", "egs")
    spec = lease["spec"]
    rt, at = spec["renewTime"], spec["acquireTime"]
    assert isinstance(rt, str) and isinstance(at, str)
    # RFC3339 with microsecond fraction and Z, e.g. 2026-09-14T10:11:12.123456Z
    import re
    pat = r"^\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2}\.\d{6}Z$"
    assert re.match(pat, rt), rt
    assert re.match(pat, at), at
    # round-trips to the time it was written at
    import time
    assert abs(lmod.parse_microtime(rt) - time.time()) < 5.0


def test_parse_microtime_kubelet_formats():
    """Formats real writers produce must parse; junk must not crash."""
    from elastic_gpu_scheduler_amd.k8s.leader import parse_microtime

    # client-go MicroTime (6-digit fraction)
    t1 = parse_microtime("2026-09-14T10:11:12.123456Z")
    assert t1 > 0
    # metav1.Time (no fraction)
    assert parse_microtime("2026-09-14T10:11:12Z") > 0
    # explicit offset
    assert abs(parse_microtime("2026-09-14T12:11:12.123456+02:00") - t1) < 1e-3
    # 9-digit (nanosecond) fraction truncates, 1-digit pads
    assert abs(parse_microtime("2026-09-14T10:11:12.123456789Z") - t1) < 1e-3
    assert parse_microtime("2026-09-14T10:11:12.1Z") > 0
    # legacy r1 numeric and garbage
    assert parse_microtime(1757844672.5) == 1757844672.5
    assert parse_microtime("not-a-time") == 0.0
    assert parse_microtime(None) == 0.0
    assert parse_microtime("") == 0.0


def test_takeover_respects_incumbent_lease_duration():
    """A candidate must honor the INCUMBENT's leaseDurationSeconds."""
    import time as _time

    from elastic_gpu_scheduler_amd.k8s.leader import format_microtime

    client = FakeKubeClient()
    # Incumbent wrote a 60 s lease (e.g. a differently-configured replica),
    # renewed 20 s ago.
    client.create_lease("ns", {
        "metadata": {"name": "egs", "namespace": "ns"},
        "spec": {"holderIdentity": "other",
                 "leaseDurationSeconds": 60,
                 "renewTime": format_microtime(_time.time() - 20)}})
    el = LeaderElector(client, "egs", "replica-b", namespace="ns",
                       lease_duration=15.0)
    # 20 s > our 15 s config, but < the incumbent's 60 s: NOT acquirable.
    assert not el._try_acquire_or_renew()
