"""Whole-cluster lifecycle simulation: hundreds of pods scheduled, some
completing, some deleted, a scheduler restart mid-way — accounting must stay
exactly consistent with the surviving pods' annotations."""
from __future__ import annotations

import random

from elastic_gpu_scheduler_amd.controller.controller import Controller
from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
from elastic_gpu_scheduler_amd.scheduler.service import BindError, SchedulerRegistry
from tests.conftest import make_node, make_pod
from tests.test_controller import wait_until

GiB = 1024**3


def expected_usage(client, nodes):
    """Ground truth from the apiserver: per-node (core, mem) used by live
    assumed pods."""
    usage = {n: [0, 0] for n in nodes}
    for pod in client.list_pods():
        if obj.is_completed_pod(pod):
            continue
        node = obj.pod_node_name(pod)
        alloc = obj.parse_allocation(pod)
        if not node or node not in usage or alloc is None:
            continue
        req = obj.pod_gpu_request(pod)
        for c, cards in enumerate(alloc):
            u = req[c]
            for _ in cards:
                if u.gpu_count > 0:
                    usage[node][0] += 100
                    usage[node][1] += 288 * GiB
                else:
                    usage[node][0] += u.core
                    usage[node][1] += u.memory
    return usage


def observed_usage(sch, nodes):
    out = {}
    for n in nodes:
        devs = sch.state.node_devices(n)
        out[n] = [sum(d.core_total - d.core_avail for d in devs),
                  sum(d.mem_total - d.mem_avail for d in devs)]
    return out


def test_lifecycle_with_restart():
    rng = random.Random(42)
    client = FakeKubeClient()
    nodes = [f"n{i}" for i in range(4)]
    for n in nodes:
        client.add_node(make_node(n))
    registry = SchedulerRegistry(client)
    ctrl = Controller(client, registry, workers=2, resync_seconds=3600)
    ctrl.start()
    sch = registry.default
    live = []
    try:
        for round_no in range(3):
            # schedule a wave of pods
            for i in range(40):
                name = f"r{round_no}-p{i}"
                kind = rng.random()
                if kind < 0.15:
                    pod = make_pod(name, per_container=[{"pgpu": 1}])
                elif kind < 0.5:
                    pod = make_pod(name, core=rng.choice([10, 25, 50]),
                                   memory=rng.choice([16, 48, 96]) * GiB)
                else:
                    pod = make_pod(name, core=rng.choice([5, 15]),
                                   memory=8 * GiB)
                created = client.create_pod(pod)
                ok, _ = sch.assume(nodes, created)
                if not ok:
                    client.delete_pod("default", name)
                    continue
                target = rng.choice(ok)
                try:
                    sch.bind(target, created)
                    live.append(name)
                except BindError:
                    client.delete_pod("default", name)

            # complete / delete a third of the live pods
            rng.shuffle(live)
            drop = live[:len(live) // 3]
            live = live[len(live) // 3:]
            for name in drop:
                if rng.random() < 0.5:
                    client.set_pod_phase("default", name, "Succeeded")
                else:
                    client.delete_pod("default", name)
            assert wait_until(lambda: observed_usage(sch, nodes) ==
                              expected_usage(client, nodes), timeout=10), \
                (observed_usage(sch, nodes), expected_usage(client, nodes))

            if round_no == 1:
                # scheduler crash + restart: rebuild from annotations only
                ctrl.stop()
                registry2 = SchedulerRegistry(client)
                ctrl2 = Controller(client, registry2, workers=2,
                                   resync_seconds=3600)
                ctrl2.start()
                sch = registry2.default
                for n in nodes:
                    sch._ensure_node(n)
                assert observed_usage(sch, nodes) == \
                    expected_usage(client, nodes)
                ctrl = ctrl2
        assert live, "simulation scheduled nothing"
    finally:
        ctrl.stop()


def test_large_cluster_simulation():
    """1,000-node cluster slice: filter fan-outs at cluster scale through
    the service layer, placements land, accounting matches ground truth."""
    import random

    rng = random.Random(3)
    client = FakeKubeClient()
    nodes = [f"big{i}" for i in range(1000)]
    for n in nodes:
        client.add_node(make_node(n))
    registry = SchedulerRegistry(client)
    sch = registry.default
    bound = []
    for i in range(60):
        pod = client.create_pod(make_pod(f"p{i}", core=rng.choice([25, 50]),
                                         memory=48 * GiB))
        ok, failed = sch.assume(nodes, pod)
        assert len(ok) == 1000, len(ok)
        target = rng.choice(ok)
        sch.bind(target, pod)
        bound.append((f"p{i}", target))
    assert observed_usage(sch, nodes) == expected_usage(client, nodes)
    # release half, re-check
    for name, _ in bound[::2]:
        sch.forget_pod(client.get_pod("default", name))
        client.delete_pod("default", name)
    assert observed_usage(sch, nodes) == expected_usage(client, nodes)


def test_warm_start_recovery_at_scale():
    """Crash recovery at scale: 10,000 bound pods across 125 nodes are
    re-accounted from annotations alone in well under a minute, with
    accounting identical to the pre-crash state."""
    import time

    client = FakeKubeClient()
    nodes = [f"n{i}" for i in range(125)]
    for n in nodes:
        client.add_node(make_node(n))
    sch = SchedulerRegistry(client).default
    for i in range(10000):
        node = nodes[i % 125]
        pod = client.create_pod(make_pod(f"p{i}", core=10, memory=2 * GiB))
        sch.assume([node], pod)
        sch.bind(node, pod)

    t0 = time.time()
    sch2 = SchedulerRegistry(client).default
    elapsed = time.time() - t0
    accounted = sum(len(sch2.state.node_pods(n))
                    for n in sch2.state.node_names())
    assert accounted == 10000
    assert elapsed < 60, elapsed
    for n in nodes[::25]:
        a = [(d.core_avail, d.mem_avail) for d in sch.state.node_devices(n)]
        b = [(d.core_avail, d.mem_avail) for d in sch2.state.node_devices(n)]
        assert a == b
