"""Fault injection: a flaky apiserver (random 500s / conflicts) must never
corrupt accounting — failed binds roll back, retries converge, and the final
state always matches the apiserver ground truth. (The reference has no fault
injection at all — SURVEY.md §5.)"""
from __future__ import annotations

import random

import pytest

from elastic_gpu_scheduler_amd.k8s.client import (ConflictError, FakeKubeClient,
                                                  NotFoundError)
from elastic_gpu_scheduler_amd.scheduler.service import (BindError,
                                                         GPUUnitScheduler)
from tests.conftest import make_node, make_pod
from tests.test_lifecycle_simulation import expected_usage, observed_usage

GiB = 1024**3


class FlakyKubeClient(FakeKubeClient):
    """Injects failures into the write paths with a given probability."""

    def __init__(self, rng, fail_rate=0.3):
        super().__init__()
        self.rng = rng
        self.fail_rate = fail_rate
        self.injected = 0

    def _maybe_fail(self):
        if self.rng.random() < self.fail_rate:
            self.injected += 1
            if self.rng.random() < 0.5:
                raise RuntimeError("injected: apiserver 500")
            raise ConflictError("injected: conflict")

    def update_pod(self, pod):
        self._maybe_fail()
        return super().update_pod(pod)

    def bind_pod(self, namespace, name, node):
        self._maybe_fail()
        return super().bind_pod(namespace, name, node)


def test_flaky_apiserver_never_corrupts_accounting():
    rng = random.Random(7)
    client = FlakyKubeClient(rng, fail_rate=0.35)
    nodes = ["n0", "n1"]
    for n in nodes:
        client.add_node(make_node(n))
    sch = GPUUnitScheduler(client)

    scheduled = 0
    for i in range(120):
        pod = client.create_pod(make_pod(f"p{i}", core=20, memory=16 * GiB))
        ok, _ = sch.assume(nodes, pod)
        if not ok:
            continue
        target = rng.choice(ok)
        # like kube-scheduler: retry a failed bind a few times
        for attempt in range(4):
            try:
                sch.bind(target, client.get_pod("default", f"p{i}"))
                scheduled += 1
                break
            except (BindError, RuntimeError, ConflictError):
                continue
        else:
            client.delete_pod("default", f"p{i}")

    assert client.injected > 10, "fault injection never fired"
    assert scheduled > 20, "nothing survived the faults"
    # accounting must exactly match the surviving annotations
    assert observed_usage(sch, nodes) == expected_usage(client, nodes)


def test_bind_all_writes_down_leaves_clean_state():
    class DownClient(FakeKubeClient):
        def update_pod(self, pod):
            raise RuntimeError("apiserver down")

    client = DownClient()
    client.add_node(make_node("n0", cards=1))
    sch = GPUUnitScheduler(client)
    for i in range(5):
        pod = client.create_pod(make_pod(f"p{i}", core=30, memory=GiB))
        sch.assume(["n0"], pod)
        with pytest.raises(RuntimeError):
            sch.bind("n0", pod)
    d = sch.state.node_devices("n0")[0]
    assert d.core_avail == 100 and d.mem_avail == 288 * GiB


def test_node_vanishes_mid_schedule():
    client = FakeKubeClient()
    client.add_node(make_node("n0"))
    sch = GPUUnitScheduler(client)
    pod = client.create_pod(make_pod("p", core=30, memory=GiB))
    ok, _ = sch.assume(["n0"], pod)
    assert ok == ["n0"]
    # node deleted between filter and bind
    with client._mu:
        del client._nodes["n0"]
    sch.invalidate_node("n0")
    with pytest.raises(BindError):
        sch.bind("n0", client.get_pod("default", "p"))


def test_get_pod_races_are_typed():
    client = FakeKubeClient()
    with pytest.raises(NotFoundError):
        client.get_pod("default", "ghost")
