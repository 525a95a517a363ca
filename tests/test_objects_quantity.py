"""Codec + quantity parsing: the wire-contract details."""
from __future__ import annotations

import pytest

from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.utils.quantity import parse_memory_bytes, parse_quantity
from tests.conftest import make_node, make_pod

GiB = 1024**3


class TestQuantity:
    def test_binary_suffixes(self):
        assert parse_quantity("64Gi") == 64 * GiB
        assert parse_quantity("512Mi") == 512 * 1024**2
        assert parse_quantity("1Ti") == 1024**4

    def test_decimal_suffixes(self):
        assert parse_quantity("1G") == 10**9
        assert parse_quantity("250M") == 250 * 10**6

    def test_plain_and_milli(self):
        assert parse_quantity("100") == 100
        assert parse_quantity(42) == 42
        assert parse_quantity("1500m") == 1

    def test_invalid(self):
        with pytest.raises(ValueError):
            parse_quantity("12XB")
        with pytest.raises(ValueError):
            parse_quantity("abc")

    def test_memory_bare_auto_heuristic(self):
        # bare small numbers are GiB (the reference test's `48` means 48 GB)
        assert parse_memory_bytes("48") == 48 * GiB
        assert parse_memory_bytes(64) == 64 * GiB
        # bare large numbers are bytes
        assert parse_memory_bytes(309237645312) == 309237645312
        # suffixed values are always exact
        assert parse_memory_bytes("64Gi") == 64 * GiB

    def test_memory_explicit_units(self):
        assert parse_memory_bytes("48", bare_unit="MiB") == 48 * 1024**2
        assert parse_memory_bytes("48", bare_unit="bytes") == 48
        assert parse_memory_bytes("48", bare_unit="GiB") == 48 * GiB


class TestPodHelpers:
    def test_is_gpu_pod(self):
        assert obj.is_gpu_pod(make_pod("p", core=10))
        assert obj.is_gpu_pod(make_pod("p", per_container=[{"pgpu": 1}]))
        assert not obj.is_gpu_pod({"spec": {"containers": [
            {"resources": {"requests": {"cpu": "1"}}}]}})

    def test_request_semantics(self):
        req = obj.pod_gpu_request(make_pod("p", core=250))
        assert req[0].gpu_count == 2  # core >= 100 -> whole cards

        req = obj.pod_gpu_request(make_pod("p", core=30, memory=64 * GiB))
        assert (req[0].gpu_count, req[0].core, req[0].memory) == (0, 30, 64 * GiB)

        req = obj.pod_gpu_request(make_pod("p", per_container=[{"pgpu": 3}]))
        assert req[0].gpu_count == 3

    def test_qgpu_resources_merge(self):
        pod = {"spec": {"containers": [{"name": "c", "resources": {"requests": {
            "elasticgpu.io/qgpu-core": "40",
            "elasticgpu.io/qgpu-memory": "32Gi"}}}]}}
        req = obj.pod_gpu_request(pod)
        assert (req[0].core, req[0].memory) == (40, 32 * GiB)

    def test_completed_pod(self):
        pod = make_pod("p", core=10)
        assert not obj.is_completed_pod(pod)
        pod["status"]["phase"] = "Failed"
        assert obj.is_completed_pod(pod)
        pod2 = make_pod("q", core=10)
        pod2["metadata"]["deletionTimestamp"] = "2026-01-01T00:00:00Z"
        assert obj.is_completed_pod(pod2)


class TestAnnotationCodec:
    def test_roundtrip(self):
        pod = make_pod("p", containers=2, core=20, memory=GiB)
        annotated = obj.apply_allocation(pod, [[0], [3]], node="n1", score=7.5)
        ann = annotated["metadata"]["annotations"]
        assert ann["elasticgpu.io/container-c0"] == "0"
        assert ann["elasticgpu.io/container-c1"] == "3"
        assert ann["elasticgpu.io/assumed"] == "true"
        assert ann["elasticgpu.io/scheduled-node"] == "n1"
        assert obj.parse_allocation(annotated) == [[0], [3]]
        assert obj.is_assumed(annotated)

    def test_multi_device_roundtrip(self):
        pod = make_pod("p", per_container=[{"pgpu": 4}])
        annotated = obj.apply_allocation(pod, [[0, 1, 2, 3]])
        assert annotated["metadata"]["annotations"][
            "elasticgpu.io/container-c0"] == "0,1,2,3"
        assert obj.parse_allocation(annotated) == [[0, 1, 2, 3]]

    def test_parse_missing_returns_none(self):
        assert obj.parse_allocation(make_pod("p", core=10)) is None

    def test_parse_empty_container_annotation(self):
        pod = make_pod("p", containers=2, core=20)
        annotated = obj.apply_allocation(pod, [[], [2]])
        assert obj.parse_allocation(annotated) == [[], [2]]


class TestNodeInventory:
    def test_from_allocatable(self):
        devs = obj.node_devices(make_node("n", cards=4, mem_per_card=288 * GiB))
        assert len(devs) == 4
        assert devs[0].mem_total == 288 * GiB

    def test_from_agent_annotation_heterogeneous(self):
        import json

        node = make_node("n", annotations={
            "elasticgpu.io/gpu-inventory": json.dumps({"cards": [
                {"core": 100, "memory_bytes": 288 * GiB},
                {"core": 100, "memory_bytes": 144 * GiB},
            ]})})
        devs = obj.node_devices(node)
        assert len(devs) == 2
        assert devs[1].mem_total == 144 * GiB  # heterogeneity preserved

    def test_from_amd_gpu_allocatable(self):
        node = {"metadata": {"name": "n"},
                "status": {"allocatable": {"amd.com/gpu": "8"}}}
        devs = obj.node_devices(node)
        assert len(devs) == 8
        assert devs[0].mem_total == 288 * GiB  # MI355X default

    def test_no_gpus(self):
        assert obj.node_devices({"metadata": {"name": "n"}, "status": {}}) == []

    def test_topology_annotation(self):
        import json

        hops = [[0, 1], [1, 0]]
        node = make_node("n", annotations={
            "elasticgpu.io/xgmi-topology": json.dumps({"hops": hops})})
        assert obj.node_topology(node) == hops
        assert obj.node_topology(make_node("n2")) == []

    def test_bad_annotations_fall_back(self):
        node = make_node("n", cards=2, annotations={
            "elasticgpu.io/gpu-inventory": "{not json",
            "elasticgpu.io/xgmi-topology": "also not"})
        assert len(obj.node_devices(node)) == 2  # falls back to allocatable
        assert obj.node_topology(node) == []

    def test_agent_inventory_out_of_order_cards(self):
        import json

        node = make_node("n", annotations={
            "elasticgpu.io/gpu-inventory": json.dumps({"cards": [
                {"index": 1, "core": 100, "memory_bytes": 144 * GiB},
                {"index": 0, "core": 100, "memory_bytes": 288 * GiB},
            ]})})
        devs = obj.node_devices(node)
        assert devs[0].mem_total == 288 * GiB  # position == physical index
        assert devs[1].mem_total == 144 * GiB
