"""Node agent: inventory/topology parsing from canned tool output (CPU),
publish flow against the fake apiserver."""
from __future__ import annotations

import json

from elastic_gpu_scheduler_amd.agent import inventory as inv
from elastic_gpu_scheduler_amd.agent import topology as topo
from elastic_gpu_scheduler_amd.agent.agent import NodeAgent
from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
from elastic_gpu_scheduler_amd._native import core

GiB = 1024**3

ROCM_SMI_VRAM = json.dumps({
    "card0": {"VRAM Total Memory (B)": "309237645312",
              "VRAM Total Used Memory (B)": "12345"},
    "card1": {"VRAM Total Memory (B)": "309237645312",
              "VRAM Total Used Memory (B)": "0"},
})

SHOWTOPOHOPS = """
========================= ROCm System Management Interface =========================
=========================== Hops between two GPUs ==================================
       GPU0   GPU1   GPU2   GPU3
GPU0   0      1      1      2
GPU1   1      0      1      2
GPU2   1      1      0      1
GPU3   2      2      1      0
====================================================================================
"""


def test_parse_rocm_smi_vram():
    cards = inv.parse_rocm_smi_vram(ROCM_SMI_VRAM)
    assert len(cards) == 2
    assert cards[0]["index"] == 0
    assert cards[0]["memory_bytes"] == 309237645312
    assert cards[0]["core"] == 100


def test_parse_rocm_smi_vram_garbage():
    assert inv.parse_rocm_smi_vram("not json") == []
    assert inv.parse_rocm_smi_vram("{}") == []


def test_parse_showtopohops():
    m = topo.parse_showtopohops(SHOWTOPOHOPS)
    assert len(m) == 4
    assert m[0][0] == 0
    assert m[0][1] == 1
    assert m[0][3] == 2
    assert m[3][2] == 1


def test_parse_showtopohops_garbage():
    assert topo.parse_showtopohops("") == []
    assert topo.parse_showtopohops("random\ntext") == []


def test_default_hive():
    m = topo.default_hive(8)
    assert m[0][0] == 0 and m[0][7] == 1 and len(m) == 8


def test_agent_publish_roundtrips_into_scheduler_inventory(monkeypatch):
    """agent.publish() -> node annotations -> scheduler node_devices must
    agree with what the agent discovered."""
    cards = [{"index": i, "name": "MI355X", "gcn_arch": "gfx950",
              "memory_bytes": 288 * GiB, "core": 100, "source": "test"}
             for i in range(8)]
    monkeypatch.setattr(inv, "discover", lambda prefer="auto": cards)
    monkeypatch.setattr(topo, "discover",
                        lambda n, prefer="auto": topo.default_hive(n))

    client = FakeKubeClient()
    client.add_node({"metadata": {"name": "gpu-node"}, "status": {}})
    agent = NodeAgent("gpu-node", client)
    ann = agent.publish()
    node = client.get_node("gpu-node")
    assert node["metadata"]["annotations"] == ann

    devs = obj.node_devices(node)
    assert len(devs) == 8
    assert devs[0].mem_total == 288 * GiB
    hops = obj.node_topology(node)
    assert len(hops) == 8 and hops[0][1] == 1


def test_agent_allocatable(monkeypatch):
    cards = [{"index": i, "memory_bytes": 288 * GiB, "core": 100}
             for i in range(8)]
    monkeypatch.setattr(inv, "discover", lambda prefer="auto": cards)
    monkeypatch.setattr(topo, "discover",
                        lambda n, prefer="auto": topo.default_hive(n))
    agent = NodeAgent("n")
    alloc = agent.allocatable()
    assert alloc["elasticgpu.io/gpu-core"] == "800"
    assert alloc["elasticgpu.io/gpu-memory"] == str(8 * 288 * GiB)
    assert alloc["amd.com/gpu"] == "8"

    node = agent.node_object()
    devs = obj.node_devices(node)
    assert len(devs) == 8


def test_discover_empty_on_cpu_only_box():
    # On this CPU-only container every real source returns [] -> empty.
    import torch

    if torch.cuda.is_available():
        return  # running on a GPU box: covered by gpu-marked tests
    assert inv.discover() == []
    assert topo.discover(8) == topo.default_hive(8)


def test_parse_compute_partition():
    payload = json.dumps({"card0": {"Compute Partition": "SPX"},
                          "card1": {"Compute Partition": "CPX"},
                          "system": {"driver": "x"}})
    assert inv.parse_compute_partition(payload) == {0: "SPX", 1: "CPX"}
    assert inv.parse_compute_partition("junk") == {}


def test_parse_amd_smi_static():
    payload = json.dumps([
        {"gpu": 0, "asic": {"market_name": "MI355X",
                            "target_graphics_version": "gfx950"},
         "vram": {"size": {"value": 294912, "unit": "MB"}}},
        {"gpu": 1, "vram": {"size": {"value": 288, "unit": "GB"}}},
    ])
    cards = inv.parse_amd_smi_static(payload)
    assert len(cards) == 2
    assert cards[0]["memory_bytes"] == 294912 * 1024**2
    assert cards[0]["name"] == "MI355X"
    assert cards[1]["memory_bytes"] == 288 * 1024**3
    assert inv.parse_amd_smi_static("garbage") == []


def test_agent_publish_includes_allocatable(monkeypatch):
    cards = [{"index": i, "memory_bytes": 288 * GiB, "core": 100}
             for i in range(8)]
    monkeypatch.setattr(inv, "discover", lambda prefer="auto": cards)
    monkeypatch.setattr(topo, "discover",
                        lambda n, prefer="auto": topo.default_hive(n))
    client = FakeKubeClient()
    client.add_node({"metadata": {"name": "gpu-node"}, "status": {}})
    NodeAgent("gpu-node", client).publish()
    node = client.get_node("gpu-node")
    assert node["status"]["allocatable"]["elasticgpu.io/gpu-core"] == "800"
    assert node["status"]["capacity"]["amd.com/gpu"] == "8"


def test_parse_amd_smi_topology():
    payload = json.dumps([
        {"gpu": 0, "links": [
            {"gpu": 1, "link_type": "XGMI", "num_hops": 1},
            {"gpu": 2, "link_type": "PCIE", "num_hops": 2}]},
        {"gpu": 1, "links": [
            {"gpu": 0, "link_type": "XGMI", "num_hops": 1},
            {"gpu": 2, "link_type": "XGMI", "num_hops": 2}]},
        {"gpu": 2, "links": []},
    ])
    m = topo.parse_amd_smi_topology(payload)
    assert m[0][1] == 1          # direct xGMI
    assert m[0][2] == 3          # PCIe link counts as routed
    assert m[1][2] == 2          # 2-hop xGMI
    assert m[2][0] == 3          # unreported direction defaults to routed
    assert [m[i][i] for i in range(3)] == [0, 0, 0]
    assert topo.parse_amd_smi_topology("junk") == []
    assert topo.parse_amd_smi_topology("[]") == []


def test_publish_with_health_excludes_sick_cards(monkeypatch):
    """A card failing the HBM health gate is published as a ZERO-CAPACITY
    placeholder: the scheduler stops placing onto it, but every other card
    keeps its physical index (list position == physical card index is the
    contract the device-index annotations rely on)."""
    cards = [{"index": i, "memory_bytes": 288 * GiB, "core": 100}
             for i in range(4)]
    monkeypatch.setattr(inv, "discover", lambda prefer="auto": cards)
    monkeypatch.setattr(topo, "discover",
                        lambda n, prefer="auto": topo.default_hive(n))
    client = FakeKubeClient()
    client.add_node({"metadata": {"name": "gpu-node"}, "status": {}})
    agent = NodeAgent("gpu-node", client)
    monkeypatch.setattr(NodeAgent, "health_check",
                        lambda self, mib=256, iters=5: [
                            {"index": 0, "hbm_gbps": 6100.0, "healthy": True},
                            {"index": 1, "hbm_gbps": 240.0, "healthy": False},
                            {"index": 2, "hbm_gbps": 6050.0, "healthy": True},
                            {"index": 3, "hbm_gbps": 6200.0, "healthy": True}])
    out = agent.publish_with_health()
    assert out["sick"] == [1]
    node = client.get_node("gpu-node")
    devs = obj.node_devices(node)
    # All four physical slots present; the sick one is unschedulable.
    assert len(devs) == 4
    assert not devs[1].schedulable() and not devs[1].whole_free()
    assert [d.schedulable() for d in devs] == [True, False, True, True]
    # Allocatable counts only healthy capacity.
    assert node["status"]["allocatable"]["elasticgpu.io/gpu-core"] == "300"
    assert node["status"]["allocatable"]["amd.com/gpu"] == "3"
    # A whole-card pod never lands on the sick card: its index is skipped.
    feasible, option, _ = core.search_placement(
        devs, [core.GPUUnit(gpu_count=3)], "binpack")
    assert feasible and sorted(option.allocated[0]) == [0, 2, 3]
