"""MI355X node agent: publishes inventory + xGMI topology, verifies health
and placement.

The reference repo relies on a sibling project (elastic-gpu-agent, NVML +
nvidia-docker) for the node side; this module is the MI355X-native
equivalent's control logic:

  * publish() — writes the per-card inventory and the xGMI hop matrix onto
    the Node object as annotations (elasticgpu.io/gpu-inventory,
    elasticgpu.io/xgmi-topology); the scheduler's node cache consumes them
    (k8s.objects.node_devices / node_topology). The whole-card-count
    extended resource itself is advertised by the ROCm k8s device plugin
    (amd.com/gpu); gpu-core/gpu-memory allocatable are derived from the
    published inventory;
  * health_check() — runs the HIP HBM-bandwidth probe per card and flags
    cards below a threshold (a sick HBM stack or wrong partition mode);
  * verify_placement() — stamps a pod-unique tag on the cards a bound pod
    was assigned, proving the scheduler/device-plugin contract held
    (BASELINE.json: "rocprof/amd-smi spot-check that containers land on the
    chosen card indices" — this is the programmatic version).
"""
from __future__ import annotations

import json
import logging
from typing import Any, Dict, List, Optional

from elastic_gpu_scheduler_amd.agent import inventory as inv
from elastic_gpu_scheduler_amd.agent import topology as topo
from elastic_gpu_scheduler_amd.k8s.client import KubeClient
from elastic_gpu_scheduler_amd.utils import types as t

log = logging.getLogger("egs.agent")

# A healthy MI355X streams HBM3E far above 1 TB/s (measured ceiling ~6.3 TB/s
# for a float4 copy); anything below this is a sick card or a misconfigured
# partition and should not be scheduled onto.
HBM_HEALTH_THRESHOLD_GBPS = 1000.0


class NodeAgent:
    def __init__(self, node_name: str, client: Optional[KubeClient] = None,
                 prefer_source: str = "auto") -> None:
        self.node_name = node_name
        self.client = client
        self.prefer_source = prefer_source

    # -- discovery --

    def snapshot(self) -> Dict[str, Any]:
        cards = inv.discover(self.prefer_source)
        hops = topo.discover(len(cards), self.prefer_source)
        return {"cards": cards, "topology": {"hops": hops}}

    def annotations(self) -> Dict[str, str]:
        snap = self.snapshot()
        return {
            t.ANNOTATION_NODE_INVENTORY: json.dumps({"cards": snap["cards"]}),
            t.ANNOTATION_NODE_TOPOLOGY: json.dumps(snap["topology"]),
        }

    def allocatable(self) -> Dict[str, str]:
        """The elasticgpu.io allocatable a device plugin would advertise for
        this node, derived from the live inventory."""
        snap = self.snapshot()
        cards = snap["cards"]
        total_core = sum(c.get("core", t.GPU_CORE_EACH_CARD) for c in cards)
        total_mem = sum(int(c.get("memory_bytes", 0)) for c in cards)
        return {
            t.RESOURCE_GPU_CORE: str(total_core),
            t.RESOURCE_GPU_MEMORY: str(total_mem),
            t.RESOURCE_AMD_GPU: str(len(cards)),
        }

    def publish(self) -> Dict[str, str]:
        """Publish inventory/topology annotations AND the elasticgpu.io
        allocatable quantities onto the Node (the role the reference
        delegates to its device plugin's node updates)."""
        if self.client is None:
            raise RuntimeError("NodeAgent.publish needs a KubeClient")
        ann = self.annotations()
        self.client.patch_node_annotations(self.node_name, ann)
        try:
            self.client.patch_node_allocatable(self.node_name,
                                               self.allocatable())
        except NotImplementedError:
            pass  # client without status-subresource support
        return ann

    def publish_with_health(self, mib: int = 256, iters: int = 5
                            ) -> Dict[str, Any]:
        """Publish inventory with unhealthy cards EXCLUDED (HBM bandwidth
        below threshold): the scheduler stops placing onto sick cards at the
        next node-cache refresh. Returns {"published": ann, "sick": [...]}.
        """
        if self.client is None:
            raise RuntimeError("publish_with_health needs a KubeClient")
        report = self.health_check(mib=mib, iters=iters)
        sick = [r["index"] for r in report if not r["healthy"]]
        ann = self.annotations()
        inv = json.loads(ann[t.ANNOTATION_NODE_INVENTORY])
        # Sick cards stay in the list as zero-capacity placeholders so list
        # position keeps equalling the PHYSICAL card index — dropping an
        # entry would shift every later card's index and the scheduler's
        # annotations (consumed as physical indexes by the device plugin)
        # would map pods onto the wrong card, including the sick one.
        # Zero-capacity devices are never schedulable (csrc/core/types.h
        # Device::schedulable).
        inv["cards"] = [
            ({**c, "core": 0, "memory_bytes": 0, "sick": True}
             if c["index"] in sick else c)
            for c in inv["cards"]
        ]
        ann[t.ANNOTATION_NODE_INVENTORY] = json.dumps(inv)
        self.client.patch_node_annotations(self.node_name, ann)
        try:
            alloc = {
                t.RESOURCE_GPU_CORE: str(sum(c.get("core", 100)
                                             for c in inv["cards"])),
                t.RESOURCE_GPU_MEMORY: str(sum(int(c.get("memory_bytes", 0))
                                               for c in inv["cards"])),
                t.RESOURCE_AMD_GPU: str(sum(1 for c in inv["cards"]
                                            if not c.get("sick"))),
            }
            self.client.patch_node_allocatable(self.node_name, alloc)
        except NotImplementedError:
            pass
        return {"published": ann, "sick": sick}

    def node_object(self) -> Dict[str, Any]:
        """A complete Node object for offline/bench use (fake apiserver)."""
        return {
            "metadata": {"name": self.node_name, "annotations": self.annotations()},
            "status": {"allocatable": self.allocatable()},
        }

    # -- GPU-side verification (requires the HIP probe + a visible GPU) --

    def health_check(self, mib: int = 256, iters: int = 5) -> List[Dict[str, Any]]:
        from elastic_gpu_scheduler_amd._native import gpuprobe

        probe = gpuprobe()
        out = []
        for i in range(probe.device_count()):
            bw = probe.hbm_bandwidth(i, mib, iters)
            out.append({"index": i, "hbm_gbps": bw,
                        "healthy": bw >= HBM_HEALTH_THRESHOLD_GBPS})
        return out

    def verify_placement(self, pod_uid: str, device_indexes: List[int],
                         mib: int = 16) -> bool:
        """Stamp+verify a pod-unique pattern on each assigned card."""
        from elastic_gpu_scheduler_amd._native import gpuprobe

        probe = gpuprobe()
        tag = _uid_tag(pod_uid)
        return all(probe.stamp(idx, tag, mib) for idx in device_indexes)

    def measured_topology(self, mib: int = 64, iters: int = 5) -> Dict[str, Any]:
        """Measured xGMI bandwidth matrix (GB/s) alongside the hop matrix.
        On a healthy single-hive OAM board every off-diagonal pair should
        sustain a similar per-link bandwidth (~150 GB/s class per link)."""
        from elastic_gpu_scheduler_amd._native import gpuprobe

        probe = gpuprobe()
        n = probe.device_count()
        bw = [[0.0] * n for _ in range(n)]
        for i in range(n):
            for j in range(n):
                if i != j:
                    bw[i][j] = probe.p2p_bandwidth(i, j, mib, iters)
        return {"hops": probe.xgmi_hop_matrix(), "bandwidth_gbps": bw}


def _uid_tag(uid: str) -> int:
    h = 1469598103934665603
    for ch in uid.encode():
        h ^= ch
        h = (h * 1099511628211) % (1 << 64)
    return h
