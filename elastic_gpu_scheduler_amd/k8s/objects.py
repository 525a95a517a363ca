"""Pod / Node helpers and the annotation codec.

Operates on plain dicts shaped like Kubernetes API JSON (the same objects the
kube-scheduler POSTs inside ExtenderArgs). Functional parity targets:
reference pkg/scheduler/pod.go (IsGPUPod, IsCompletedPod, request extraction,
GetUpdatedPodAnnotationSpec, NewGPUOptionFromPod) — redesigned, with the
whole-card-drops-from-result bug of GetContainerGPUResource (pod.go:142-144)
deliberately not reproduced.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from elastic_gpu_scheduler_amd._native import core
from elastic_gpu_scheduler_amd.utils import types as t
from elastic_gpu_scheduler_amd.utils.quantity import parse_memory_bytes, parse_quantity

Pod = Dict[str, Any]
Node = Dict[str, Any]


# --- basic metadata -------------------------------------------------------

def pod_uid(pod: Pod) -> str:
    return str(pod.get("metadata", {}).get("uid", ""))


def pod_name(pod: Pod) -> str:
    return str(pod.get("metadata", {}).get("name", ""))


def pod_namespace(pod: Pod) -> str:
    return str(pod.get("metadata", {}).get("namespace", "default"))


def pod_key(pod: Pod) -> str:
    return f"{pod_namespace(pod)}/{pod_name(pod)}"


def pod_node_name(pod: Pod) -> str:
    return str(pod.get("spec", {}).get("nodeName", "") or "")


def is_completed_pod(pod: Pod) -> bool:
    """Succeeded/Failed, or being deleted. Mirrors reference pod.go:16-25."""
    if pod.get("metadata", {}).get("deletionTimestamp"):
        return True
    phase = pod.get("status", {}).get("phase", "")
    return phase in ("Succeeded", "Failed")


# --- GPU demand extraction ------------------------------------------------

def _container_resources(container: Dict[str, Any]) -> Dict[str, Any]:
    res = container.get("resources", {}) or {}
    merged: Dict[str, Any] = {}
    merged.update(res.get("requests", {}) or {})
    # Extended resources require requests == limits; trust limits if only they
    # are set.
    for k, v in (res.get("limits", {}) or {}).items():
        merged.setdefault(k, v)
    return merged


def is_gpu_pod(pod: Pod) -> bool:
    """True when any container requests one of the five elasticgpu resources
    (reference pod.go:27-34)."""
    for c in pod.get("spec", {}).get("containers", []) or []:
        res = _container_resources(c)
        if any(name in res for name in t.GPU_RESOURCE_NAMES):
            return True
    return False


def container_gpu_unit(container: Dict[str, Any], bare_unit: str = "auto"):
    """One container's demand as a native GPUUnit.

    Semantics (reference NewGPURequest, pkg/scheduler/allocate.go:35-58):
      core == 0 and mem == 0  -> no GPU
      core >= 100             -> core/100 whole cards (memory ignored)
      else                    -> fractional {core%, memory bytes} of one card
    Extension: `elasticgpu.io/pgpu: N` is N whole cards (the reference keeps
    pgpu mode as a commented-out TODO, scheduler.go:296-302).
    """
    res = _container_resources(container)
    core_units = 0
    for name in (t.RESOURCE_GPU_CORE, t.RESOURCE_QGPU_CORE):
        if name in res:
            core_units += int(parse_quantity(res[name]))
    mem_bytes = 0
    for name in (t.RESOURCE_GPU_MEMORY, t.RESOURCE_QGPU_MEMORY):
        if name in res:
            mem_bytes += parse_memory_bytes(res[name], bare_unit)
    pgpu = int(parse_quantity(res[t.RESOURCE_PGPU])) if t.RESOURCE_PGPU in res else 0

    if pgpu > 0:
        return core.GPUUnit(gpu_count=pgpu, core=0, memory=0)
    if core_units == 0 and mem_bytes == 0:
        return core.GPUUnit(gpu_count=0, core=0, memory=0)
    if core_units >= t.GPU_CORE_EACH_CARD:
        return core.GPUUnit(gpu_count=core_units // t.GPU_CORE_EACH_CARD,
                            core=0, memory=0)
    return core.GPUUnit(gpu_count=0, core=core_units, memory=mem_bytes)


def pod_gpu_request(pod: Pod, bare_unit: str = "auto") -> List[Any]:
    """Per-container GPURequest for the native core."""
    return [container_gpu_unit(c, bare_unit)
            for c in pod.get("spec", {}).get("containers", []) or []]


# --- annotation codec (contract with the node agent) ----------------------

def wants_container_spread(pod: Pod) -> bool:
    """elasticgpu.io/spread-containers=true: place each container of this
    pod on a distinct card."""
    ann = pod.get("metadata", {}).get("annotations", {}) or {}
    labels = pod.get("metadata", {}).get("labels", {}) or {}
    return (ann.get(t.ANNOTATION_SPREAD_CONTAINERS) == "true" or
            labels.get(t.ANNOTATION_SPREAD_CONTAINERS) == "true")


def is_assumed(pod: Pod) -> bool:
    ann = pod.get("metadata", {}).get("annotations", {}) or {}
    return ann.get(t.EGPU_ASSUMED) == "true"


def allocation_annotations(pod: Pod, allocated: List[List[int]],
                           node: str = "", score: float = 0.0) -> Dict[str, str]:
    """Annotations recording a placement: per-container device indexes plus
    the assumed marker (reference GetUpdatedPodAnnotationSpec, pod.go:57-78)
    and our MI355X extensions (node, score) for observability/recovery."""
    out: Dict[str, str] = {t.EGPU_ASSUMED: "true"}
    containers = pod.get("spec", {}).get("containers", []) or []
    for i, c in enumerate(containers):
        ids = allocated[i] if i < len(allocated) else []
        out[t.ANNOTATION_EGPU_CONTAINER_PREFIX + c.get("name", str(i))] = (
            ",".join(str(x) for x in ids))
    if node:
        out[t.ANNOTATION_EGPU_NODE] = node
    out[t.ANNOTATION_EGPU_SCORE] = f"{score:.3f}"
    return out


def apply_allocation(pod: Pod, allocated: List[List[int]], node: str = "",
                     score: float = 0.0, copy: bool = True) -> Pod:
    """Return the pod with placement annotations + assumed label applied.

    copy=False mutates `pod` in place — valid when the caller owns the dict
    (e.g. a fresh get_pod result); the bind hot path uses it to skip one
    deep copy per bind."""
    from elastic_gpu_scheduler_amd.k8s.client import _jcopy

    p = _jcopy(pod) if copy else pod
    meta = p.setdefault("metadata", {})
    ann = meta.setdefault("annotations", {})
    ann.update(allocation_annotations(pod, allocated, node, score))
    labels = meta.setdefault("labels", {})
    labels[t.EGPU_ASSUMED] = "true"
    return p


def parse_allocation(pod: Pod) -> Optional[List[List[int]]]:
    """Reconstruct the per-container device indexes from annotations
    (reference NewGPUOptionFromPod, pkg/scheduler/allocate.go:75-93).
    Returns None when the pod carries no complete placement."""
    ann = pod.get("metadata", {}).get("annotations", {}) or {}
    containers = pod.get("spec", {}).get("containers", []) or []
    out: List[List[int]] = []
    found = False
    for i, c in enumerate(containers):
        key = t.ANNOTATION_EGPU_CONTAINER_PREFIX + c.get("name", str(i))
        raw = ann.get(key)
        if raw is None:
            out.append([])
            continue
        found = True
        out.append([int(x) for x in raw.split(",") if x.strip() != ""])
    return out if found else None


# --- node inventory -------------------------------------------------------

def node_devices(node: Node, bare_unit: str = "auto") -> List[Any]:
    """Build the per-card Device vector for a node.

    Priority order:
      1. the agent-published JSON inventory annotation (exact per-card
         core/memory — supports heterogeneous cards, unlike the reference's
         evenly-divided assumption at node.go:37-38);
      2. allocatable elasticgpu.io/gpu-core + gpu-memory (cards =
         gpu-core / 100, memory split evenly — reference-compatible);
      3. allocatable amd.com/gpu count with MI355X defaults (288 GiB/card).
    """
    import json

    meta = node.get("metadata", {}) or {}
    ann = meta.get("annotations", {}) or {}
    inv_raw = ann.get(t.ANNOTATION_NODE_INVENTORY)
    if inv_raw:
        try:
            inv = json.loads(inv_raw)
            # Device list position == PHYSICAL card index. The vector is
            # keyed by each card's published "index", and any gap (a card an
            # agent omitted, e.g. a sick one gated by health) is filled with
            # a zero-capacity placeholder — never schedulable, but keeping
            # every later card at its true physical index. A sick card
            # published as a zero-capacity entry (agent.publish_with_health)
            # lands here identically.
            by_index: Dict[int, Dict[str, Any]] = {}
            for card in inv.get("cards", []) or []:
                by_index[int(card.get("index", len(by_index)))] = card
            devices = []
            if by_index:
                for i in range(max(by_index) + 1):
                    card = by_index.get(i)
                    if card is None or card.get("sick"):
                        devices.append(core.Device(core_total=0, core_avail=0,
                                                   mem_total=0, mem_avail=0))
                        continue
                    mem = int(card.get("memory_bytes", t.MI355X_MEMORY_BYTES))
                    cr = int(card.get("core", t.GPU_CORE_EACH_CARD))
                    devices.append(core.Device(core_total=cr, core_avail=cr,
                                               mem_total=mem, mem_avail=mem))
            if devices:
                return devices
        except (ValueError, TypeError):
            pass  # fall through to allocatable

    allocatable = node.get("status", {}).get("allocatable", {}) or {}
    if t.RESOURCE_GPU_CORE in allocatable:
        total_core = int(parse_quantity(allocatable[t.RESOURCE_GPU_CORE]))
        count = max(total_core // t.GPU_CORE_EACH_CARD, 0)
        if count > 0:
            if t.RESOURCE_GPU_MEMORY in allocatable:
                total_mem = parse_memory_bytes(allocatable[t.RESOURCE_GPU_MEMORY],
                                               bare_unit)
                per_card = total_mem // count
            else:
                per_card = t.MI355X_MEMORY_BYTES
            return [core.Device(core_total=t.GPU_CORE_EACH_CARD,
                                core_avail=t.GPU_CORE_EACH_CARD,
                                mem_total=per_card, mem_avail=per_card)
                    for _ in range(count)]

    if t.RESOURCE_AMD_GPU in allocatable:
        count = int(parse_quantity(allocatable[t.RESOURCE_AMD_GPU]))
        return [core.Device() for _ in range(count)]

    return []


def node_topology(node: Node) -> List[List[int]]:
    """Agent-published xGMI hop matrix; empty = assume a single fully
    connected hive (every MI355X OAM pair is one xGMI hop)."""
    import json

    ann = node.get("metadata", {}).get("annotations", {}) or {}
    raw = ann.get(t.ANNOTATION_NODE_TOPOLOGY)
    if not raw:
        return []
    try:
        m = json.loads(raw)
        if isinstance(m, dict):
            m = m.get("hops", [])
        if (isinstance(m, list) and all(isinstance(r, list) for r in m)):
            return [[int(x) for x in row] for row in m]
    except (ValueError, TypeError):
        pass
    return []
