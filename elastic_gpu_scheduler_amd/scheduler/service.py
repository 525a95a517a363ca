"""Resource-scheduler service: the verbs behind filter/priorities/bind.

Functional parity with the reference GPUUnitScheduler (pkg/scheduler/
scheduler.go:86-290) and its registry (BuildResourceSchedulers,
scheduler.go:292-334), redesigned:

  * state lives in the native C++ ClusterState (per-node locks, thread-pool
    filter fan-out) instead of Go maps behind one global mutex;
  * Bind writes annotations with a typed-conflict retry loop (the reference
    matches the optimistic-lock error by *text*, scheduler.go:201-209) and
    ROLLS BACK the in-memory allocation if the apiserver writes ultimately
    fail (the reference silently swallows non-conflict errors and returns
    success, scheduler.go:210-211);
  * the bind protocol is crash-consistent: annotations carry the target node
    (elasticgpu.io/scheduled-node), so a crash between annotate and bind is
    replayed idempotently at startup;
  * real scheduling Events are emitted (the reference wires an event
    recorder but never uses it, controller.go:57-65).
"""
from __future__ import annotations

import json
import logging
import queue
import threading
import time
from typing import Any, Dict, List, Optional

from elastic_gpu_scheduler_amd._native import core
from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.k8s.client import ConflictError, KubeClient, NotFoundError
from elastic_gpu_scheduler_amd.utils import types as t

log = logging.getLogger("egs.scheduler")


class BindError(Exception):
    pass


class GPUUnitScheduler:
    """Schedules pods requesting elasticgpu.io/gpu-core + gpu-memory."""

    def __init__(self, client: KubeClient, policy: str = t.PRIORITY_BINPACK,
                 seed: int = 0, threads: int = 0, bare_unit: str = "auto",
                 name: str = "gpushare", topology_weight: float = 0.3) -> None:
        self.client = client
        self.name = name
        self.policy = policy
        self.bare_unit = bare_unit
        self.state = core.ClusterState(policy, seed, threads, topology_weight)
        # released-pod tombstones (reference releasedPodMap, scheduler.go:47):
        # a DELETE seen before the final MODIFIED must not re-add the pod.
        self._released: Dict[str, float] = {}
        self._released_mu = threading.Lock()
        # fast-path membership cache over state's node map (adds only; a
        # removed node falls back to the authoritative native check)
        self._known_nodes: set = set()
        # Binds in flight per node + deferred invalidations: evicting a
        # node between a bind's in-memory allocate and its annotation
        # write would lose the reservation — the refill replays only
        # apiserver-VISIBLE pods, so a concurrent bind on the refilled
        # state could double-book the same cards. Invalidation therefore
        # defers until the node's last in-flight bind completes.
        self._bind_mu = threading.Lock()
        self._binds_inflight: Dict[str, int] = {}
        self._dirty_nodes: set = set()
        # Serialises evict/refill transitions: two threads refilling the
        # same node would replace each other's allocator MID-REPLAY and
        # drop accounting for already-replayed pods. Never taken on the
        # warm path (known-node check first). Lock order: _bind_mu may be
        # held when taking _fill_mu, never the reverse.
        self._fill_mu = threading.Lock()
        # events are emitted asynchronously: an apiserver event write must
        # never sit on the bind critical path
        self._event_q: "queue.Queue" = queue.Queue(maxsize=4096)
        self._event_thread = threading.Thread(target=self._event_loop,
                                              name="egs-events", daemon=True)
        self._event_thread.start()
        self.warm_start()

    # ---- cache management ------------------------------------------------

    def warm_start(self) -> None:
        """Rebuild all accounting from assumed-pod annotations — the
        annotation store on the apiserver IS the checkpoint (reference
        NewGPUUnitScheduler, scheduler.go:86-106)."""
        try:
            pods = self.client.list_pods(label_selector={t.EGPU_ASSUMED: "true"})
        except Exception:
            log.exception("warm start: listing assumed pods failed")
            return
        for pod in pods:
            if obj.is_completed_pod(pod):
                continue
            node = obj.pod_node_name(pod) or (
                pod.get("metadata", {}).get("annotations", {}) or {}).get(
                    t.ANNOTATION_EGPU_NODE, "")
            if not node:
                continue
            self._ensure_node(node, replay=False)
            self._replay_pod(node, pod)

    def _ensure_node(self, node_name: str, replay: bool = True) -> bool:
        """Lazily fill the node cache from a live Get + assumed-pod replay
        (reference getNodeInfo, scheduler.go:62-84)."""
        if node_name in self._known_nodes:
            return True
        with self._fill_mu:
            if self.state.has_node(node_name):  # lost the refill race: done
                self._known_nodes.add(node_name)
                return True
            try:
                node = self.client.get_node(node_name)
            except NotFoundError:
                return False
            except Exception:
                log.exception("get node %s failed", node_name)
                return False
            devices = obj.node_devices(node, self.bare_unit)
            if not devices:
                return False
            topo = obj.node_topology(node)
            self.state.add_node(node_name, devices, topo)
            if replay:
                try:
                    pods = self.client.list_pods(
                        label_selector={t.EGPU_ASSUMED: "true"},
                        field_selector={"spec.nodeName": node_name})
                except Exception:
                    pods = []
                for pod in pods:
                    if not obj.is_completed_pod(pod):
                        self._replay_pod(node_name, pod)
            self._known_nodes.add(node_name)
        return True

    def invalidate_node(self, node_name: str) -> None:
        """Evict a node from the cache so fresh inventory/topology (e.g. the
        agent republishing annotations) is re-read on next use. If a bind
        is mid-flight on this node the eviction is DEFERRED to the last
        bind's completion — evicting now would drop its not-yet-annotated
        reservation and let the refilled cache double-book those cards
        (found by tests/test_concurrency.py invalidation-churn)."""
        with self._bind_mu:
            if self._binds_inflight.get(node_name, 0) > 0:
                self._dirty_nodes.add(node_name)
                return
            # evict UNDER the lock: a bind entering concurrently must only
            # proceed after the eviction, so its _ensure_node refills first
            self._evict_node(node_name)

    def _evict_node(self, node_name: str) -> None:
        with self._fill_mu:  # atomic vs any in-progress refill
            self._known_nodes.discard(node_name)
            self.state.remove_node(node_name)

    def _bind_enter(self, node_name: str) -> None:
        with self._bind_mu:
            self._binds_inflight[node_name] = \
                self._binds_inflight.get(node_name, 0) + 1

    def _bind_exit(self, node_name: str) -> None:
        with self._bind_mu:
            n = self._binds_inflight.get(node_name, 1) - 1
            if n <= 0:
                self._binds_inflight.pop(node_name, None)
                if node_name in self._dirty_nodes:
                    self._dirty_nodes.discard(node_name)
                    self._evict_node(node_name)  # under the lock, see above
            else:
                self._binds_inflight[node_name] = n

    def _replay_pod(self, node_name: str, pod: Dict[str, Any]) -> None:
        allocated = obj.parse_allocation(pod)
        if allocated is None:
            return
        req = obj.pod_gpu_request(pod, self.bare_unit)
        option = core.GPUOption()
        option.allocated = allocated
        try:
            self.state.add_pod(node_name, obj.pod_uid(pod), req, option)
        except RuntimeError as exc:
            # Double-booked annotations: surface loudly, keep serving.
            log.error("replaying pod %s onto %s failed: %s",
                      obj.pod_key(pod), node_name, exc)

    # ---- verbs (ResourceScheduler interface, scheduler.go:30-39) --------

    def assume(self, node_names: List[str], pod: Dict[str, Any]):
        """Filter: returns (ok_nodes, failed_nodes: {name: reason})."""
        req = obj.pod_gpu_request(pod, self.bare_unit)
        uid = obj.pod_uid(pod)
        distinct = obj.wants_container_spread(pod)
        for n in node_names:
            self._ensure_node(n)
        verdicts = self.state.assume(node_names, uid, req, distinct)
        ok, failed = [], {}
        for name, v in zip(node_names, verdicts):
            if v == 0:
                ok.append(name)
            elif v == 1:
                failed[name] = "insufficient GPU resources"
            else:
                failed[name] = "node has no GPU inventory"
        return ok, failed

    def score(self, node_names: List[str], pod: Dict[str, Any]) -> List[float]:
        req = obj.pod_gpu_request(pod, self.bare_unit)
        uid = obj.pod_uid(pod)
        distinct = obj.wants_container_spread(pod)
        for n in node_names:
            self._ensure_node(n)
        return self.state.score(node_names, uid, req, distinct)

    def bind(self, node_name: str, pod: Dict[str, Any]) -> None:
        """Allocate -> annotate (conflict-retried) -> bind. Rolls back the
        allocation if the apiserver writes fail."""
        uid = obj.pod_uid(pod)
        req = obj.pod_gpu_request(pod, self.bare_unit)
        # Inside the enter/exit window any invalidate_node defers: the
        # in-memory allocation below is invisible to the apiserver until
        # _write_bind lands, and an eviction+refill in between would let a
        # concurrent bind double-book these cards.
        self._bind_enter(node_name)
        try:
            if not self._ensure_node(node_name):
                raise BindError(f"unknown or GPU-less node {node_name}")
            try:
                option = self.state.allocate(node_name, uid, req,
                                             obj.wants_container_spread(pod))
            except RuntimeError as exc:
                raise BindError(str(exc)) from exc
            try:
                self._write_bind(node_name, pod, option)
            except Exception:
                self.state.forget_pod(uid)
                raise
        finally:
            self._bind_exit(node_name)
        self._emit_event(pod, "Scheduled",
                         f"placed on {node_name} gpus "
                         f"{[list(a) for a in option.allocated]} "
                         f"score {option.score:.2f}")

    def _write_bind(self, node_name: str, pod: Dict[str, Any], option) -> None:
        ns, name = obj.pod_namespace(pod), obj.pod_name(pod)
        # the handler fetched `pod` fresh for this bind; annotate in place
        annotated = obj.apply_allocation(pod, [list(a) for a in option.allocated],
                                         node=node_name, score=option.score,
                                         copy=False)
        for attempt in range(3):
            try:
                self.client.update_pod(annotated)
                break
            except ConflictError:
                if attempt == 2:
                    raise BindError(
                        f"pod {ns}/{name}: annotation update kept conflicting")
                try:
                    fresh = self.client.get_pod(ns, name)
                except NotFoundError:
                    raise BindError(f"pod {ns}/{name} vanished during bind")
                if obj.pod_uid(fresh) != obj.pod_uid(pod):
                    raise BindError(f"pod {ns}/{name} was recreated during bind")
                annotated = obj.apply_allocation(
                    fresh, [list(a) for a in option.allocated],
                    node=node_name, score=option.score, copy=False)
        self.client.bind_pod(ns, name, node_name)

    def add_pod(self, pod: Dict[str, Any]) -> None:
        """Controller sync: account an already-placed pod (idempotent)."""
        uid = obj.pod_uid(pod)
        with self._released_mu:
            if uid in self._released:
                return
        node_name = obj.pod_node_name(pod)
        if not node_name or self.state.known_pod(uid):
            return
        if not self._ensure_node(node_name):
            return
        self._replay_pod(node_name, pod)
        self.state.note_pod_node(uid, node_name)

    def forget_pod(self, pod: Dict[str, Any]) -> None:
        uid = obj.pod_uid(pod)
        with self._released_mu:
            # dict preserves insertion order: evicting the oldest entry keeps
            # the tombstone set bounded in O(1) per forget (a wholesale
            # rebuild here measurably decayed sustained throughput)
            self._released.pop(uid, None)
            self._released[uid] = time.time()
            while len(self._released) > 4096:
                self._released.pop(next(iter(self._released)))
        self.state.forget_pod(uid)

    def known_pod(self, pod: Dict[str, Any]) -> bool:
        return self.state.known_pod(obj.pod_uid(pod))

    def released_pod(self, pod: Dict[str, Any]) -> bool:
        with self._released_mu:
            return obj.pod_uid(pod) in self._released

    def process_preemption(self, pod: Dict[str, Any],
                           node_to_victims: Dict[str, List[str]]
                           ) -> Dict[str, List[str]]:
        """Extender preemption: for each candidate node, decide whether
        evicting (a minimal subset of) the proposed victim pods makes this
        pod feasible. Returns node -> victim UIDs actually required; nodes
        where even all victims don't help are omitted. Capability beyond the
        reference (neither it nor this verb exist upstream)."""
        req = obj.pod_gpu_request(pod, self.bare_unit)
        uid = obj.pod_uid(pod)
        out: Dict[str, List[str]] = {}
        for node, victims in node_to_victims.items():
            if not self._ensure_node(node):
                continue
            victims = sorted(set(victims))  # deterministic
            if not self.state.feasible_with_victims(node, uid, req, victims):
                continue  # even evicting everything doesn't fit
            # greedy minimisation: drop victims that aren't needed
            needed = list(victims)
            for v in victims:
                trial = [x for x in needed if x != v]
                if self.state.feasible_with_victims(node, uid, req, trial):
                    needed = trial
            out[node] = needed
        return out

    # ---- observability ---------------------------------------------------

    def status(self) -> Dict[str, Any]:
        """Per-node per-card availability (reference Status,
        scheduler.go:283-290 + /scheduler/status route)."""
        nodes = {}
        for name in self.state.node_names():
            devices = self.state.node_devices(name)
            nodes[name] = {
                "policy": self.policy,
                "pending_assumes": self.state.node_assumed_count(name),
                "gpus": [{
                    "core_available": d.core_avail,
                    "core_total": d.core_total,
                    "memory_available": d.mem_avail,
                    "memory_total": d.mem_total,
                } for d in devices],
                "pods": self.state.node_pod_placements(name),
            }
        return {"name": self.name, "policy": self.policy, "nodes": nodes}

    def _emit_event(self, pod: Dict[str, Any], reason: str, message: str) -> None:
        try:
            self._event_q.put_nowait((obj.pod_namespace(pod), pod, reason,
                                      message))
        except queue.Full:
            log.debug("event queue full; dropping %s event", reason)

    def _event_loop(self) -> None:
        while True:
            ns, pod, reason, message = self._event_q.get()
            try:
                self._write_event(ns, pod, reason, message)
            except Exception:
                log.debug("event emit failed", exc_info=True)

    def flush_events(self, timeout: float = 5.0) -> None:
        """Wait for queued events to drain (tests)."""
        deadline = time.time() + timeout
        while not self._event_q.empty() and time.time() < deadline:
            time.sleep(0.01)

    def _write_event(self, ns: str, pod: Dict[str, Any], reason: str,
                     message: str) -> None:
        try:
            self.client.create_event(ns, {
                "metadata": {"generateName": "egs-"},
                "involvedObject": {
                    "kind": "Pod",
                    "namespace": obj.pod_namespace(pod),
                    "name": obj.pod_name(pod),
                    "uid": obj.pod_uid(pod),
                },
                "reason": reason,
                "message": message,
                "type": "Normal",
                "source": {"component": "elastic-gpu-scheduler-amd"},
            })
        except Exception:
            log.debug("event emit failed", exc_info=True)


class SchedulerRegistry:
    """resource-name -> scheduler map (reference BuildResourceSchedulers,
    scheduler.go:292-334). All five elasticgpu resource names route to the
    GPUUnitScheduler; mode selects which names are active."""

    def __init__(self, client: KubeClient, mode: str = t.MODE_GPUSHARE,
                 policy: str = t.PRIORITY_BINPACK, seed: int = 0,
                 threads: int = 0, bare_unit: str = "auto",
                 topology_weight: float = 0.3) -> None:
        self.schedulers: Dict[str, GPUUnitScheduler] = {}
        unit = GPUUnitScheduler(client, policy=policy, seed=seed, threads=threads,
                                bare_unit=bare_unit, name=mode,
                                topology_weight=topology_weight)
        if mode == t.MODE_QGPU:
            names = (t.RESOURCE_QGPU_CORE, t.RESOURCE_QGPU_MEMORY)
        elif mode == t.MODE_PGPU:
            names = (t.RESOURCE_PGPU,)
        else:  # gpushare: accept every elasticgpu resource (most permissive)
            names = t.GPU_RESOURCE_NAMES
        for n in names:
            self.schedulers[n] = unit
        self.default = unit

    def for_pod(self, pod: Dict[str, Any]) -> Optional[GPUUnitScheduler]:
        """Match the pod's first registered resource request (reference
        GetResourceScheduler, scheduler.go:323-334)."""
        for c in pod.get("spec", {}).get("containers", []) or []:
            res = c.get("resources", {}) or {}
            for section in ("requests", "limits"):
                for rname in (res.get(section, {}) or {}):
                    if rname in self.schedulers:
                        return self.schedulers[rname]
        return None

    def status_json(self) -> str:
        seen = {}
        for sch in self.schedulers.values():
            seen[sch.name] = sch.status()
        return json.dumps(seen, indent=2)
