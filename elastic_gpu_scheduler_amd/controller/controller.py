"""Reconcile controller: keeps scheduler accounting honest against the
apiserver.

Analogue of the reference's informer controller (pkg/controller/
controller.go): watches pods, filters to GPU pods, runs a rate-limited
workqueue with N workers, and reconciles lifecycle into the scheduler —
AddPod for running assigned pods, ForgetPod for completed/deleted ones
(controller.go:154-185, 301-331). Differences:

  * a plain threaded workqueue with per-key dedup instead of client-go
    machinery; resync is an explicit periodic relist (reference uses a 30 s
    shared-informer resync, controller.go:24);
  * the node informer the reference creates but never consults
    (controller.go:97-99) is ALIVE here: a node watch invalidates the
    per-node allocator cache immediately on inventory/topology republish
    or node deletion (plus a relist fallback in the periodic resync), so
    agent updates take effect without waiting out the resync period.
"""
from __future__ import annotations

import logging
import queue
import threading
import time
from typing import Any, Dict, Optional

from elastic_gpu_scheduler_amd.k8s import objects as obj
from elastic_gpu_scheduler_amd.k8s.client import KubeClient
from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry

log = logging.getLogger("egs.controller")


class Controller:
    def __init__(self, client: KubeClient, registry: SchedulerRegistry,
                 workers: int = 1, resync_seconds: float = 30.0) -> None:
        self.client = client
        self.registry = registry
        self.workers = max(1, workers)
        self.resync_seconds = resync_seconds
        self._queue: "queue.Queue[Optional[str]]" = queue.Queue()
        self._pending: Dict[str, Dict[str, Any]] = {}  # key -> last seen pod
        self._retries: Dict[str, int] = {}
        self.max_retries = 5
        self._node_rv: Dict[str, Any] = {}  # node -> last seen resourceVersion
        self._missing_once: set = set()  # uids absent from the last relist
        self._pending_mu = threading.Lock()
        self._stop = threading.Event()
        self._threads: list[threading.Thread] = []
        self._unsubscribe = None

    # -- lifecycle --

    def start(self) -> None:
        self._unsubscribe = self.client.watch_pods(self._on_event)
        # Node watch: the agent republishing inventory/topology (or a node
        # vanishing) invalidates the per-node allocator cache IMMEDIATELY
        # instead of waiting for the periodic resync. The reference builds
        # a node informer and never consults it (controller.go:97-99).
        try:
            self._unsubscribe_nodes = self.client.watch_nodes(
                self._on_node_event)
            # Baseline the RVs so the FIRST change after start already
            # invalidates (watch backends without an initial replay would
            # otherwise record-only the first event per node).
            for n in self.client.list_nodes():
                name = (n.get("metadata", {}) or {}).get("name")
                if name:
                    self._node_rv.setdefault(
                        name, n.get("metadata", {}).get("resourceVersion"))
        except NotImplementedError:
            self._unsubscribe_nodes = None
        for i in range(self.workers):
            t = threading.Thread(target=self._worker, name=f"egs-sync-{i}",
                                 daemon=True)
            t.start()
            self._threads.append(t)
        t = threading.Thread(target=self._resync_loop, name="egs-resync",
                             daemon=True)
        t.start()
        self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()
        if self._unsubscribe:
            self._unsubscribe()
        if getattr(self, "_unsubscribe_nodes", None):
            self._unsubscribe_nodes()
        for _ in self._threads:
            self._queue.put(None)
        for t in self._threads:
            t.join(timeout=2.0)

    def wait_idle(self, timeout: float = 5.0) -> bool:
        """Wait until the workqueue has drained (tests / shutdown)."""
        deadline = time.time() + timeout
        while time.time() < deadline:
            with self._pending_mu:
                empty = not self._pending
            if empty and self._queue.empty():
                return True
            time.sleep(0.01)
        return False

    # -- watch handling (reference controller.go:212-299) --

    def _on_event(self, event_type: str, pod: Dict[str, Any]) -> None:
        if not obj.is_gpu_pod(pod):
            return
        key = obj.pod_key(pod)
        if event_type == "DELETED":
            pod = dict(pod)
            pod.setdefault("metadata", {})["_egs_deleted"] = True
        self._enqueue(key, pod)

    def _on_node_event(self, event_type: str, node: Dict[str, Any]) -> None:
        """Node ADDED/MODIFIED/DELETED: evict the allocator cache when the
        node's resourceVersion moved past what we last saw (first sight is
        record-only — invalidating a node we never tracked would churn the
        cache at watch startup). Benign race with _resync_nodes: the worst
        case is one extra invalidation, which only costs a lazy refill."""
        name = (node.get("metadata", {}) or {}).get("name")
        if not name:
            return
        schedulers = {id(s): s for s in self.registry.schedulers.values()}
        if event_type == "DELETED":
            self._node_rv.pop(name, None)
            for sch in schedulers.values():
                sch.invalidate_node(name)
            return
        rv = (node.get("metadata", {}) or {}).get("resourceVersion")
        prev = self._node_rv.get(name)
        self._node_rv[name] = rv
        if prev is not None and rv != prev:
            log.info("node %s changed (rv %s -> %s, watch); refreshing cache",
                     name, prev, rv)
            for sch in schedulers.values():
                sch.invalidate_node(name)

    def _enqueue(self, key: str, pod: Dict[str, Any]) -> None:
        with self._pending_mu:
            fresh = key not in self._pending
            self._pending[key] = pod
        if fresh:
            self._queue.put(key)

    # -- workers (reference processNextWorkItem, controller.go:189-210) --

    def _worker(self) -> None:
        while not self._stop.is_set():
            key = self._queue.get()
            if key is None:
                return
            with self._pending_mu:
                pod = self._pending.pop(key, None)
            if pod is None:
                continue
            try:
                self._sync_pod(pod)
                self._retries.pop(key, None)
            except Exception:
                # rate-limited requeue (reference uses a client-go
                # rate-limited workqueue, controller.go:64,189-210)
                n = self._retries.get(key, 0) + 1
                if n > self.max_retries:
                    log.exception("sync of %s failed %d times; dropping",
                                  key, n)
                    self._retries.pop(key, None)
                    continue
                self._retries[key] = n
                log.warning("sync of %s failed (attempt %d); requeueing",
                            key, n, exc_info=True)
                delay = min(0.05 * (2 ** n), 5.0)
                threading.Timer(delay, self._enqueue, args=(key, pod)).start()

    def _sync_pod(self, pod: Dict[str, Any]) -> None:
        """Reference syncPod (controller.go:154-185): completed/deleted ->
        release; running & assigned -> account."""
        sch = self.registry.for_pod(pod)
        if sch is None:
            return
        deleted = pod.get("metadata", {}).get("_egs_deleted", False)
        if deleted or obj.is_completed_pod(pod):
            sch.forget_pod(pod)
            return
        if obj.pod_node_name(pod) and obj.is_assumed(pod):
            sch.add_pod(pod)

    # -- periodic resync (stand-in for informer resync) --

    def _resync_loop(self) -> None:
        while not self._stop.wait(self.resync_seconds):
            try:
                self.resync_once()
            except Exception:
                log.exception("resync failed")

    def resync_once(self) -> None:
        # Relist only ASSUMED pods (label selector, server-side): every pod
        # the scheduler accounts carries elasticgpu.io/assumed=true (set at
        # bind), so this is sufficient for both replay and eviction — and at
        # real cluster scale it avoids relisting every pod in the cluster
        # each period (the reference pays that via its informer's full
        # resync, controller.go:24).
        from elastic_gpu_scheduler_amd.utils import types as t
        pods = self.client.list_pods(label_selector={t.EGPU_ASSUMED: "true"})
        live = set()
        for pod in pods:
            if not obj.is_gpu_pod(pod):
                continue
            live.add(obj.pod_uid(pod))
            self._sync_pod(pod)
        schedulers = {id(s): s for s in self.registry.schedulers.values()}
        # Evict accounting for pods that vanished without a DELETE event —
        # but only after they have been missing from TWO consecutive relists.
        # The pod list is a snapshot taken before the sweep: a pod bound
        # between list_pods() and the sweep would otherwise be wrongly
        # forgotten, transiently over-freeing its cards until the next event
        # re-adds it (another pod could double-book them in that window).
        missing_now = set()
        for sch in schedulers.values():
            for node in sch.state.node_names():
                for uid in sch.state.node_pods(node):
                    if uid in live:
                        continue
                    if uid in self._missing_once:
                        log.info("pod uid %s missing from two consecutive "
                                 "relists; forgetting", uid)
                        sch.state.forget_pod(uid)
                    else:
                        missing_now.add(uid)
        self._missing_once = missing_now
        self._resync_nodes(schedulers.values())

    def _resync_nodes(self, schedulers) -> None:
        """Invalidate node caches whose objects changed (the agent
        republishing inventory/topology bumps resourceVersion) or vanished;
        the next use lazily refills and replays assumed pods."""
        try:
            nodes = self.client.list_nodes()
        except Exception:
            log.debug("node relist failed", exc_info=True)
            return
        current = {n["metadata"]["name"]: n["metadata"].get("resourceVersion")
                   for n in nodes if n.get("metadata", {}).get("name")}
        for sch in schedulers:
            for name in sch.state.node_names():
                if name not in current:
                    log.info("node %s deleted; evicting from cache", name)
                    sch.invalidate_node(name)
                elif self._node_rv.get(name) is not None and \
                        current[name] != self._node_rv[name]:
                    log.info("node %s changed (rv %s -> %s); refreshing cache",
                             name, self._node_rv[name], current[name])
                    sch.invalidate_node(name)
        self._node_rv = current
