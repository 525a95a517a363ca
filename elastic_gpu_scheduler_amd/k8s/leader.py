"""Lease-based leader election.

The reference must run as a single replica — two replicas would double-book
GPUs (deploy yaml replicas: 1). This adds the standard
coordination.k8s.io Lease protocol so standby replicas are safe: exactly one
elector holds the lease; the others wait and take over when renewals stop.

Usage: LeaderElector(client, name, identity).run(on_started, on_stopped)
blocks, calling on_started() when leadership is acquired and on_stopped()
when it is lost (the caller should exit and let Kubernetes restart it — the
same crash-recovery path as any restart, docs/ARCHITECTURE.md).
"""
from __future__ import annotations

import logging
import threading
import time
from typing import Callable, Optional

from elastic_gpu_scheduler_amd.k8s.client import ConflictError, KubeClient, NotFoundError

log = logging.getLogger("egs.leader")


class LeaderElector:
    def __init__(self, client: KubeClient, name: str, identity: str,
                 namespace: str = "kube-system",
                 lease_duration: float = 15.0, renew_period: float = 5.0,
                 retry_period: float = 2.0) -> None:
        self.client = client
        self.name = name
        self.identity = identity
        self.namespace = namespace
        self.lease_duration = lease_duration
        self.renew_period = renew_period
        self.retry_period = retry_period
        self._stop = threading.Event()
        self.is_leader = False

    # -- lease helpers (monotonic-free: uses apiserver-side renew times) --

    def _try_acquire_or_renew(self) -> bool:
        now = time.time()
        try:
            lease = self.client.get_lease(self.namespace, self.name)
        except NotFoundError:
            lease = {
                "metadata": {"name": self.name, "namespace": self.namespace},
                "spec": {},
            }
            lease["spec"] = self._owned_spec(now)
            try:
                self.client.create_lease(self.namespace, lease)
                return True
            except ConflictError:
                return False
        spec = lease.get("spec", {}) or {}
        holder = spec.get("holderIdentity")
        renew = float(spec.get("renewTime", 0) or 0)
        expired = now - renew > self.lease_duration
        if holder not in (None, "", self.identity) and not expired:
            return False
        lease["spec"] = self._owned_spec(now, previous=spec)
        try:
            self.client.update_lease(self.namespace, lease)
            return True
        except (ConflictError, NotFoundError):
            return False

    def _owned_spec(self, now: float, previous: Optional[dict] = None) -> dict:
        transitions = int((previous or {}).get("leaseTransitions", 0) or 0)
        if previous and previous.get("holderIdentity") != self.identity:
            transitions += 1
        return {
            "holderIdentity": self.identity,
            "leaseDurationSeconds": int(self.lease_duration),
            "renewTime": now,
            "leaseTransitions": transitions,
        }

    # -- main loop --

    def run(self, on_started: Callable[[], None],
            on_stopped: Callable[[], None]) -> None:
        """Block until stop(): acquire -> on_started -> renew loop; a lost
        lease calls on_stopped and returns."""
        while not self._stop.is_set():
            if self._try_acquire_or_renew():
                break
            self._stop.wait(self.retry_period)
        if self._stop.is_set():
            return
        self.is_leader = True
        log.info("%s acquired leadership of %s", self.identity, self.name)
        on_started()
        while not self._stop.wait(self.renew_period):
            if not self._try_acquire_or_renew():
                log.warning("%s lost leadership of %s", self.identity, self.name)
                self.is_leader = False
                on_stopped()
                return
        # graceful stop: release so a standby takes over immediately
        self._release()

    def stop(self) -> None:
        self._stop.set()

    def _release(self) -> None:
        try:
            lease = self.client.get_lease(self.namespace, self.name)
            if (lease.get("spec", {}) or {}).get("holderIdentity") == self.identity:
                lease["spec"]["holderIdentity"] = ""
                lease["spec"]["renewTime"] = 0
                self.client.update_lease(self.namespace, lease)
        except Exception:
            log.debug("lease release failed", exc_info=True)
        self.is_leader = False
