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
import re
import threading
import time
from datetime import datetime, timezone
from typing import Callable, Optional

from elastic_gpu_scheduler_amd.k8s.client import ConflictError, KubeClient, NotFoundError

log = logging.getLogger("egs.leader")

_MICROTIME_RE = re.compile(
    r"^(\d{4})-(\d{2})-(\d{2})[Tt](\d{2}):(\d{2}):(\d{2})"
    r"(?:\.(\d{1,9}))?([Zz]|[+-]\d{2}:?\d{2})$")


def format_microtime(ts: float) -> str:
    """Epoch seconds -> coordination.k8s.io MicroTime (RFC3339 with
    microseconds, e.g. "2026-09-14T10:11:12.123456Z"). A real apiserver
    REJECTS a non-RFC3339 renewTime — the r1 implementation wrote a unix
    float and only ever worked against the in-memory fake."""
    dt = datetime.fromtimestamp(ts, tz=timezone.utc)
    return dt.strftime("%Y-%m-%dT%H:%M:%S.%f") + "Z"


def parse_microtime(value) -> float:
    """MicroTime/Time string -> epoch seconds; 0.0 when absent/invalid.

    Tolerant of what real writers produce: kubelet/client-go MicroTime
    (6-digit fraction, Z), metav1.Time (no fraction), explicit UTC offsets,
    and 1-9 fractional digits. Numeric input is accepted for back-compat
    with leases written by the r1 format."""
    if value in (None, "", 0):
        return 0.0
    if isinstance(value, (int, float)):
        return float(value)
    m = _MICROTIME_RE.match(str(value).strip())
    if not m:
        return 0.0
    y, mo, d, h, mi, s = (int(m.group(i)) for i in range(1, 7))
    frac = m.group(7) or ""
    micro = int(frac.ljust(6, "0")[:6]) if frac else 0
    tz = m.group(8)
    if tz in ("Z", "z"):
        tzinfo = timezone.utc
    else:
        sign = 1 if tz[0] == "+" else -1
        hh, mm = int(tz[1:3]), int(tz[-2:])
        from datetime import timedelta
        tzinfo = timezone(sign * timedelta(hours=hh, minutes=mm))
    try:
        return datetime(y, mo, d, h, mi, s, micro, tzinfo=tzinfo).timestamp()
    except ValueError:
        return 0.0


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
        renew = parse_microtime(spec.get("renewTime"))
        # Honor the HOLDER's advertised duration when it differs from ours
        # (client-go semantics: a candidate must respect the incumbent's
        # leaseDurationSeconds, not its own config).
        duration = float(spec.get("leaseDurationSeconds") or self.lease_duration)
        expired = now - renew > duration
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
        taking_over = (previous or {}).get("holderIdentity") != self.identity
        if previous and taking_over:
            transitions += 1
        if previous and not taking_over and previous.get("acquireTime"):
            acquire = previous["acquireTime"]
            if isinstance(acquire, (int, float)):  # r1-format lease: rewrite
                acquire = format_microtime(float(acquire))
        else:
            acquire = format_microtime(now)
        return {
            "holderIdentity": self.identity,
            "leaseDurationSeconds": int(self.lease_duration),
            "acquireTime": acquire,
            "renewTime": format_microtime(now),
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
                # client-go releases by clearing the holder; renewTime stays
                # a valid MicroTime (a real apiserver rejects non-RFC3339).
                lease["spec"]["holderIdentity"] = ""
                self.client.update_lease(self.namespace, lease)
        except Exception:
            log.debug("lease release failed", exc_info=True)
        self.is_leader = False
