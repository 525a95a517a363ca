"""Prometheus metrics (new vs the reference, which has none — SURVEY.md §5).

North-star metrics from BASELINE.json: pods scheduled/sec and filter→bind
latency. Exposed on GET /metrics in the standard text format.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, Optional

from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest)

REGISTRY = CollectorRegistry()

REQUESTS = Counter("egs_requests_total", "Extender HTTP requests",
                   ["verb", "outcome"], registry=REGISTRY)
VERB_LATENCY = Histogram(
    "egs_verb_latency_seconds", "Latency per extender verb", ["verb"],
    buckets=(.0001, .00025, .0005, .001, .0025, .005, .01, .025, .05, .1,
             .25, .5, 1., 2.5),
    registry=REGISTRY)
PODS_SCHEDULED = Counter("egs_pods_scheduled_total",
                         "Pods successfully bound", registry=REGISTRY)
FILTER_TO_BIND = Histogram(
    "egs_filter_to_bind_seconds",
    "Wall time from a pod's first filter to its successful bind",
    buckets=(.0005, .001, .0025, .005, .01, .025, .05, .1, .25, .5, 1., 2.5,
             5., 10.),
    registry=REGISTRY)
NODES_CACHED = Gauge("egs_nodes_cached", "Nodes in the scheduler cache",
                     registry=REGISTRY)


class FilterToBindTracker:
    """Remembers each pod's first filter timestamp to measure the
    filter→bind latency on successful bind. Bounded."""

    def __init__(self, cap: int = 65536) -> None:
        self._mu = threading.Lock()
        self._first_filter: Dict[str, float] = {}
        self._cap = cap

    def saw_filter(self, uid: str) -> None:
        now = time.perf_counter()
        with self._mu:
            if len(self._first_filter) >= self._cap:
                self._first_filter.clear()  # pathological backlog: reset
            self._first_filter.setdefault(uid, now)

    def saw_bind(self, uid: str) -> Optional[float]:
        now = time.perf_counter()
        with self._mu:
            t0 = self._first_filter.pop(uid, None)
        if t0 is None:
            return None
        dt = now - t0
        FILTER_TO_BIND.observe(dt)
        return dt


TRACKER = FilterToBindTracker()


def render() -> bytes:
    return generate_latest(REGISTRY)
