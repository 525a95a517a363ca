"""Offline test infrastructure: the strict wire-format apiserver mock.

There is no kind/kube-apiserver binary in the build environment, so the
"run against a real control plane" requirement is met the envtest way: a
real HTTPS server that ENFORCES apiserver wire formats (MicroTime leases,
typed Status errors, resourceVersion optimistic concurrency, watch framing
with bookmarks and 410 Gone, mTLS/Bearer auth) instead of the permissive
in-memory fake. Wire-format bugs that only a live apiserver would catch —
like r1's unix-float Lease renewTime — fail loudly here.

Lazy exports: `python -m elastic_gpu_scheduler_amd.testing.strict_apiserver`
runs the server as its own process (bench use), and an eager import here
would double-import the module under runpy.
"""
from __future__ import annotations

__all__ = ["StrictAPIServer", "generate_pki"]


def __getattr__(name: str):
    if name in __all__:
        from elastic_gpu_scheduler_amd.testing import strict_apiserver as m

        return getattr(m, name)
    raise AttributeError(name)
