"""Offline test infrastructure: the strict wire-format apiserver mock.

There is no kind/kube-apiserver binary in the build environment, so the
"run against a real control plane" requirement is met the envtest way: a
real HTTPS server that ENFORCES apiserver wire formats (MicroTime leases,
typed Status errors, resourceVersion optimistic concurrency, watch framing
with bookmarks and 410 Gone, mTLS/Bearer auth) instead of the permissive
in-memory fake. Wire-format bugs that only a live apiserver would catch —
like r1's unix-float Lease renewTime — fail loudly here.
"""
from elastic_gpu_scheduler_amd.testing.strict_apiserver import (  # noqa: F401
    StrictAPIServer,
    generate_pki,
)
