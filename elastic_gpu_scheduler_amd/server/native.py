"""Native-server front end: C++ HTTP server with the Python app as fallback.

serve_native(registry, ...) starts the in-process C++ HTTP/1.1 server
(csrc/httpd) sharing the default scheduler's ClusterState. filter and
priorities are answered entirely in C++ (no GIL) once a node is warm in the
cache; bind, status, version, metrics, debug — and any request the C++ path
can't serve (cold nodes, malformed JSON) — fall back to ExtenderApp.handle.
"""
from __future__ import annotations

from typing import Optional

from elastic_gpu_scheduler_amd._native import core
from elastic_gpu_scheduler_amd.server.app import ExtenderApp


class NativeFrontend:
    def __init__(self, app: ExtenderApp, host: str = "0.0.0.0", port: int = 0,
                 tls_cert: str = "", tls_key: str = "",
                 tls_client_ca: str = ""):
        """tls_cert/tls_key serve HTTPS (extender enableHTTPS) with OpenSSL
        terminating TLS inside the C++ server — the GIL-free fast path is
        kept (r1 silently fell back to uvicorn under TLS).
        tls_client_ca additionally requires verified client certs (mTLS)."""
        self.app = app
        sch = app.registry.default
        self.server = core.NativeExtenderServer(
            state=sch.state, bare_unit=sch.bare_unit, host=host, port=port,
            fallback=self._fallback, tls_cert=tls_cert, tls_key=tls_key,
            tls_client_ca=tls_client_ca)
        app.native_server = self.server

    def _fallback(self, method: str, path: str, body: bytes):
        return self.app.handle(method, path, body)

    def start(self) -> None:
        self.server.start()

    def stop(self) -> None:
        self.server.stop()

    @property
    def port(self) -> int:
        return self.server.port

    def stats(self) -> dict:
        return dict(self.server.stats())


def serve_native(registry, host: str = "0.0.0.0", port: int = 0,
                 app: Optional[ExtenderApp] = None) -> NativeFrontend:
    app = app or ExtenderApp(registry)
    fe = NativeFrontend(app, host, port)
    fe.start()
    return fe
