"""A strict wire-format Kubernetes apiserver mock, served over real TLS.

Purpose ("close the real-control-plane gap"): the in-memory
FakeKubeClient accepts anything dict-shaped, so wire-format bugs — the
r1 Lease renewTime written as a unix float, missing client-cert auth —
survived every test and would only have died against a live apiserver.
There is no kube-apiserver/etcd/kind binary in this environment, so this
module is the envtest-style stand-in: a real HTTPS server that speaks the
apiserver's REST surface and REJECTS malformed writes the way a live
apiserver does:

  * coordination.k8s.io Lease: spec.renewTime/acquireTime must be RFC3339
    MicroTime STRINGS ("2026-09-14T10:11:12.123456Z"); numbers are a 400
    decode error exactly like metav1.MicroTime unmarshalling;
  * optimistic concurrency: PUT with a stale metadata.resourceVersion is a
    409 with a typed Status body (reason=Conflict, the real "object has
    been modified" message) — not a bare string;
  * every error is a v1.Status object; every create stamps uid,
    creationTimestamp (RFC3339) and a fresh resourceVersion;
  * watch: newline-delimited JSON event framing over chunked HTTP/1.1,
    resourceVersion resumption, BOOKMARK events, and a Status-410 ERROR
    event when the requested RV has been compacted away;
  * auth: Bearer token or mTLS client certificate (CERT_OPTIONAL TLS with
    an issued client cert — what a kind kubeconfig carries); anonymous
    requests are 401;
  * Content-Type enforcement: JSON bodies and the two merge-patch types
    only (415 otherwise).

Reference behavior being stood in for: the reference's client-go bootstrap
and informers (pkg/utils/utils.go:44-68, pkg/controller/controller.go)
talk to exactly this surface.

Test hooks: `drop_watches()` severs live watch streams (reconnect/resume
testing), `compact()` forgets old events (410 testing), and the internal
`seed_*` helpers mutate state while emitting real watch events.
"""
from __future__ import annotations

import json
import re
import ssl
import subprocess
import threading
import time
import uuid
from datetime import datetime, timezone
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from pathlib import Path
from typing import Any, Dict, List, Optional, Tuple

_MICROTIME_RE = re.compile(
    r"^\d{4}-\d{2}-\d{2}[Tt]\d{2}:\d{2}:\d{2}(?:\.\d{1,9})?([Zz]|[+-]\d{2}:?\d{2})$")

JSON_CT = "application/json"
PATCH_CTS = ("application/strategic-merge-patch+json",
             "application/merge-patch+json")


def _now_rfc3339() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def _status(code: int, reason: str, message: str) -> Dict[str, Any]:
    return {"kind": "Status", "apiVersion": "v1", "metadata": {},
            "status": "Failure", "message": message, "reason": reason,
            "code": code}


class _ApiError(Exception):
    def __init__(self, code: int, reason: str, message: str) -> None:
        super().__init__(message)
        self.body = _status(code, reason, message)
        self.code = code


def _deep_merge(dst: Dict[str, Any], patch: Dict[str, Any]) -> None:
    """Merge-patch semantics (sufficient for the annotation/status patches
    the scheduler and agent send; null deletes a key)."""
    for k, v in patch.items():
        if v is None:
            dst.pop(k, None)
        elif isinstance(v, dict) and isinstance(dst.get(k), dict):
            _deep_merge(dst[k], v)
        else:
            dst[k] = v


def generate_pki(directory: "str | Path") -> Dict[str, str]:
    """Generate a CA, a server cert for 127.0.0.1/localhost, and a client
    cert signed by the CA (openssl CLI; no network). Returns file paths."""
    d = Path(directory)
    d.mkdir(parents=True, exist_ok=True)
    ca_key, ca_crt = d / "ca.key", d / "ca.crt"
    srv_key, srv_csr, srv_crt = d / "server.key", d / "server.csr", d / "server.crt"
    cli_key, cli_csr, cli_crt = d / "client.key", d / "client.csr", d / "client.crt"
    ext = d / "san.ext"
    ext.write_text("subjectAltName=IP:127.0.0.1,DNS:localhost\n")

    def run(*args: str) -> None:
        subprocess.run(list(args), check=True, capture_output=True)

    run("openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
        "-keyout", str(ca_key), "-out", str(ca_crt), "-days", "2",
        "-subj", "/CN=egs-strict-ca")
    run("openssl", "req", "-newkey", "rsa:2048", "-nodes",
        "-keyout", str(srv_key), "-out", str(srv_csr), "-subj", "/CN=127.0.0.1")
    run("openssl", "x509", "-req", "-in", str(srv_csr), "-CA", str(ca_crt),
        "-CAkey", str(ca_key), "-CAcreateserial", "-out", str(srv_crt),
        "-days", "2", "-extfile", str(ext))
    run("openssl", "req", "-newkey", "rsa:2048", "-nodes",
        "-keyout", str(cli_key), "-out", str(cli_csr),
        "-subj", "/CN=egs-e2e-client/O=system:masters")
    run("openssl", "x509", "-req", "-in", str(cli_csr), "-CA", str(ca_crt),
        "-CAkey", str(ca_key), "-CAcreateserial", "-out", str(cli_crt),
        "-days", "2")
    return {"ca_crt": str(ca_crt), "ca_key": str(ca_key),
            "server_crt": str(srv_crt), "server_key": str(srv_key),
            "client_crt": str(cli_crt), "client_key": str(cli_key)}


class StrictAPIServer:
    """See module docstring. Start with `start()`; `base_url` is https://."""

    def __init__(self, pki: Dict[str, str], token: Optional[str] = None,
                 host: str = "127.0.0.1", port: int = 0,
                 event_retention: int = 4096,
                 bookmark_interval: float = 0.25) -> None:
        self.pki = pki
        self.token = token
        self.event_retention = event_retention
        self.bookmark_interval = bookmark_interval

        self._mu = threading.RLock()
        self._watch_cv = threading.Condition(self._mu)
        self._rv = 100
        self._pods: Dict[str, Dict[str, Any]] = {}
        self._nodes: Dict[str, Dict[str, Any]] = {}
        self._leases: Dict[str, Dict[str, Any]] = {}
        self.events: List[Dict[str, Any]] = []
        # watch log: (rv, type, kind, snapshot) — compacted to event_retention
        self._log: List[Tuple[int, str, str, Dict[str, Any]]] = []
        self._watch_gen = 0  # bumped by drop_watches()
        self.pause_watches = False  # test hook: refuse new watch streams
        self.request_counts: Dict[str, int] = {}
        # chaos hook: probability that a MUTATING request (POST/PUT/PATCH/
        # DELETE, except watch) fails with a 500 BEFORE touching state —
        # models a flaky apiserver for fault-injection e2e.
        self.fault_rate = 0.0
        self._fault_seq = 0

        outer = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"
            disable_nagle_algorithm = True  # else ~40 ms delayed-ACK stalls
            wbufsize = 65536

            # silence default stderr logging
            def log_message(self, fmt, *args):  # noqa: D401
                pass

            def _authorized(self) -> bool:
                try:
                    peer = self.connection.getpeercert()
                except (ssl.SSLError, OSError, AttributeError):
                    peer = None
                if peer:  # any cert accepted by the CA-verified context
                    return True
                auth = self.headers.get("Authorization", "")
                return bool(outer.token) and auth == f"Bearer {outer.token}"

            def _send_json(self, code: int, obj: Dict[str, Any]) -> None:
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", JSON_CT)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _read_body(self) -> bytes:
                length = int(self.headers.get("Content-Length", "0") or "0")
                return self.rfile.read(length) if length else b""

            def _dispatch(self, method: str) -> None:
                outer.request_counts[method] = \
                    outer.request_counts.get(method, 0) + 1
                if not self._authorized():
                    self._send_json(401, _status(
                        401, "Unauthorized",
                        "Unauthorized: no valid bearer token or client "
                        "certificate"))
                    return
                try:
                    outer._route(self, method)
                except _ApiError as e:
                    self._send_json(e.code, e.body)
                except (BrokenPipeError, ConnectionResetError):
                    pass
                except Exception as e:  # noqa: BLE001 — surface as 500 Status
                    self._send_json(500, _status(
                        500, "InternalError", f"{type(e).__name__}: {e}"))

            def do_GET(self):
                self._dispatch("GET")

            def do_POST(self):
                self._dispatch("POST")

            def do_PUT(self):
                self._dispatch("PUT")

            def do_PATCH(self):
                self._dispatch("PATCH")

            def do_DELETE(self):
                self._dispatch("DELETE")

        class Server(ThreadingHTTPServer):
            daemon_threads = True

            def handle_error(self, request, client_address):
                pass  # TLS handshake failures from unauthorized probes

        self._httpd = Server((host, port), Handler)
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(pki["server_crt"], pki["server_key"])
        ctx.load_verify_locations(pki["ca_crt"])
        ctx.verify_mode = ssl.CERT_OPTIONAL  # mTLS when the client offers one
        self._httpd.socket = ctx.wrap_socket(self._httpd.socket,
                                             server_side=True)
        self._thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    # lifecycle
    @property
    def port(self) -> int:
        return self._httpd.server_address[1]

    @property
    def base_url(self) -> str:
        return f"https://127.0.0.1:{self.port}"

    def start(self) -> "StrictAPIServer":
        self._thread = threading.Thread(target=self._httpd.serve_forever,
                                        name="strict-apiserver", daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        self.drop_watches()
        self._httpd.shutdown()
        self._httpd.server_close()
        if self._thread:
            self._thread.join(timeout=5)

    # test hooks ---------------------------------------------------------
    def drop_watches(self) -> None:
        """Sever every live watch stream (tests reconnect/resume)."""
        with self._watch_cv:
            self._watch_gen += 1
            self._watch_cv.notify_all()

    def compact(self) -> None:
        """Forget the retained watch log: the next resume from an old RV
        gets a Status-410 ERROR event (etcd compaction analogue)."""
        with self._watch_cv:
            self._log.clear()

    # state helpers ------------------------------------------------------
    def _next_rv_locked(self) -> str:
        self._rv += 1
        return str(self._rv)

    def _emit_locked(self, etype: str, obj: Dict[str, Any],
                     kind: str = "Pod") -> None:
        self._log.append((self._rv, etype, kind, json.loads(json.dumps(obj))))
        if len(self._log) > self.event_retention:
            self._log = self._log[-self.event_retention:]
        self._watch_cv.notify_all()

    def _stamp_new_locked(self, obj: Dict[str, Any],
                          namespace: Optional[str]) -> None:
        meta = obj.setdefault("metadata", {})
        if namespace:
            meta.setdefault("namespace", namespace)
        meta.setdefault("uid", str(uuid.uuid4()))
        meta.setdefault("creationTimestamp", _now_rfc3339())
        meta["resourceVersion"] = self._next_rv_locked()

    # seeding (direct state mutation WITH real watch events) -------------
    def seed_node(self, node: Dict[str, Any]) -> Dict[str, Any]:
        with self._mu:
            node = json.loads(json.dumps(node))
            self._stamp_new_locked(node, None)
            self._nodes[node["metadata"]["name"]] = node
            self._emit_locked("ADDED", node, kind="Node")
            return json.loads(json.dumps(node))

    def seed_pod(self, pod: Dict[str, Any]) -> Dict[str, Any]:
        with self._mu:
            pod = json.loads(json.dumps(pod))
            self._stamp_new_locked(pod, "default")
            key = f"{pod['metadata']['namespace']}/{pod['metadata']['name']}"
            self._pods[key] = pod
            self._emit_locked("ADDED", pod)
            return json.loads(json.dumps(pod))

    def seed_delete_pod(self, namespace: str, name: str) -> None:
        with self._mu:
            pod = self._pods.pop(f"{namespace}/{name}", None)
            if pod is not None:
                self._next_rv_locked()
                pod["metadata"]["resourceVersion"] = str(self._rv)
                self._emit_locked("DELETED", pod)

    def seed_pod_phase(self, namespace: str, name: str, phase: str) -> None:
        with self._mu:
            pod = self._pods[f"{namespace}/{name}"]
            pod.setdefault("status", {})["phase"] = phase
            pod["metadata"]["resourceVersion"] = self._next_rv_locked()
            self._emit_locked("MODIFIED", pod)

    def pod(self, namespace: str, name: str) -> Dict[str, Any]:
        with self._mu:
            return json.loads(json.dumps(self._pods[f"{namespace}/{name}"]))

    def lease(self, namespace: str, name: str) -> Dict[str, Any]:
        with self._mu:
            return json.loads(json.dumps(self._leases[f"{namespace}/{name}"]))

    # ------------------------------------------------------------------
    # strict validation
    @staticmethod
    def _require_json_ct(h, allow_patch: bool = False) -> None:
        ct = (h.headers.get("Content-Type") or "").split(";")[0].strip()
        if allow_patch:
            if ct not in PATCH_CTS:
                raise _ApiError(
                    415, "UnsupportedMediaType",
                    f"unsupported patch content type {ct!r}; supported: "
                    f"{', '.join(PATCH_CTS)}")
            return
        if ct != JSON_CT:
            raise _ApiError(415, "UnsupportedMediaType",
                            f"content type {ct!r} is not {JSON_CT}")

    @staticmethod
    def _parse_json(raw: bytes) -> Dict[str, Any]:
        try:
            obj = json.loads(raw.decode() or "null")
        except (ValueError, UnicodeDecodeError) as e:
            raise _ApiError(400, "BadRequest",
                            f"the object provided is unrecognized: {e}")
        if not isinstance(obj, dict):
            raise _ApiError(400, "BadRequest", "expected a JSON object body")
        return obj

    @staticmethod
    def _validate_lease(lease: Dict[str, Any]) -> None:
        """metav1.MicroTime decodes ONLY RFC3339 strings — a numeric
        renewTime (the r1 bug) is a 400, exactly like a real apiserver."""
        spec = lease.get("spec", {}) or {}
        for field in ("renewTime", "acquireTime"):
            v = spec.get(field)
            if v is None:
                continue
            if not isinstance(v, str):
                raise _ApiError(
                    400, "BadRequest",
                    f"v1.LeaseSpec.{field}: decode: could not decode "
                    f"{type(v).__name__} into metav1.MicroTime (expected "
                    "RFC3339 string)")
            if not _MICROTIME_RE.match(v):
                raise _ApiError(
                    400, "BadRequest",
                    f"v1.LeaseSpec.{field}: parsing time {v!r}: invalid "
                    "RFC3339 timestamp")
        for field in ("leaseDurationSeconds", "leaseTransitions"):
            v = spec.get(field)
            if v is not None and not isinstance(v, int):
                raise _ApiError(
                    400, "BadRequest",
                    f"v1.LeaseSpec.{field}: decode: could not decode "
                    f"{type(v).__name__} into int32")
        holder = spec.get("holderIdentity")
        if holder is not None and not isinstance(holder, str):
            raise _ApiError(400, "BadRequest",
                            "v1.LeaseSpec.holderIdentity: expected string")

    # ------------------------------------------------------------------
    # routing
    _POD = re.compile(r"^/api/v1/namespaces/([^/]+)/pods/([^/]+)$")
    _POD_BIND = re.compile(r"^/api/v1/namespaces/([^/]+)/pods/([^/]+)/binding$")
    _PODS_NS = re.compile(r"^/api/v1/namespaces/([^/]+)/pods$")
    _NODE = re.compile(r"^/api/v1/nodes/([^/]+)$")
    _NODE_STATUS = re.compile(r"^/api/v1/nodes/([^/]+)/status$")
    _EVENTS = re.compile(r"^/api/v1/namespaces/([^/]+)/events$")
    _LEASES = re.compile(
        r"^/apis/coordination\.k8s\.io/v1/namespaces/([^/]+)/leases$")
    _LEASE = re.compile(
        r"^/apis/coordination\.k8s\.io/v1/namespaces/([^/]+)/leases/([^/]+)$")

    def _route(self, h, method: str) -> None:
        if self.fault_rate > 0 and method in ("POST", "PUT", "PATCH",
                                              "DELETE"):
            # deterministic pseudo-random sequence so runs are replayable
            with self._mu:
                self._fault_seq += 1
                seq = self._fault_seq
            if (seq * 2654435761 % 997) / 997.0 < self.fault_rate:
                raise _ApiError(500, "InternalError",
                                "etcdserver: request timed out (injected)")
        path, _, query = h.path.partition("?")
        params: Dict[str, str] = {}
        for part in query.split("&"):
            if "=" in part:
                k, _, v = part.partition("=")
                from urllib.parse import unquote_plus
                params[unquote_plus(k)] = unquote_plus(v)

        if method == "GET" and path == "/api/v1/pods":
            if params.get("watch") == "true":
                self._serve_watch(h, params)
            else:
                self._serve_pod_list(h, params)
            return
        if method == "GET" and path == "/api/v1/nodes":
            if params.get("watch") == "true":
                self._serve_watch(h, params, kind="Node")
                return
            with self._mu:
                items = [json.loads(json.dumps(n))
                         for n in self._nodes.values()]
                rv = str(self._rv)
            h._send_json(200, {"kind": "NodeList", "apiVersion": "v1",
                               "metadata": {"resourceVersion": rv},
                               "items": items})
            return
        if method == "POST" and path == "/api/v1/nodes":
            self._require_json_ct(h)
            node = self._parse_json(h._read_body())
            name = node.get("metadata", {}).get("name")
            if not name:
                raise _ApiError(422, "Invalid", "metadata.name is required")
            with self._mu:
                if name in self._nodes:
                    raise _ApiError(409, "AlreadyExists",
                                    f'nodes "{name}" already exists')
                self._stamp_new_locked(node, None)
                self._nodes[name] = node
                self._emit_locked("ADDED", node, kind="Node")
                h._send_json(201, json.loads(json.dumps(node)))
            return

        m = self._POD_BIND.match(path)
        if m and method == "POST":
            self._require_json_ct(h)
            self._handle_binding(h, m.group(1), m.group(2))
            return
        m = self._POD.match(path)
        if m:
            if method == "GET":
                self._handle_get_pod(h, m.group(1), m.group(2))
            elif method == "PUT":
                self._require_json_ct(h)
                self._handle_put_pod(h, m.group(1), m.group(2))
            elif method == "DELETE":
                self._handle_delete_pod(h, m.group(1), m.group(2))
            else:
                raise _ApiError(405, "MethodNotAllowed", method)
            return
        m = self._PODS_NS.match(path)
        if m and method == "POST":
            self._require_json_ct(h)
            self._handle_create_pod(h, m.group(1))
            return
        m = self._NODE_STATUS.match(path)
        if m and method == "PATCH":
            self._require_json_ct(h, allow_patch=True)
            self._handle_patch_node(h, m.group(1))
            return
        m = self._NODE.match(path)
        if m:
            if method == "GET":
                with self._mu:
                    node = self._nodes.get(m.group(1))
                    if node is None:
                        raise _ApiError(404, "NotFound",
                                        f'nodes "{m.group(1)}" not found')
                    h._send_json(200, json.loads(json.dumps(node)))
            elif method == "PATCH":
                self._require_json_ct(h, allow_patch=True)
                self._handle_patch_node(h, m.group(1))
            else:
                raise _ApiError(405, "MethodNotAllowed", method)
            return
        m = self._EVENTS.match(path)
        if m and method == "POST":
            self._require_json_ct(h)
            event = self._parse_json(h._read_body())
            with self._mu:
                self._stamp_new_locked(event, m.group(1))
                self.events.append(event)
            h._send_json(201, event)
            return
        m = self._LEASES.match(path)
        if m and method == "POST":
            self._require_json_ct(h)
            self._handle_create_lease(h, m.group(1))
            return
        m = self._LEASE.match(path)
        if m:
            if method == "GET":
                self._handle_get_lease(h, m.group(1), m.group(2))
            elif method == "PUT":
                self._require_json_ct(h)
                self._handle_put_lease(h, m.group(1), m.group(2))
            else:
                raise _ApiError(405, "MethodNotAllowed", method)
            return
        raise _ApiError(404, "NotFound", f"no route for {method} {path}")

    # ------------------------------------------------------------------
    # pods
    @staticmethod
    def _match_selectors(pod: Dict[str, Any], params: Dict[str, str]) -> bool:
        sel = params.get("labelSelector")
        if sel:
            labels = pod.get("metadata", {}).get("labels", {}) or {}
            for clause in sel.split(","):
                k, _, v = clause.partition("=")
                if labels.get(k) != v:
                    return False
        fsel = params.get("fieldSelector")
        if fsel:
            for clause in fsel.split(","):
                k, _, v = clause.partition("=")
                if k == "spec.nodeName":
                    if (pod.get("spec", {}) or {}).get("nodeName", "") != v:
                        return False
                elif k == "metadata.name":
                    if pod.get("metadata", {}).get("name") != v:
                        return False
        return True

    def _serve_pod_list(self, h, params: Dict[str, str]) -> None:
        with self._mu:
            items = [json.loads(json.dumps(p)) for p in self._pods.values()
                     if self._match_selectors(p, params)]
            rv = str(self._rv)
        h._send_json(200, {"kind": "PodList", "apiVersion": "v1",
                           "metadata": {"resourceVersion": rv},
                           "items": items})

    def _handle_get_pod(self, h, ns: str, name: str) -> None:
        with self._mu:
            pod = self._pods.get(f"{ns}/{name}")
            if pod is None:
                raise _ApiError(404, "NotFound",
                                f'pods "{name}" not found')
            h._send_json(200, json.loads(json.dumps(pod)))

    def _handle_create_pod(self, h, ns: str) -> None:
        pod = self._parse_json(h._read_body())
        name = pod.get("metadata", {}).get("name")
        if not name:
            raise _ApiError(422, "Invalid", "metadata.name is required")
        with self._mu:
            key = f"{ns}/{name}"
            if key in self._pods:
                raise _ApiError(409, "AlreadyExists",
                                f'pods "{name}" already exists')
            pod.setdefault("metadata", {})["namespace"] = ns
            self._stamp_new_locked(pod, ns)
            self._pods[key] = pod
            out = json.loads(json.dumps(pod))
            self._emit_locked("ADDED", pod)
        h._send_json(201, out)

    def _handle_put_pod(self, h, ns: str, name: str) -> None:
        pod = self._parse_json(h._read_body())
        with self._mu:
            key = f"{ns}/{name}"
            current = self._pods.get(key)
            if current is None:
                raise _ApiError(404, "NotFound", f'pods "{name}" not found')
            sent_rv = pod.get("metadata", {}).get("resourceVersion")
            cur_rv = current["metadata"]["resourceVersion"]
            if sent_rv != cur_rv:
                raise _ApiError(
                    409, "Conflict",
                    f'Operation cannot be fulfilled on pods "{name}": the '
                    "object has been modified; please apply your changes to "
                    "the latest version and try again")
            # immutable fields a real apiserver enforces
            if pod.get("metadata", {}).get("uid") not in (
                    None, current["metadata"].get("uid")):
                raise _ApiError(422, "Invalid", "metadata.uid is immutable")
            pod["metadata"]["uid"] = current["metadata"].get("uid")
            pod["metadata"]["namespace"] = ns
            pod["metadata"].setdefault(
                "creationTimestamp", current["metadata"].get("creationTimestamp"))
            pod["metadata"]["resourceVersion"] = self._next_rv_locked()
            self._pods[key] = pod
            out = json.loads(json.dumps(pod))
            self._emit_locked("MODIFIED", pod)
        h._send_json(200, out)

    def _handle_delete_pod(self, h, ns: str, name: str) -> None:
        with self._mu:
            pod = self._pods.pop(f"{ns}/{name}", None)
            if pod is None:
                raise _ApiError(404, "NotFound", f'pods "{name}" not found')
            pod["metadata"]["resourceVersion"] = self._next_rv_locked()
            self._emit_locked("DELETED", pod)
        h._send_json(200, _status(200, "", "Success") | {"status": "Success"})

    def _handle_binding(self, h, ns: str, name: str) -> None:
        binding = self._parse_json(h._read_body())
        target = binding.get("target", {}) or {}
        if target.get("kind") != "Node" or not target.get("name"):
            raise _ApiError(422, "Invalid",
                            "binding.target must be a Node with a name")
        with self._mu:
            pod = self._pods.get(f"{ns}/{name}")
            if pod is None:
                raise _ApiError(404, "NotFound", f'pods "{name}" not found')
            if target["name"] not in self._nodes:
                raise _ApiError(404, "NotFound",
                                f'nodes "{target["name"]}" not found')
            existing = (pod.get("spec", {}) or {}).get("nodeName")
            if existing and existing != target["name"]:
                raise _ApiError(
                    409, "Conflict",
                    f'Operation cannot be fulfilled on pods/binding "{name}": '
                    f"pod {name} is already assigned to node {existing}")
            pod.setdefault("spec", {})["nodeName"] = target["name"]
            pod["metadata"]["resourceVersion"] = self._next_rv_locked()
            self._emit_locked("MODIFIED", pod)
        h._send_json(201, _status(201, "", "Success") | {"status": "Success"})

    # nodes ---------------------------------------------------------------
    def _handle_patch_node(self, h, name: str) -> None:
        patch = self._parse_json(h._read_body())
        with self._mu:
            node = self._nodes.get(name)
            if node is None:
                raise _ApiError(404, "NotFound", f'nodes "{name}" not found')
            _deep_merge(node, patch)
            node["metadata"]["resourceVersion"] = self._next_rv_locked()
            self._emit_locked("MODIFIED", node, kind="Node")
            h._send_json(200, json.loads(json.dumps(node)))

    # leases --------------------------------------------------------------
    def _handle_get_lease(self, h, ns: str, name: str) -> None:
        with self._mu:
            lease = self._leases.get(f"{ns}/{name}")
            if lease is None:
                raise _ApiError(
                    404, "NotFound",
                    f'leases.coordination.k8s.io "{name}" not found')
            h._send_json(200, json.loads(json.dumps(lease)))

    def _handle_create_lease(self, h, ns: str) -> None:
        lease = self._parse_json(h._read_body())
        self._validate_lease(lease)
        name = lease.get("metadata", {}).get("name")
        if not name:
            raise _ApiError(422, "Invalid", "metadata.name is required")
        with self._mu:
            key = f"{ns}/{name}"
            if key in self._leases:
                raise _ApiError(
                    409, "AlreadyExists",
                    f'leases.coordination.k8s.io "{name}" already exists')
            lease.setdefault("metadata", {})["namespace"] = ns
            self._stamp_new_locked(lease, ns)
            self._leases[key] = lease
            h._send_json(201, json.loads(json.dumps(lease)))

    def _handle_put_lease(self, h, ns: str, name: str) -> None:
        lease = self._parse_json(h._read_body())
        self._validate_lease(lease)
        with self._mu:
            key = f"{ns}/{name}"
            current = self._leases.get(key)
            if current is None:
                raise _ApiError(
                    404, "NotFound",
                    f'leases.coordination.k8s.io "{name}" not found')
            if lease.get("metadata", {}).get("resourceVersion") != \
                    current["metadata"]["resourceVersion"]:
                raise _ApiError(
                    409, "Conflict",
                    f'Operation cannot be fulfilled on '
                    f'leases.coordination.k8s.io "{name}": the object has '
                    "been modified; please apply your changes to the latest "
                    "version and try again")
            lease["metadata"]["namespace"] = ns
            lease["metadata"]["resourceVersion"] = self._next_rv_locked()
            self._leases[key] = lease
            h._send_json(200, json.loads(json.dumps(lease)))

    # watch ---------------------------------------------------------------
    def _serve_watch(self, h, params: Dict[str, str],
                     kind: str = "Pod") -> None:
        if self.pause_watches:
            raise _ApiError(503, "ServiceUnavailable",
                            "watch refused (test pause)")
        since: Optional[int] = None
        if params.get("resourceVersion"):
            try:
                since = int(params["resourceVersion"])
            except ValueError:
                raise _ApiError(400, "BadRequest",
                                "resourceVersion: invalid value")
        bookmarks = params.get("allowWatchBookmarks") == "true"

        h.send_response(200)
        h.send_header("Content-Type", JSON_CT)
        h.send_header("Transfer-Encoding", "chunked")
        h.end_headers()

        def chunk(obj: Dict[str, Any]) -> None:
            data = (json.dumps(obj) + "\n").encode()
            h.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
            h.wfile.flush()

        def end() -> None:
            try:
                h.wfile.write(b"0\r\n\r\n")
                h.wfile.flush()
            except (BrokenPipeError, ConnectionResetError, OSError):
                pass

        with self._watch_cv:
            gen = self._watch_gen
            too_old = False
            if since is not None and since != 0:
                oldest = self._log[0][0] if self._log else self._rv + 1
                too_old = since < oldest - 1 and since < self._rv
            cursor = since if since not in (None, 0) else self._rv
        if too_old:
            # compacted away: Status 410, the "too old" signal
            try:
                chunk({"type": "ERROR",
                       "object": _status(
                           410, "Expired",
                           f"too old resource version: {since}")
                       | {"status": "Failure"}})
            except (BrokenPipeError, ConnectionResetError, OSError):
                pass
            end()
            return
        last_bookmark = time.monotonic()
        try:
            while True:
                # Collect pending events UNDER the lock; write them to the
                # (possibly slow) client socket OUTSIDE it — a stalled watch
                # reader must never block apiserver writes.
                with self._watch_cv:
                    if gen != self._watch_gen:
                        break
                    pending = [(rv, etype, snapshot)
                               for rv, etype, k, snapshot in self._log
                               if rv > cursor and k == kind]
                    if not pending:
                        self._watch_cv.wait(timeout=self.bookmark_interval)
                        pending = [(rv, etype, snapshot)
                                   for rv, etype, k, snapshot in self._log
                                   if rv > cursor and k == kind]
                        if gen != self._watch_gen:
                            break
                    rv_now = self._rv
                for rv, etype, snapshot in pending:
                    chunk({"type": etype, "object": snapshot})
                    cursor = rv
                if bookmarks and \
                        time.monotonic() - last_bookmark >= \
                        self.bookmark_interval:
                    chunk({"type": "BOOKMARK",
                           "object": {"kind": kind, "apiVersion": "v1",
                                      "metadata": {"resourceVersion":
                                                   str(rv_now)}}})
                    cursor = max(cursor, rv_now)
                    last_bookmark = time.monotonic()
        except (BrokenPipeError, ConnectionResetError, OSError):
            return  # client went away
        end()  # dropped by drop_watches(): clean end-of-stream


def main(argv=None) -> int:
    """Run the strict apiserver as its OWN process (bench/e2e use): a real
    apiserver is out-of-process, and keeping it in-process would serialize
    it behind the scheduler's GIL, measuring the mock instead of the
    scheduler. Prints one line `READY <port>` on stdout when serving."""
    import argparse
    import sys

    p = argparse.ArgumentParser(description=main.__doc__)
    p.add_argument("--pki", required=True,
                   help="directory with/for CA+server+client certs "
                        "(generated if missing)")
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--token", default=None)
    args = p.parse_args(argv)

    d = Path(args.pki)
    if not (d / "server.crt").exists():
        pki = generate_pki(d)
    else:
        pki = {"ca_crt": str(d / "ca.crt"), "ca_key": str(d / "ca.key"),
               "server_crt": str(d / "server.crt"),
               "server_key": str(d / "server.key"),
               "client_crt": str(d / "client.crt"),
               "client_key": str(d / "client.key")}
    server = StrictAPIServer(pki, token=args.token, port=args.port).start()
    print(f"READY {server.port}", flush=True)
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        pass
    finally:
        server.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
