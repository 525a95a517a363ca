#!/usr/bin/env python3
"""Flagship benchmark: sustained scheduling throughput of the MI355X-native
extender.

Measures the BASELINE.json headline metric — pods scheduled/sec and p50
filter->bind latency — on BASELINE config #4's workload: batches of 64 mixed
gpu-core/gpu-memory pods scheduled onto 8x-MI355X-288GB nodes, binpack
policy, through the FULL pipeline: real HTTP over TCP (native C++ server by
default; --server uvicorn for the ASGI stack) -> extender filter ->
priorities -> bind (annotation write + binding) -> controller release,
backed by an in-process fake apiserver (there is no cluster on the
bench box; the reference's apiserver round-trips are replaced by the fake's
in-memory writes for BOTH warm and timed phases, stated here so the number
is interpretable).

One step = schedule a full batch of B pods end-to-end, verify a sample of
placements on the local MI355X (HIP stamp kernel), then delete the pods and
wait for the reconcile controller to release every card.

Scaling: weak — each rank runs an independent scheduler shard (its own
fake cluster + HTTP server), one process per GPU; ranks synchronise per
step via torch.distributed (RCCL on GPU, gloo on CPU). `value` is the
whole-job aggregate pods/sec over all ranks, timed as MAX over ranks.

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W] [--batch B]
        [--policy binpack|spread|random] [--no-http] [--no-verify]
For N>1 the driver launches this under torch.distributed.run.
"""
from __future__ import annotations

import argparse
import json
import os
import socket
import statistics
import threading
import time
import uuid

GiB = 1024**3


# ---------------------------------------------------------------- workload

def mixed_pod_spec(i: int, step: int, rank: int, pad_bytes: int = 0):
    """BASELINE config #4 mix: whole-card, half-card, quarter and small
    fractional pods, deterministic by index."""
    k = i % 8
    if k == 0:
        core, mem = 100, 0          # whole card
    elif k in (1, 2):
        core, mem = 50, 96 * GiB
    elif k in (3, 4, 5):
        core, mem = 25, 48 * GiB
    else:
        core, mem = 10, 16 * GiB
    name = f"bench-r{rank}-s{step}-{i}"
    req = {"elasticgpu.io/gpu-core": str(core)}
    if mem:
        req["elasticgpu.io/gpu-memory"] = str(mem)
    pod = {
        "metadata": {"name": name, "namespace": "default",
                     "uid": str(uuid.uuid4())},
        "spec": {"containers": [{"name": "main",
                                 "resources": {"requests": req,
                                               "limits": dict(req)}}]},
        "status": {"phase": "Pending"},
    }
    if pad_bytes > 0:
        # realistic pod objects carry kB of env/volumes/labels; model that
        # wire weight with an opaque annotation
        pod["metadata"]["annotations"] = {"bench.pad": "x" * pad_bytes}
    return pod


def build_cluster(n_nodes: int, cards: int, use_gpu_inventory: bool):
    from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient
    from elastic_gpu_scheduler_amd.utils import types as t

    client = FakeKubeClient()
    template_ann = None
    if use_gpu_inventory:
        # Live MI355X inventory from the HIP probe, extended to an 8-card
        # node template (the bench box exposes 1 card; the node model is the
        # 8x OAM board of BASELINE's configs).
        from elastic_gpu_scheduler_amd.agent.agent import NodeAgent

        agent = NodeAgent("template")
        snap = agent.snapshot()
        card = snap["cards"][0]
        import json as _json

        cards_list = [dict(card, index=i) for i in range(cards)]
        template_ann = {
            t.ANNOTATION_NODE_INVENTORY: _json.dumps({"cards": cards_list}),
            t.ANNOTATION_NODE_TOPOLOGY: _json.dumps(
                {"hops": [[0 if a == b else 1 for b in range(cards)]
                          for a in range(cards)]}),
        }
    for i in range(n_nodes):
        node = {
            "metadata": {"name": f"node-{i}"},
            "status": {"allocatable": {
                t.RESOURCE_GPU_CORE: str(100 * cards),
                t.RESOURCE_GPU_MEMORY: str(t.MI355X_MEMORY_BYTES * cards),
            }},
        }
        if template_ann:
            node["metadata"]["annotations"] = dict(template_ann)
        client.add_node(node)
    return client


# ---------------------------------------------------------------- pipeline

class MiniHttpClient:
    """Minimal blocking HTTP/1.1 keep-alive client over one socket
    (optionally TLS: pass an ssl.SSLContext)."""

    def __init__(self, host: str, port: int, ssl_context=None):
        raw = socket.create_connection((host, port))
        raw.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        if ssl_context is not None:
            self.sock = ssl_context.wrap_socket(raw, server_hostname=host)
        else:
            self.sock = raw
        self.buf = b""

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass

    def post_json(self, path: str, payload) -> "tuple[int, object]":
        body = json.dumps(payload).encode()
        req = (f"POST {path} HTTP/1.1\r\nhost: bench\r\n"
               f"content-type: application/json\r\n"
               f"content-length: {len(body)}\r\n\r\n").encode() + body
        self.sock.sendall(req)
        # read headers
        while b"\r\n\r\n" not in self.buf:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise ConnectionError("server closed connection")
            self.buf += chunk
        head, _, rest = self.buf.partition(b"\r\n\r\n")
        lines = head.split(b"\r\n")
        status = int(lines[0].split()[1])
        clen = None
        chunked = False
        for line in lines[1:]:
            k, _, v = line.partition(b":")
            key = k.strip().lower()
            if key == b"content-length":
                clen = int(v.strip())
            elif key == b"transfer-encoding" and b"chunked" in v.lower():
                chunked = True
        if chunked:
            body, rest = self._read_chunked(rest)
            self.buf = rest
            return status, json.loads(body or b"{}")
        clen = clen or 0
        while len(rest) < clen:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise ConnectionError("server closed connection")
            rest += chunk
        self.buf = rest[clen:]
        return status, json.loads(rest[:clen] or b"{}")

    def _read_chunked(self, rest: bytes) -> "tuple[bytes, bytes]":
        body = b""
        while True:
            while b"\r\n" not in rest:
                chunk = self.sock.recv(65536)
                if not chunk:
                    raise ConnectionError("server closed connection")
                rest += chunk
            size_line, _, rest = rest.partition(b"\r\n")
            size = int(size_line.split(b";")[0], 16)
            while len(rest) < size + 2:
                chunk = self.sock.recv(65536)
                if not chunk:
                    raise ConnectionError("server closed connection")
                rest += chunk
            body += rest[:size]
            rest = rest[size + 2:]  # skip trailing CRLF
            if size == 0:
                return body, rest




class BenchPipeline:
    def __init__(self, rank: int, args, device_index: int, use_gpu: bool):
        from elastic_gpu_scheduler_amd.controller.controller import Controller
        from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
        from elastic_gpu_scheduler_amd.server.app import make_app

        self.rank = rank
        self.args = args
        self.use_gpu = use_gpu
        self.device_index = device_index
        self.client = build_cluster(args.nodes, args.cards, use_gpu)
        self.registry = SchedulerRegistry(self.client, policy=args.policy,
                                          threads=args.filter_threads)
        self.controller = Controller(self.client, self.registry, workers=2,
                                     resync_seconds=3600)
        self.controller.start()
        self.app = make_app(self.registry)
        self.node_names = [f"node-{i}" for i in range(args.nodes)]
        self.latencies: list[float] = []
        self._lat_mu = threading.Lock()
        self.bind_retries = 0
        self.port = None
        self._server = None
        self._server_thread = None
        self._native = None
        self.base_url = None
        self._tls_dir = None
        self._client_ssl = None
        if not args.no_http:
            self._start_server()
        if use_gpu and not args.no_verify:
            from elastic_gpu_scheduler_amd._native import gpuprobe

            self.probe = gpuprobe()  # fails loudly if the HIP ext is missing
        else:
            self.probe = None

    def _start_server(self):
        tls_kw = {}
        if self.args.tls:
            # full pipeline over HTTPS: OpenSSL terminates inside the C++
            # server; the load generator speaks TLS on every connection
            import ssl
            import tempfile

            from elastic_gpu_scheduler_amd.testing import generate_pki

            self._tls_dir = tempfile.TemporaryDirectory(prefix="egs-bench-tls-")
            pki = generate_pki(self._tls_dir.name)
            tls_kw = {"tls_cert": pki["server_crt"],
                      "tls_key": pki["server_key"]}
            self._client_ssl = ssl.create_default_context(
                cafile=pki["ca_crt"])
        if self.args.server == "native":
            from elastic_gpu_scheduler_amd.server.native import NativeFrontend

            self._native = NativeFrontend(self.app, host="127.0.0.1", port=0,
                                          **tls_kw)
            self._native.start()
            self.port = self._native.port
            self.base_url = f"http://127.0.0.1:{self.port}"
            return
        if tls_kw:
            raise SystemExit("--tls requires --server native in the bench")
        import uvicorn

        port = self._free_port()
        config = uvicorn.Config(self.app, host="127.0.0.1", port=port,
                                log_level="error", access_log=False)
        self._server = uvicorn.Server(config)
        self._server_thread = threading.Thread(target=self._server.run,
                                               daemon=True)
        self._server_thread.start()
        self.port = port
        self.base_url = f"http://127.0.0.1:{port}"
        deadline = time.time() + 30
        import httpx

        while time.time() < deadline:
            try:
                if httpx.get(self.base_url + "/healthz",
                             timeout=1.0).status_code == 200:
                    return
            except Exception:
                time.sleep(0.05)
        raise RuntimeError("bench HTTP server failed to start")

    @staticmethod
    def _free_port() -> int:
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        return port

    # -- one step --

    def step(self, step_idx: int, record_latency: bool):
        pods = [self.client.create_pod(
                    mixed_pod_spec(i, step_idx, self.rank,
                                   self.args.pod_pad_bytes))
                for i in range(self.args.batch)]
        if self.args.no_http:
            self._schedule_direct(pods, record_latency)
        elif self.args.kube_sim:
            self._schedule_kube_sim(pods, record_latency)
        else:
            self._schedule_http(pods, record_latency)
        self._verify_sample(pods)
        self._release(pods)

    def _schedule_http(self, pods, record_latency: bool):
        """Threaded load generator: each worker holds one keep-alive HTTP
        connection (kube-scheduler style) and drives pods through
        filter -> priorities -> bind, retrying a failed bind like the real
        scheduler requeues a pod. Raw sockets keep client overhead ~tens of
        us so the SERVER is what gets measured."""
        import queue as _queue

        q: "_queue.Queue" = _queue.Queue()
        for pod in pods:
            q.put(pod)
        errors = []
        n_workers = min(self.args.concurrency, len(pods))

        def worker():
            conn = MiniHttpClient("127.0.0.1", self.port,
                                  ssl_context=self._client_ssl)
            try:
                while True:
                    try:
                        pod = q.get_nowait()
                    except _queue.Empty:
                        return
                    try:
                        self._schedule_one(conn, pod, record_latency)
                    except Exception as exc:  # noqa: BLE001
                        errors.append(exc)
            finally:
                conn.close()

        threads = [threading.Thread(target=worker) for _ in range(n_workers)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        if errors:
            raise errors[0]

    def _schedule_one(self, conn, pod, record_latency: bool):
        t0 = time.perf_counter()
        for attempt in range(8):
            status, body = conn.post_json(
                "/scheduler/filter",
                {"pod": pod, "nodenames": self.node_names})
            ok = body.get("nodenames") or []
            if not ok:
                raise RuntimeError(f"no feasible node: {body}")
            status, prio = conn.post_json(
                "/scheduler/priorities", {"pod": pod, "nodenames": ok})
            top = max(e["score"] for e in prio)
            tied = [e["host"] for e in prio if e["score"] == top]
            best = tied[(hash(pod["metadata"]["uid"]) + attempt) % len(tied)]
            status, out = conn.post_json("/scheduler/bind", {
                "podName": pod["metadata"]["name"],
                "podNamespace": "default",
                "podUID": pod["metadata"]["uid"],
                "node": best})
            if status == 200:
                break
            with self._lat_mu:
                self.bind_retries += 1
        else:
            raise RuntimeError(f"bind kept failing: {out}")
        if record_latency:
            with self._lat_mu:
                self.latencies.append(time.perf_counter() - t0)

    def _schedule_kube_sim(self, pods, record_latency: bool):
        """--kube-sim: ONE kube-scheduler's actual concurrency shape —
        scheduleOne runs filter -> priorities -> select SEQUENTIALLY (one
        pod at a time), while binds are dispatched asynchronously (the
        real scheduler's bind goroutine). The default load generator
        models N independent schedulers racing; this mode shows the
        conflict rate a standard single-scheduler deployment sees."""
        import concurrent.futures as cf
        import queue as _queue

        sched_conn = MiniHttpClient("127.0.0.1", self.port,
                                    ssl_context=self._client_ssl)
        bind_conns: "_queue.Queue" = _queue.Queue()
        for _ in range(4):
            bind_conns.put(MiniHttpClient("127.0.0.1", self.port,
                                          ssl_context=self._client_ssl))
        errors = []

        def do_bind(pod, node, t0):
            conn = bind_conns.get()
            try:
                for attempt in range(8):
                    status, out = conn.post_json("/scheduler/bind", {
                        "podName": pod["metadata"]["name"],
                        "podNamespace": "default",
                        "podUID": pod["metadata"]["uid"],
                        "node": node})
                    if status == 200:
                        break
                    with self._lat_mu:
                        self.bind_retries += 1
                    # requeue: refilter like the real scheduler would
                    status, body = conn.post_json(
                        "/scheduler/filter",
                        {"pod": pod, "nodenames": self.node_names})
                    ok = body.get("nodenames") or []
                    if not ok:
                        errors.append(RuntimeError(f"infeasible: {body}"))
                        return
                    status, prio = conn.post_json(
                        "/scheduler/priorities",
                        {"pod": pod, "nodenames": ok})
                    top = max(e["score"] for e in prio)
                    tied = [e["host"] for e in prio if e["score"] == top]
                    node = tied[(hash(pod["metadata"]["uid"]) + attempt + 1)
                                % len(tied)]
                else:
                    errors.append(RuntimeError("bind kept failing"))
                    return
                if record_latency:
                    with self._lat_mu:
                        self.latencies.append(time.perf_counter() - t0)
            finally:
                bind_conns.put(conn)

        with cf.ThreadPoolExecutor(max_workers=4) as binder:
            futures = []
            for pod in pods:
                t0 = time.perf_counter()
                _, body = sched_conn.post_json(
                    "/scheduler/filter",
                    {"pod": pod, "nodenames": self.node_names})
                ok = body.get("nodenames") or []
                if not ok:
                    raise RuntimeError(f"no feasible node: {body}")
                _, prio = sched_conn.post_json(
                    "/scheduler/priorities", {"pod": pod, "nodenames": ok})
                top = max(e["score"] for e in prio)
                tied = [e["host"] for e in prio if e["score"] == top]
                best = tied[hash(pod["metadata"]["uid"]) % len(tied)]
                futures.append(binder.submit(do_bind, pod, best, t0))
            for f in futures:
                f.result()
        sched_conn.close()
        while not bind_conns.empty():
            bind_conns.get().close()
        if errors:
            raise errors[0]

    def _schedule_direct(self, pods, record_latency: bool):
        """--no-http: drive the handlers in-process (core profiling mode)."""
        from elastic_gpu_scheduler_amd.scheduler.service import BindError

        for pod in pods:
            t0 = time.perf_counter()
            sch = self.registry.for_pod(pod)
            for attempt in range(8):
                ok, _ = sch.assume(self.node_names, pod)
                if not ok:
                    raise RuntimeError("no feasible node")
                scores = sch.score(ok, pod)
                top = max(scores)
                tied = [n for n, s in zip(ok, scores) if s == top]
                best = tied[(hash(pod["metadata"]["uid"]) + attempt) % len(tied)]
                try:
                    sch.bind(best, self.client.get_pod(
                        "default", pod["metadata"]["name"]))
                    break
                except BindError:
                    self.bind_retries += 1
            else:
                raise RuntimeError("bind kept failing (direct mode)")
            if record_latency:
                self.latencies.append(time.perf_counter() - t0)

    def _verify_sample(self, pods):
        if self.probe is None:
            return
        from elastic_gpu_scheduler_amd.k8s import objects as obj

        verified = 0
        for pod in pods:
            if verified >= self.args.verify_sample:
                break
            bound = self.client.get_pod("default", pod["metadata"]["name"])
            alloc = obj.parse_allocation(bound)
            if not alloc or not alloc[0]:
                continue
            # stamp on the local physical card (the bench box has 1 visible
            # card per rank; logical index maps onto it)
            tag = hash(bound["metadata"]["uid"]) & ((1 << 64) - 1)
            if not self.probe.stamp(self.device_index, tag, 8):
                raise RuntimeError("placement stamp verification failed")
            verified += 1

    def _release(self, pods):
        for pod in pods:
            self.client.delete_pod("default", pod["metadata"]["name"])
        # wait until the reconcile controller released every card
        sch = self.registry.default
        deadline = time.time() + 30
        while time.time() < deadline:
            if all(d.core_avail == d.core_total
                   for n in self.node_names
                   for d in sch.state.node_devices(n)):
                return
            time.sleep(0.001)
        raise RuntimeError("controller failed to release pods in time")

    def close(self):
        self.controller.stop()
        if self._tls_dir is not None:
            self._tls_dir.cleanup()
        if self._native is not None:
            self._native.stop()
        if self._server is not None:
            self._server.should_exit = True
            self._server_thread.join(timeout=5)


# ---------------------------------------------------------------- main

def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=64,
                   help="pods per step (BASELINE config 4: 64)")
    p.add_argument("--nodes", type=int, default=8,
                   help="fake 8-card MI355X nodes per scheduler shard")
    p.add_argument("--cards", type=int, default=8)
    p.add_argument("--policy", default="binpack",
                   choices=("binpack", "spread", "random"))
    p.add_argument("--concurrency", type=int, default=1,
                   help="in-flight pods in the load generator. 1 measured "
                        "best on the r2 build (MI355X box: 3078 pods/s at "
                        "p50 0.24 ms vs 4-way's 2485 at 1.17 ms — the C++ "
                        "fast paths got cheap enough that extra in-flight "
                        "pods only buy GIL contention on the bind path); "
                        "--kube-sim models the real single-scheduler "
                        "regime with async binds")
    p.add_argument("--filter-threads", type=int, default=0)
    p.add_argument("--server", default="native",
                   choices=("native", "uvicorn"),
                   help="HTTP front end: native C++ (default) or uvicorn")
    p.add_argument("--no-http", action="store_true",
                   help="bypass TCP; drive handlers in-process")
    p.add_argument("--tls", action="store_true",
                   help="serve and drive the full pipeline over HTTPS "
                        "(native front end, self-signed PKI)")
    p.add_argument("--kube-sim", action="store_true",
                   help="model ONE kube-scheduler (sequential scheduleOne, "
                        "async binds) instead of N racing schedulers")
    p.add_argument("--no-verify", action="store_true",
                   help="skip on-GPU placement stamping")
    p.add_argument("--self-profile", default="", metavar="PATH",
                   help="capture /debug/profile (collapsed stacks) during "
                        "the timed steps and write it to PATH")
    p.add_argument("--pod-pad-bytes", type=int, default=0,
                   help="pad each pod object with N annotation bytes to "
                        "model realistic (multi-kB) pod specs on the wire")
    p.add_argument("--verify-sample", type=int, default=2,
                   help="placements stamped on-device per step")
    args = p.parse_args()

    import torch

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size > 1 and args.filter_threads == 0:
        # one scheduler shard per GPU shares the host's CPUs: size each
        # shard's fan-out pool to its fair share instead of ncpu threads
        # PER RANK (8 ranks x 256 threads would thrash a 256-core box)
        args.filter_threads = max(4, (os.cpu_count() or 8) // world_size)
    use_gpu = torch.cuda.is_available()
    distributed = world_size > 1

    if use_gpu:
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
    if distributed:
        import contextlib
        import torch.distributed as dist

        @contextlib.contextmanager
        def _quiet_stdout():
            # gloo prints its rendezvous banner straight to fd 1, which
            # would pollute the single-JSON-line stdout contract
            import sys as _sys

            _sys.stdout.flush()
            saved = os.dup(1)
            devnull = os.open(os.devnull, os.O_WRONLY)
            os.dup2(devnull, 1)
            try:
                yield
            finally:
                _sys.stdout.flush()
                os.dup2(saved, 1)
                os.close(saved)
                os.close(devnull)

        with _quiet_stdout():
            dist.init_process_group(backend="nccl" if use_gpu else "gloo")

    # Opportunistic multi-GPU topology capture (the
    # >=2-card branches of probe.hip had never seen hardware): when this
    # bench lands on a multi-GPU box (the driver's 8-GPU scaling run),
    # rank 0 measures the REAL xGMI hop matrix + p2p bandwidths once,
    # before the timed region, in a SUBPROCESS with a hard timeout so a
    # wedged p2p path can never hang the bench itself.
    xgmi_measured = None
    if use_gpu and rank == 0 and torch.cuda.device_count() > 1:
        import subprocess
        import sys as _sys

        try:
            probe_out = subprocess.run(
                [_sys.executable, "-c",
                 "import json\n"
                 "from elastic_gpu_scheduler_amd.agent.agent import NodeAgent\n"
                 "t = NodeAgent('bench').measured_topology(mib=8, iters=2)\n"
                 "print(json.dumps(t))"],
                capture_output=True, text=True, timeout=150,
                cwd=os.path.dirname(os.path.abspath(__file__)))
            if probe_out.returncode == 0:
                topo = json.loads(probe_out.stdout.strip().splitlines()[-1])
                bw = topo.get("bandwidth_gbps") or []
                xgmi_measured = {
                    "hops": topo.get("hops"),
                    "bandwidth_gbps": [[round(x) for x in row]
                                       for row in bw],
                }
            else:
                print(f"xgmi capture failed: {probe_out.stderr[-400:]}",
                      file=__import__("sys").stderr)
        except Exception as exc:  # never fail the bench for the probe
            print(f"xgmi capture skipped: {exc}",
                  file=__import__("sys").stderr)

    device_index = local_rank % max(
        torch.cuda.device_count(), 1) if use_gpu else 0
    pipe = BenchPipeline(rank, args, device_index, use_gpu)

    def barrier():
        if distributed:
            import torch.distributed as dist

            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    try:
        for w in range(args.warmup):
            pipe.step(-1 - w, record_latency=False)

        profile_thread = None
        if args.self_profile and rank == 0 and not args.no_http:
            def grab_profile():
                import urllib.request

                try:
                    with urllib.request.urlopen(
                            pipe.base_url + "/debug/profile",
                            timeout=30) as r:
                        data = r.read()
                    with open(args.self_profile, "wb") as f:
                        f.write(data)
                except Exception as exc:  # profiling must never fail the bench
                    print(f"self-profile failed: {exc}", file=__import__("sys").stderr)

            profile_thread = threading.Thread(target=grab_profile, daemon=True)

        barrier()
        t0 = time.perf_counter()
        for s in range(args.steps):
            if s == 1 and profile_thread is not None:
                profile_thread.start()  # sample while under steady load
            pipe.step(s, record_latency=True)
        barrier()
        elapsed = time.perf_counter() - t0

        # MAX over ranks
        if distributed:
            import torch.distributed as dist

            dev = "cuda" if use_gpu else "cpu"
            t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t.item())

        if rank == 0:
            pods_total = args.batch * args.steps * world_size
            lat_ms = sorted(x * 1000 for x in pipe.latencies)
            p50 = statistics.median(lat_ms) if lat_ms else None
            p99 = (lat_ms[int(len(lat_ms) * 0.99) - 1]
                   if len(lat_ms) >= 2 else None)
            result = {
                "metric": "pods_scheduled_per_sec",
                "value": round(pods_total / elapsed, 2),
                "unit": "pods/s",
                "n_gpus": world_size,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(elapsed * 1000 / args.steps, 3),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "n/a",
                "data": "synthetic",
                "config": {
                    "model": "extender pipeline: HTTP filter+priorities+bind"
                             " + controller release (fake apiserver)",
                    "workload": f"{args.batch} mixed gpu-core/gpu-memory pods"
                                " per step (BASELINE config 4)",
                    "global_batch": args.batch * world_size,
                    "parallelism": f"shard{world_size}",
                    "policy": args.policy,
                    "nodes_per_shard": args.nodes,
                    "cards_per_node": args.cards,
                    "card": "MI355X 288GB",
                    "http": not args.no_http,
                    "server": "none" if args.no_http else args.server,
                    "native_stats": (pipe._native.stats()
                                     if pipe._native else None),
                    "concurrency": args.concurrency,
                    "kube_sim": args.kube_sim,
                    "tls": args.tls,
                    "pod_pad_bytes": args.pod_pad_bytes,
                    "p50_filter_bind_ms": round(p50, 3) if p50 else None,
                    "p99_filter_bind_ms": round(p99, 3) if p99 else None,
                    "verify_on_device": pipe.probe is not None,
                    "bind_retries": pipe.bind_retries,
                    "xgmi_measured": xgmi_measured,
                },
            }
            print(json.dumps(result), flush=True)
    finally:
        pt = locals().get("profile_thread")
        if pt is not None and pt.is_alive():
            pt.join(timeout=30)
        pipe.close()
        if distributed:
            import torch.distributed as dist

            dist.destroy_process_group()


if __name__ == "__main__":
    main()
