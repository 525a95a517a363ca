"""Entrypoint: flags -> clients -> registry -> controller -> HTTP server.

Analogue of the reference cmd/main.go:32-101 with the same knobs:
  --priority  binpack|spread|random   (reference -priority, main.go:27;
                                       random is new — upstream README
                                       promises it but ships a stub)
  --mode      gpushare|pgpu|qgpu      (reference -mode, main.go:29)
  --kubeconf  path                    (reference -kubeconf, main.go:28)
  --port      default 39999          (reference env PORT, main.go:69-72)
  --threadness N controller workers   (reference env THREADNESS, main.go:68)
plus MI355X-native additions: --bare-memory-unit, --seed, --fake-cluster
(an in-process fake apiserver seeded with MI355X nodes, for local runs and
benchmarking without a cluster).

Env vars PORT and THREADNESS are honoured for drop-in compatibility.
"""
from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import threading

from elastic_gpu_scheduler_amd.utils import types as t


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="elastic-gpu-scheduler-amd",
                                description=__doc__)
    p.add_argument("--priority", default=t.PRIORITY_BINPACK, choices=t.PRIORITIES)
    p.add_argument("--mode", default=t.MODE_GPUSHARE, choices=t.MODES)
    p.add_argument("--kubeconf", default=os.environ.get("KUBECONFIG", ""))
    p.add_argument("--port", type=int,
                   default=int(os.environ.get("PORT", t.DEFAULT_PORT)))
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--threadness", type=int,
                   default=int(os.environ.get("THREADNESS", 1)),
                   help="controller worker threads")
    p.add_argument("--filter-threads", type=int, default=0,
                   help="native filter fan-out threads (0 = one per host CPU)")
    p.add_argument("--bare-memory-unit", default="auto",
                   choices=("auto", "bytes", "GiB", "MiB"),
                   help="how to read suffix-less gpu-memory quantities")
    p.add_argument("--seed", type=int, default=0, help="random-policy seed")
    p.add_argument("--topology-weight", type=float, default=0.3,
                   help="xGMI-locality share of a multi-card placement's "
                        "score, 0..1 (0 disables topology steering)")
    p.add_argument("--fake-cluster", type=int, default=0, metavar="NODES",
                   help="serve against an in-process fake apiserver with N "
                        "8x-MI355X nodes (local dev / benchmarking)")
    p.add_argument("--server", default="native", choices=("native", "uvicorn"),
                   help="HTTP front end: native C++ (default) or uvicorn ASGI")
    p.add_argument("--leader-elect", action="store_true",
                   help="Lease-based leader election: standby replicas wait "
                        "for the lease instead of double-booking GPUs (the "
                        "reference supports only replicas=1)")
    p.add_argument("--tls-cert", default="", metavar="PEM",
                   help="serve HTTPS (extender enableHTTPS); works on BOTH "
                        "front ends — the native C++ server terminates TLS "
                        "with OpenSSL and keeps the GIL-free fast path")
    p.add_argument("--tls-key", default="", metavar="PEM")
    p.add_argument("--tls-client-ca", default="", metavar="PEM",
                   help="require+verify client certificates (mTLS), native "
                        "front end")
    p.add_argument("--leader-identity",
                   default=os.environ.get("POD_NAME", "") or os.uname().nodename)
    p.add_argument("--log-level", default="info")
    return p


def make_fake_cluster(n_nodes: int):
    from elastic_gpu_scheduler_amd.k8s.client import FakeKubeClient

    client = FakeKubeClient()
    for i in range(n_nodes):
        client.add_node({
            "metadata": {"name": f"mi355x-node-{i}"},
            "status": {"allocatable": {
                t.RESOURCE_GPU_CORE: str(t.GPU_CORE_EACH_CARD *
                                         t.MI355X_CARDS_PER_NODE),
                t.RESOURCE_GPU_MEMORY: str(t.MI355X_MEMORY_BYTES *
                                           t.MI355X_CARDS_PER_NODE),
                t.RESOURCE_AMD_GPU: str(t.MI355X_CARDS_PER_NODE),
            }},
        })
    return client


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    logging.basicConfig(
        level=getattr(logging, args.log_level.upper(), logging.INFO),
        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    log = logging.getLogger("egs.main")

    if args.fake_cluster > 0:
        client = make_fake_cluster(args.fake_cluster)
        log.info("serving against in-process fake apiserver with %d nodes",
                 args.fake_cluster)
    else:
        from elastic_gpu_scheduler_amd.k8s.client import RealKubeClient

        if args.kubeconf:
            os.environ["KUBECONFIG"] = args.kubeconf
        client = RealKubeClient.from_env()

    if args.leader_elect:
        from elastic_gpu_scheduler_amd.k8s.leader import LeaderElector

        elector = LeaderElector(client, "elastic-gpu-scheduler-amd",
                                args.leader_identity)
        acquired = threading.Event()
        lost = threading.Event()
        threading.Thread(target=elector.run,
                         args=(acquired.set, lost.set), daemon=True).start()
        log.info("waiting for leadership (%s)...", args.leader_identity)
        acquired.wait()
        log.info("leadership acquired")

        def watch_loss():
            lost.wait()
            log.error("leadership lost; exiting for restart")
            os._exit(1)

        threading.Thread(target=watch_loss, daemon=True).start()

    from elastic_gpu_scheduler_amd.controller.controller import Controller
    from elastic_gpu_scheduler_amd.scheduler.service import SchedulerRegistry
    from elastic_gpu_scheduler_amd.server.app import make_app

    registry = SchedulerRegistry(client, mode=args.mode, policy=args.priority,
                                 seed=args.seed, threads=args.filter_threads,
                                 bare_unit=args.bare_memory_unit,
                                 topology_weight=args.topology_weight)
    controller = Controller(client, registry, workers=args.threadness)
    controller.start()

    app = make_app(registry)

    # SIGINT/SIGTERM -> graceful stop; second signal -> hard exit
    # (reference pkg/utils/signals/signal.go:16-30).
    stop_event = threading.Event()
    signal_count = {"n": 0}

    def handle(sig, frame):
        signal_count["n"] += 1
        if signal_count["n"] >= 2:
            os._exit(1)
        stop_event.set()

    signal.signal(signal.SIGINT, handle)
    signal.signal(signal.SIGTERM, handle)

    log.info("listening on %s:%d (policy=%s mode=%s server=%s)", args.host,
             args.port, args.priority, args.mode, args.server)
    if args.server == "native":
        from elastic_gpu_scheduler_amd.server.native import NativeFrontend

        # TLS (extender enableHTTPS) is terminated by OpenSSL inside the
        # C++ server — the GIL-free fast path is kept under HTTPS.
        fe = NativeFrontend(app, host=args.host, port=args.port,
                            tls_cert=args.tls_cert, tls_key=args.tls_key,
                            tls_client_ca=args.tls_client_ca)
        fe.start()
        stop_event.wait()
        fe.stop()
    else:
        import uvicorn

        import ssl as _ssl

        config = uvicorn.Config(app, host=args.host, port=args.port,
                                log_level=args.log_level, access_log=False,
                                ssl_certfile=args.tls_cert or None,
                                ssl_keyfile=args.tls_key or None,
                                ssl_ca_certs=args.tls_client_ca or None,
                                ssl_cert_reqs=_ssl.CERT_REQUIRED
                                if args.tls_client_ca else _ssl.CERT_NONE)
        server = uvicorn.Server(config)

        def watch_stop():
            stop_event.wait()
            server.should_exit = True

        threading.Thread(target=watch_stop, daemon=True).start()
        server.run()
    controller.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
