"""Node-agent entrypoint: discover MI355X inventory + xGMI topology and
publish them onto this Node object, then keep them fresh.

The reference ecosystem runs elastic-gpu-agent (NVML-based, separate repo)
for this role; here it is part of the same framework. Run as a DaemonSet on
GPU nodes (deploy/elastic-gpu-agent-amd.yaml):

    python -m elastic_gpu_scheduler_amd.cmd.agent_main --node $NODE_NAME

Publishes:
  elasticgpu.io/gpu-inventory   per-card {core, memory_bytes, name, arch}
  elasticgpu.io/xgmi-topology   hop matrix for locality-aware placement
and optionally runs the HBM health probe each cycle, publishing sick cards
as zero-capacity placeholders (list position always equals the physical card
index) so the scheduler stops placing pods on them.
"""
from __future__ import annotations

import argparse
import json
import logging
import os
import sys
import time


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="elastic-gpu-agent-amd", description=__doc__)
    p.add_argument("--node", default=os.environ.get("NODE_NAME", ""),
                   help="node object to annotate (default: $NODE_NAME)")
    p.add_argument("--interval", type=float, default=300.0,
                   help="republish interval seconds (0 = publish once and exit)")
    p.add_argument("--health-check", action="store_true",
                   help="run the HBM bandwidth probe each cycle and exclude "
                        "unhealthy cards from the inventory")
    p.add_argument("--source", default="auto",
                   choices=("auto", "gpuprobe", "amdsmi", "amd-smi",
                            "rocm-smi", "torch"))
    p.add_argument("--dry-run", action="store_true",
                   help="print the annotations instead of patching the node")
    p.add_argument("--log-level", default="info")
    return p


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    logging.basicConfig(
        level=getattr(logging, args.log_level.upper(), logging.INFO),
        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    log = logging.getLogger("egs.agent.main")

    from elastic_gpu_scheduler_amd.agent.agent import NodeAgent

    if args.dry_run:
        agent = NodeAgent(args.node or "dry-run-node", client=None,
                          prefer_source=args.source)
        print(json.dumps(agent.annotations(), indent=2))
        return 0

    if not args.node:
        print("--node (or $NODE_NAME) is required", file=sys.stderr)
        return 2

    from elastic_gpu_scheduler_amd.k8s.client import RealKubeClient

    client = RealKubeClient.from_env()
    agent = NodeAgent(args.node, client, prefer_source=args.source)

    while True:
        try:
            if args.health_check:
                out = agent.publish_with_health()
                if out["sick"]:
                    log.warning("unhealthy cards (published zero-capacity): %s",
                                out["sick"])
            else:
                agent.publish()
            log.info("published inventory for %s", args.node)
        except Exception:
            log.exception("publish failed; retrying next cycle")
        if args.interval <= 0:
            return 0
        time.sleep(args.interval)


if __name__ == "__main__":
    sys.exit(main())
