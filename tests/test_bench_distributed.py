"""Distributed bench path on CPU: 2 ranks over gloo, exactly the code the
driver runs with torch.distributed.run on GPU nodes (RCCL there)."""
from __future__ import annotations

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_two_rank_gloo(tmp_path):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    # force the CPU/gloo path even on a GPU box: two ranks sharing one GPU
    # is not a supported RCCL topology (the driver gives each rank its own)
    env["HIP_VISIBLE_DEVICES"] = ""
    env["CUDA_VISIBLE_DEVICES"] = ""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29571", str(REPO / "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--batch", "16",
         "--no-verify"],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    result = json.loads(lines[0])
    assert result["metric"] == "pods_scheduled_per_sec"
    assert result["n_gpus"] == 2
    assert result["value"] > 0
    assert result["scaling"] == "weak"
    assert result["config"]["global_batch"] == 32


def test_bench_single_rank_json_contract():
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "2",
         "--warmup", "1", "--batch", "16", "--no-verify"],
        capture_output=True, text=True, timeout=600, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-3000:]
    result = json.loads([l for l in out.stdout.splitlines()
                         if l.startswith("{")][0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in result, key
    assert result["n_gpus"] == 1
    assert result["data"] == "synthetic"
