"""CLI entrypoints: flags, fake cluster seeding, agent dry-run, and a live
serve smoke over the native server (SIGTERM shutdown)."""
from __future__ import annotations

import json
import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

import httpx

from elastic_gpu_scheduler_amd.cmd.agent_main import build_parser as agent_parser
from elastic_gpu_scheduler_amd.cmd.main import build_parser, make_fake_cluster

REPO = Path(__file__).resolve().parent.parent


def test_parser_defaults_match_reference():
    args = build_parser().parse_args([])
    assert args.priority == "binpack"
    assert args.mode == "gpushare"
    assert args.port == 39999  # reference default port (cmd/main.go:69-72)
    assert args.threadness == 1
    assert args.server == "native"


def test_parser_env_compat(monkeypatch):
    monkeypatch.setenv("PORT", "12345")
    monkeypatch.setenv("THREADNESS", "4")
    args = build_parser().parse_args([])
    assert args.port == 12345
    assert args.threadness == 4


def test_make_fake_cluster():
    client = make_fake_cluster(3)
    nodes = client.list_nodes()
    assert len(nodes) == 3
    alloc = nodes[0]["status"]["allocatable"]
    assert alloc["elasticgpu.io/gpu-core"] == "800"
    assert alloc["amd.com/gpu"] == "8"


def test_agent_dry_run():
    out = subprocess.run(
        [sys.executable, "-m", "elastic_gpu_scheduler_amd.cmd.agent_main",
         "--dry-run"],
        capture_output=True, text=True, timeout=120, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    ann = json.loads(out.stdout)
    assert "elasticgpu.io/gpu-inventory" in ann
    assert "elasticgpu.io/xgmi-topology" in ann


def test_serve_fake_cluster_and_shutdown():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [sys.executable, "-m", "elastic_gpu_scheduler_amd.cmd.main",
         "--fake-cluster", "2", "--port", str(port), "--host", "127.0.0.1"],
        cwd=str(REPO), stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        base = f"http://127.0.0.1:{port}"
        deadline = time.time() + 60
        up = False
        while time.time() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=1.0).status_code == 200:
                    up = True
                    break
            except Exception:
                time.sleep(0.1)
        assert up, "server never came up"
        pod = {
            "metadata": {"name": "p", "namespace": "default", "uid": "u1"},
            "spec": {"containers": [{"name": "c", "resources": {"requests": {
                "elasticgpu.io/gpu-core": "30"}}}]},
        }
        r = httpx.post(base + "/scheduler/filter",
                       json={"pod": pod,
                             "nodenames": ["mi355x-node-0", "mi355x-node-1"]},
                       timeout=10.0)
        assert r.status_code == 200
        assert len(r.json()["nodenames"]) == 2
        assert httpx.get(base + "/version", timeout=5.0).status_code == 200
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            raise AssertionError("server did not shut down on SIGTERM")


def test_agent_parser():
    args = agent_parser().parse_args(["--node", "n1", "--interval", "0"])
    assert args.node == "n1"
    assert args.interval == 0


def test_serve_https_uvicorn(tmp_path):
    """--tls-cert/--tls-key serve the extender over HTTPS (enableHTTPS)."""
    cert, key = tmp_path / "tls.crt", tmp_path / "tls.key"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "1",
         "-subj", "/CN=localhost"],
        check=True, capture_output=True, timeout=120)
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [sys.executable, "-m", "elastic_gpu_scheduler_amd.cmd.main",
         "--fake-cluster", "1", "--port", str(port), "--host", "127.0.0.1",
         "--server", "uvicorn", "--tls-cert", str(cert),
         "--tls-key", str(key)],
        cwd=str(REPO), stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        base = f"https://127.0.0.1:{port}"
        deadline = time.time() + 60
        ok = False
        while time.time() < deadline:
            try:
                if httpx.get(base + "/healthz", verify=False,
                             timeout=1.0).status_code == 200:
                    ok = True
                    break
            except Exception:
                time.sleep(0.1)
        assert ok, "HTTPS server never came up"
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
