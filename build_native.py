#!/usr/bin/env python3
"""Build the native extensions in-tree.

Two extensions:
  * elastic_gpu_scheduler_amd/_core.<abi>.so     — C++ scheduler core (g++)
  * elastic_gpu_scheduler_amd/_gpuprobe.<abi>.so — HIP gfx950 device probe (hipcc)

Both are built in-tree so the .so travels to GPU boxes with the repo snapshot
(they are git-ignored; history stays source-only). hipcc cross-compiles gfx950
without a GPU, so the probe builds on CPU-only machines too.
"""
from __future__ import annotations

import os
import shutil
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent
PKG = REPO / "elastic_gpu_scheduler_amd"
CORE_SRC = PKG / "csrc" / "core"
HTTPD_SRC = PKG / "csrc" / "httpd"
PROBE_SRC = PKG / "csrc" / "gpuprobe"
GFX_ARCH = os.environ.get("EGS_GFX_ARCH", "gfx950")


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def _py_includes() -> list[str]:
    import pybind11

    return [
        "-I" + sysconfig.get_paths()["include"],
        "-I" + pybind11.get_include(),
    ]


def _needs_rebuild(target: Path, sources: list[Path]) -> bool:
    if not target.exists():
        return True
    t = target.stat().st_mtime
    return any(s.stat().st_mtime > t for s in sources if s.exists())


def _run(cmd: list[str]) -> None:
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def build_core(force: bool = False) -> Path:
    out = PKG / ("_core" + _ext_suffix())
    sources = (sorted(CORE_SRC.glob("*.cc")) + sorted(CORE_SRC.glob("*.h")) +
               sorted(HTTPD_SRC.glob("*.cc")) + sorted(HTTPD_SRC.glob("*.h")))
    if force or _needs_rebuild(out, sources):
        cmd = (
            ["g++", "-O3", "-std=c++17", "-shared", "-fPIC", "-fvisibility=hidden",
             "-Wall", "-pthread"]
            + _py_includes()
            + ["-I" + str(CORE_SRC)]
            + [str(s) for s in sorted(CORE_SRC.glob("*.cc"))]
            + ["-o", str(out), "-lssl", "-lcrypto"]  # native-server TLS
        )
        _run(cmd)
    return out


def build_gpuprobe(force: bool = False) -> Path | None:
    hipcc = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
    if not Path(hipcc).exists():
        print("hipcc not found; skipping _gpuprobe build", file=sys.stderr)
        return None
    out = PKG / ("_gpuprobe" + _ext_suffix())
    sources = sorted(PROBE_SRC.glob("*.hip"))
    if force or _needs_rebuild(out, sources):
        cmd = (
            [hipcc, f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17", "-shared",
             "-fPIC", "-fvisibility=hidden"]
            + _py_includes()
            + [str(s) for s in sources]
            + ["-o", str(out)]
        )
        _run(cmd)
    return out


def build_all(force: bool = False) -> None:
    build_core(force)
    build_gpuprobe(force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
