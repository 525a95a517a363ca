"""Loaders for the in-tree native extensions.

The native core (`_core`) is mandatory everywhere — the scheduler's hot path
is C++ and there is deliberately no Python fallback (a silent eager fallback
would hide a broken build). The GPU probe (`_gpuprobe`) is mandatory whenever
a GPU is actually present; on CPU-only machines callers gate on
`gpuprobe_available()`.
"""
from __future__ import annotations

import importlib


def _load(name: str):
    return importlib.import_module(f"elastic_gpu_scheduler_amd.{name}")


try:
    core = _load("_core")
except ImportError as exc:  # pragma: no cover - build failure is fatal
    raise ImportError(
        "elastic_gpu_scheduler_amd._core native extension is missing. "
        "Build it in-tree with `python build_native.py` (repo root)."
    ) from exc

_gpuprobe = None
_gpuprobe_err: Exception | None = None
try:
    _gpuprobe = _load("_gpuprobe")
except ImportError as exc:
    _gpuprobe_err = exc


def gpuprobe_available() -> bool:
    return _gpuprobe is not None


def gpuprobe():
    """Return the HIP probe module; raise loudly if it was not built.

    On a GPU box the probe must exist — we never silently fall back.
    """
    if _gpuprobe is None:
        raise ImportError(
            "elastic_gpu_scheduler_amd._gpuprobe (HIP/gfx950) is missing: "
            f"{_gpuprobe_err}. Build with `python build_native.py`."
        )
    return _gpuprobe
