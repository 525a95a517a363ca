"""Loaders for the in-tree native extensions.

The native core (`_core`) is mandatory everywhere — the scheduler's hot path
is C++ and there is deliberately no Python fallback (a silent eager fallback
would hide a broken build). The GPU probe (`_gpuprobe`) loads lazily and is
mandatory whenever a GPU is actually present; on CPU-only machines callers
gate on `gpuprobe_available()`.

Load-order constraint (measured on MI355X / ROCm 7.2 + PyTorch-ROCm):
PyTorch ships its own libamdhip64/libhsa-runtime64 while `_gpuprobe` links
/opt/rocm's. If OUR runtime loads first and torch's second, our
hipGetDeviceCount sees 0 devices; torch-first always works. So `_gpuprobe`
is imported lazily and torch (when importable) is imported right before it.
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

# Push the bare-number "auto" GiB/bytes threshold into the C++ fast path so
# there is ONE definition (utils/quantity.py) — a drift between the two would
# split memory-quantity semantics per request path.
from elastic_gpu_scheduler_amd.utils.quantity import BARE_AUTO_GIB_THRESHOLD as _thr

core.set_bare_auto_gib_threshold(_thr)

_gpuprobe = None
_gpuprobe_err: Exception | None = None
_gpuprobe_tried = False


def _try_load_gpuprobe():
    global _gpuprobe, _gpuprobe_err, _gpuprobe_tried
    if _gpuprobe_tried:
        return
    _gpuprobe_tried = True
    try:
        # Ensure torch's bundled HIP/HSA runtime is resident first (see
        # module docstring); harmless when torch is absent or CPU-only.
        importlib.import_module("torch")
    except ImportError:
        pass
    try:
        _gpuprobe = _load("_gpuprobe")
    except ImportError as exc:
        _gpuprobe_err = exc


def gpuprobe_available() -> bool:
    _try_load_gpuprobe()
    return _gpuprobe is not None


def gpuprobe():
    """Return the HIP probe module; raise loudly if it was not built.

    On a GPU box the probe must exist — we never silently fall back.
    """
    _try_load_gpuprobe()
    if _gpuprobe is None:
        raise ImportError(
            "elastic_gpu_scheduler_amd._gpuprobe (HIP/gfx950) is missing: "
            f"{_gpuprobe_err}. Build with `python build_native.py`."
        )
    return _gpuprobe
