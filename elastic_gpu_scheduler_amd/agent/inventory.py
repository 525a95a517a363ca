"""MI355X device inventory discovery.

The reference scheduler ecosystem delegates device discovery to a separate
node agent (elastic-gpu-agent) built on NVML; this is the MI355X-native
replacement. Discovery order:

  1. the in-tree HIP probe (`_gpuprobe`) — authoritative when a GPU is
     visible: exact HBM3E bytes, CU count, gfx arch, measured health;
  2. the `amdsmi` python bindings (ROCm's supported NVML analogue);
  3. `rocm-smi --showmeminfo vram --json` / `amd-smi` subprocess parsing;
  4. torch.cuda device properties (PyTorch-ROCm);
  5. none (CPU-only machine) -> empty inventory.

Each card is reported as {"index", "name", "gcn_arch", "memory_bytes",
"core": 100}; the scheduler's node annotation codec
(k8s.objects.node_devices) consumes exactly this shape.
"""
from __future__ import annotations

import json
import logging
import shutil
import subprocess
from typing import Any, Dict, List

from elastic_gpu_scheduler_amd.utils import types as t

log = logging.getLogger("egs.agent")


def _via_gpuprobe() -> List[Dict[str, Any]]:
    from elastic_gpu_scheduler_amd._native import gpuprobe_available, gpuprobe

    if not gpuprobe_available():
        return []
    probe = gpuprobe()
    cards = []
    for info in probe.inventory():
        cards.append({
            "index": info["index"],
            "name": info["name"],
            "gcn_arch": info["gcn_arch"],
            "memory_bytes": int(info["total_mem_bytes"]),
            "compute_units": info["multi_processor_count"],
            "core": t.GPU_CORE_EACH_CARD,
            "source": "gpuprobe",
        })
    return cards


def _via_amdsmi() -> List[Dict[str, Any]]:
    try:
        import amdsmi  # type: ignore
    except ImportError:
        return []
    try:
        amdsmi.amdsmi_init()
        cards = []
        for i, handle in enumerate(amdsmi.amdsmi_get_processor_handles()):
            asic = amdsmi.amdsmi_get_gpu_asic_info(handle)
            mem = amdsmi.amdsmi_get_gpu_memory_total(
                handle, amdsmi.AmdSmiMemoryType.VRAM)
            cards.append({
                "index": i,
                "name": asic.get("market_name", "AMD GPU"),
                "gcn_arch": asic.get("target_graphics_version", ""),
                "memory_bytes": int(mem),
                "core": t.GPU_CORE_EACH_CARD,
                "source": "amdsmi",
            })
        amdsmi.amdsmi_shut_down()
        return cards
    except Exception:
        log.debug("amdsmi inventory failed", exc_info=True)
        return []


def parse_rocm_smi_vram(payload: str) -> List[Dict[str, Any]]:
    """Parse `rocm-smi --showmeminfo vram --json` output into cards.

    Shape: {"card0": {"VRAM Total Memory (B)": "309237645312", ...}, ...}
    (exercised against canned output in tests — no GPU needed).
    """
    try:
        data = json.loads(payload)
    except json.JSONDecodeError:
        return []
    cards = []
    for key in sorted(data.keys(), key=lambda k: (len(k), k)):
        if not key.startswith("card"):
            continue
        entry = data[key]
        total = None
        for field in ("VRAM Total Memory (B)", "vram_total", "VRAM Total Memory"):
            if field in entry:
                try:
                    total = int(str(entry[field]).strip())
                except ValueError:
                    total = None
                break
        if total is None:
            continue
        cards.append({
            "index": int(key[len("card"):]),
            "name": "AMD GPU",
            "gcn_arch": "",
            "memory_bytes": total,
            "core": t.GPU_CORE_EACH_CARD,
            "source": "rocm-smi",
        })
    return cards


def _via_rocm_smi() -> List[Dict[str, Any]]:
    exe = shutil.which("rocm-smi")
    if not exe:
        return []
    try:
        out = subprocess.run([exe, "--showmeminfo", "vram", "--json"],
                             capture_output=True, text=True, timeout=30)
        return parse_rocm_smi_vram(out.stdout)
    except (subprocess.SubprocessError, OSError):
        return []


def _via_torch() -> List[Dict[str, Any]]:
    try:
        import torch
    except ImportError:
        return []
    if not torch.cuda.is_available():
        return []
    cards = []
    for i in range(torch.cuda.device_count()):
        prop = torch.cuda.get_device_properties(i)
        cards.append({
            "index": i,
            "name": prop.name,
            "gcn_arch": getattr(prop, "gcnArchName", ""),
            "memory_bytes": int(prop.total_memory),
            "core": t.GPU_CORE_EACH_CARD,
            "source": "torch",
        })
    return cards


def parse_amd_smi_static(payload: str) -> List[Dict[str, Any]]:
    """Parse `amd-smi static --json` into cards.

    amd-smi (the supported successor of rocm-smi) emits a list of per-GPU
    objects: [{"gpu": 0, "vram": {"size": {"value": 294912, "unit": "MB"}},
    "asic": {"market_name": ...}}, ...]. Field layouts drifted across ROCm
    releases; this accepts the common shapes.
    """
    try:
        data = json.loads(payload)
    except json.JSONDecodeError:
        return []
    if isinstance(data, dict):
        data = data.get("gpu_data", data.get("gpus", []))
    if not isinstance(data, list):
        return []
    cards = []
    for entry in data:
        if not isinstance(entry, dict):
            continue
        idx = entry.get("gpu", entry.get("gpu_id", len(cards)))
        vram = entry.get("vram", {})
        size = vram.get("size", vram.get("vram_size", {}))
        mem_bytes = None
        if isinstance(size, dict):
            val = size.get("value")
            unit = str(size.get("unit", "MB")).upper()
            if val is not None:
                mult = {"B": 1, "KB": 1024, "MB": 1024**2,
                        "GB": 1024**3}.get(unit, 1024**2)
                mem_bytes = int(float(val) * mult)
        elif isinstance(size, (int, float)):
            mem_bytes = int(size) * 1024**2
        if mem_bytes is None:
            continue
        asic = entry.get("asic", {}) if isinstance(entry.get("asic"), dict) else {}
        cards.append({
            "index": int(idx),
            "name": asic.get("market_name", "AMD GPU"),
            "gcn_arch": asic.get("target_graphics_version", ""),
            "memory_bytes": mem_bytes,
            "core": t.GPU_CORE_EACH_CARD,
            "source": "amd-smi",
        })
    return cards


def _via_amd_smi() -> List[Dict[str, Any]]:
    exe = shutil.which("amd-smi")
    if not exe:
        return []
    try:
        out = subprocess.run([exe, "static", "--json"], capture_output=True,
                             text=True, timeout=30)
        return parse_amd_smi_static(out.stdout)
    except (subprocess.SubprocessError, OSError):
        return []


def parse_compute_partition(payload: str) -> Dict[int, str]:
    """Parse `rocm-smi --showcomputepartition --json`:
    {"card0": {"Compute Partition": "SPX"}, ...} -> {0: "SPX"}.

    MI355X partition modes change what a "card" is: SPX exposes one 288 GB
    device per OAM; CPX exposes 8 partitions of ~36 GB each. The scheduler
    treats whatever is enumerated as the card vector, and the partition tag
    published with the inventory lets operators see which mode produced it.
    """
    try:
        data = json.loads(payload)
    except json.JSONDecodeError:
        return {}
    out = {}
    for key, entry in data.items():
        if not key.startswith("card") or not isinstance(entry, dict):
            continue
        for field in ("Compute Partition", "compute_partition"):
            if field in entry:
                try:
                    out[int(key[len("card"):])] = str(entry[field]).strip()
                except ValueError:
                    pass
                break
    return out


def _query_partitions() -> Dict[int, str]:
    exe = shutil.which("rocm-smi")
    if not exe:
        return {}
    try:
        out = subprocess.run([exe, "--showcomputepartition", "--json"],
                             capture_output=True, text=True, timeout=30)
        return parse_compute_partition(out.stdout)
    except (subprocess.SubprocessError, OSError):
        return {}


def discover(prefer: str = "auto") -> List[Dict[str, Any]]:
    """Discover the node's GPU cards. `prefer` forces one source in tests."""
    sources = {
        "gpuprobe": _via_gpuprobe,
        "amdsmi": _via_amdsmi,
        "amd-smi": _via_amd_smi,
        "rocm-smi": _via_rocm_smi,
        "torch": _via_torch,
    }
    if prefer != "auto":
        return sources[prefer]()
    for name in ("gpuprobe", "amdsmi", "amd-smi", "rocm-smi", "torch"):
        try:
            cards = sources[name]()
        except Exception:
            log.debug("inventory source %s failed", name, exc_info=True)
            cards = []
        if cards:
            partitions = _query_partitions()
            for c in cards:
                if c["index"] in partitions:
                    c["partition"] = partitions[c["index"]]
            return cards
    return []
