"""Race detection: build the native-core stress harness with
ThreadSanitizer and run it (SURVEY.md §5 — the reference has no race
detection; this is the rebuild's)."""
from __future__ import annotations

import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_core_clean_under_tsan(tmp_path):
    binary = tmp_path / "stress"
    build = subprocess.run(
        ["g++", "-O1", "-g", "-std=c++17", "-fsanitize=thread", "-pthread",
         "-I", str(REPO / "elastic_gpu_scheduler_amd/csrc/core"),
         str(REPO / "elastic_gpu_scheduler_amd/csrc/stress/stress_main.cc"),
         "-o", str(binary)],
        capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr[-3000:]
    run = subprocess.run([str(binary)], capture_output=True, text=True,
                         timeout=300,
                         env={"TSAN_OPTIONS": "halt_on_error=1"})
    if "ThreadSanitizer" in run.stderr and "WARNING" in run.stderr:
        raise AssertionError(f"TSAN reported races:\n{run.stderr[-3000:]}")
    if "unexpected memory mapping" in run.stderr or (
            run.returncode < 0 and not run.stdout and not run.stderr):
        # TSAN cannot run under this kernel's ASLR/mmap layout (needs
        # vm.mmap_rnd_bits <= 30); it either prints the mapping error or
        # segfaults before producing any output. Environment, not product.
        import pytest

        pytest.skip("ThreadSanitizer unsupported by this kernel's mmap layout")
    assert run.returncode == 0, (run.stdout[-1000:], run.stderr[-3000:])
    assert "stress ok" in run.stdout


def test_core_clean_under_asan(tmp_path):
    """Same stress harness under AddressSanitizer + UBSan (heap errors,
    overflow, UB in the allocator core)."""
    binary = tmp_path / "stress_asan"
    build = subprocess.run(
        ["g++", "-O1", "-g", "-std=c++17",
         "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
         "-pthread",
         "-I", str(REPO / "elastic_gpu_scheduler_amd/csrc/core"),
         str(REPO / "elastic_gpu_scheduler_amd/csrc/stress/stress_main.cc"),
         "-o", str(binary)],
        capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr[-3000:]
    run = subprocess.run([str(binary)], capture_output=True, text=True,
                         timeout=300)
    assert run.returncode == 0, (run.stdout[-1000:], run.stderr[-3000:])
    assert "stress ok" in run.stdout
