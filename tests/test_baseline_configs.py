"""BASELINE.json configuration scenarios as tests (fast variants of
benchmarks/sweep.py: placements must be correct under each named config)."""
from __future__ import annotations

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "benchmarks"))

import sweep  # noqa: E402


def test_config1_single_gpu_node():
    r = sweep.config1.__wrapped__() if hasattr(sweep.config1, "__wrapped__") \
        else sweep.config1()
    assert r["pods_per_sec"] > 0
    assert r["p50_ms"] < 100


def test_config2_memory_sharing_one_card():
    assert sweep.config2()["ok"]


def test_config3_whole_card_spread():
    assert sweep.config3()["ok"]


def test_config5_topology_adjacent():
    assert sweep.config5()["ok"]
