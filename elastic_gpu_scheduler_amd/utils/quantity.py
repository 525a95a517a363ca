"""Kubernetes resource Quantity parsing.

The reference keeps gpu-memory as a unitless integer and divides node
allocatable evenly over cards (pkg/scheduler/node.go:25-40, with a literal
"TODO: GB only" comment); its own test uses 48 (meaning GiB) while the README
example uses 256 (meaning... something else). We parse real Kubernetes
quantities ("64Gi", "256M", "2304Gi") into BYTES, and make the bare-number
heuristic explicit and overridable:

  * suffixed quantities are exact (binary Ki/Mi/Gi/Ti/Pi/Ei, decimal
    k/M/G/T/P/E, and milli "m" which rounds down);
  * bare numbers follow `bare_unit` — "auto" (default: values < 8192 are
    GiB, larger values are bytes — matches how the reference ecosystem
    writes small per-card GiB counts), or "bytes" / "GiB" / "MiB".
"""
from __future__ import annotations

import re

_BINARY = {"Ki": 1024, "Mi": 1024**2, "Gi": 1024**3, "Ti": 1024**4,
           "Pi": 1024**5, "Ei": 1024**6}
_DECIMAL = {"k": 10**3, "M": 10**6, "G": 10**9, "T": 10**12, "P": 10**15,
            "E": 10**18}

_QTY_RE = re.compile(r"^\s*([+-]?[0-9.]+(?:[eE][+-]?[0-9]+)?)\s*([A-Za-z]*)\s*$")
_INT_RE = re.compile(r"^[+-]?[0-9]+$")

BARE_AUTO_GIB_THRESHOLD = 8192


def parse_quantity(value: "str | int | float", bare_unit: str = "auto") -> int:
    """Parse a k8s quantity into an integer (suffix-scaled, rounded down).

    Integer numeric parts are scaled with exact integer arithmetic (large
    byte counts like 4611686019G must not lose precision to float)."""
    if isinstance(value, int):
        num_s, suffix = str(value), ""
    elif isinstance(value, float):
        num_s, suffix = repr(value), ""
    else:
        m = _QTY_RE.match(str(value))
        if not m:
            raise ValueError(f"invalid quantity: {value!r}")
        num_s, suffix = m.group(1), m.group(2)
    is_int = _INT_RE.match(num_s) is not None
    num = int(num_s) if is_int else float(num_s)
    if suffix in _BINARY:
        return int(num * _BINARY[suffix])
    if suffix in _DECIMAL:
        return int(num * _DECIMAL[suffix])
    if suffix == "m":
        return int(num // 1000) if is_int else int(num / 1000)
    if suffix == "":
        return int(num)
    raise ValueError(f"unknown quantity suffix {suffix!r} in {value!r}")


def parse_memory_bytes(value: "str | int | float", bare_unit: str = "auto") -> int:
    """Parse a memory quantity into bytes, applying the bare-number policy."""
    if isinstance(value, str):
        m = _QTY_RE.match(value)
        suffixed = bool(m and m.group(2))
    else:
        suffixed = False
    raw = parse_quantity(value)
    if suffixed:
        return raw
    if bare_unit == "bytes":
        return raw
    if bare_unit == "GiB":
        return raw * 1024**3
    if bare_unit == "MiB":
        return raw * 1024**2
    # auto
    if 0 < raw < BARE_AUTO_GIB_THRESHOLD:
        return raw * 1024**3
    return raw
