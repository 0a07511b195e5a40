"""Size-unit handling.

Reference contract: units KB/MB/GB/TB, 1024-based
(/root/reference/internal/models/memory.go:3-14, utils/file.go:24-48).
"""
from __future__ import annotations

from ..xerrors import GdaError

MEMORY_UNITS: dict[str, int] = {
    "KB": 1024,
    "MB": 1024**2,
    "GB": 1024**3,
    "TB": 1024**4,
}


class UnsupportedSizeUnit(GdaError):
    pass


def parse_size(s: str) -> tuple[float, str]:
    """Split ``"20GB"`` into ``(20.0, "GB")``. Raises UnsupportedSizeUnit."""
    s = s.strip()
    if len(s) < 3:
        raise UnsupportedSizeUnit(s)
    num, unit = s[:-2], s[-2:].upper()
    if unit not in MEMORY_UNITS:
        raise UnsupportedSizeUnit(s)
    try:
        value = float(num)
    except ValueError:
        raise UnsupportedSizeUnit(s) from None
    return value, unit


def to_bytes(s: str) -> int:
    """``"20GB"`` -> 21474836480. Reference: utils/file.go:24 (ToBytes)."""
    value, unit = parse_size(s)
    return int(value * MEMORY_UNITS[unit])


def format_bytes(n: int) -> str:
    """Render bytes with the largest exact-ish 1024 unit, e.g. 2147483648 -> "2GB".

    Used when re-materializing a memory limit from a stored spec; the
    reference gets this wrong (Memory/1024/1024 rendered "GB",
    services/replicaset.go:408 — a 1024x inflation we do not copy).
    """
    for unit in ("TB", "GB", "MB", "KB"):
        factor = MEMORY_UNITS[unit]
        if n >= factor and n % factor == 0:
            return f"{n // factor}{unit}"
    for unit in ("TB", "GB", "MB", "KB"):
        factor = MEMORY_UNITS[unit]
        if n >= factor:
            return f"{n / factor:g}{unit}"
    return f"{max(n, 0) / 1024:g}KB"
