"""Human-readable sizes (reference: pkg/client/units/size.go:41-48 — decimal)."""
from __future__ import annotations

_UNITS = ["B", "kB", "MB", "GB", "TB", "PB", "EB"]


def human_size(n: float) -> str:
    """Decimal (1000-based) size like the reference's HumanSize."""
    size = float(n)
    i = 0
    while size >= 1000.0 and i < len(_UNITS) - 1:
        size /= 1000.0
        i += 1
    if i == 0:
        return f"{int(size)}{_UNITS[i]}"
    return f"{size:.4g}{_UNITS[i]}"


def human_rate(bytes_per_s: float) -> str:
    return human_size(bytes_per_s) + "/s"
