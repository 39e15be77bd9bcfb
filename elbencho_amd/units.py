"""Unit parsing and human-readable formatting.

Behavior parity with the reference's UnitTk
(/root/reference/source/toolkits/UnitTk.cpp): base-2 suffix parsing
("4M" -> 4 MiB), elapsed-time humanization ("2m3.456s"), base-2/-10 number
scaling. Independent implementation.
"""

from __future__ import annotations

_BASE2 = {"k": 1 << 10, "m": 1 << 20, "g": 1 << 30, "t": 1 << 40, "p": 1 << 50, "e": 1 << 60}
_BASE10 = {"k": 10**3, "m": 10**6, "g": 10**9, "t": 10**12, "p": 10**15, "e": 10**18}


def parse_size(value: str | int | None, base10: bool = False) -> int:
    """Parse a byte size with optional unit suffix: "4K", "1m", "2GB", "512"."""
    if value is None:
        return 0
    if isinstance(value, (int, float)):
        return int(value)
    s = str(value).strip()
    if not s:
        return 0
    s_low = s.lower()
    # strip trailing "b" / "ib" (e.g. "4kb", "4kib")
    mult = 1
    table = _BASE10 if base10 else _BASE2
    if s_low.endswith("ib") and len(s_low) > 2 and s_low[-3] in _BASE2:
        mult = _BASE2[s_low[-3]]
        s_low = s_low[:-3]
    elif s_low.endswith("b") and len(s_low) > 1 and s_low[-2] in table:
        mult = table[s_low[-2]]
        s_low = s_low[:-2]
    elif s_low and s_low[-1] in table:
        mult = table[s_low[-1]]
        s_low = s_low[:-1]
    return int(float(s_low) * mult) if s_low else 0


def elapsed_ms_to_human(elapsed_ms: int) -> str:
    """Format milliseconds like the reference: 1ms / 1.001s / 2m3.456s / 3h25m45s."""
    elapsed_sec = elapsed_ms // 1000
    hours = elapsed_sec // 3600
    mins = (elapsed_sec % 3600) // 60
    secs = elapsed_sec % 60
    ms = elapsed_ms % 1000
    if hours:
        return f"{hours}h{mins}m{secs}s"
    if mins:
        return f"{mins}m{secs}.{ms:03d}s"
    if secs:
        return f"{secs}.{ms:03d}s"
    return f"{ms}ms"


_B2_UNITS = ["", "Ki", "Mi", "Gi", "Ti", "Pi", "Ei"]
_B10_UNITS = ["", "K", "M", "G", "T", "P", "E"]


def num_to_human(number: float, base10: bool = False, max_len: int = 10) -> str:
    """Scale a number with a unit suffix when it exceeds max_len digits."""
    s = str(int(number))
    if len(s) <= max_len:
        return s
    units = _B10_UNITS if base10 else _B2_UNITS
    base = 1000 if base10 else 1024
    val = float(number)
    for u in units:
        if val < base or u == units[-1]:
            if val == int(val):
                return f"{int(val)}{u}"
            return f"{val:.1f}{u}"
        val /= base
    return s


def bytes_to_mib(n: int) -> int:
    return n // (1024 * 1024)
