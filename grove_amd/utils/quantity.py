"""Kubernetes resource-quantity parsing (subset: suffixes used in pod specs)."""
from __future__ import annotations

_BIN = {"Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40, "Pi": 2**50}
_DEC = {"n": 1e-9, "u": 1e-6, "m": 1e-3, "k": 1e3, "M": 1e6, "G": 1e9, "T": 1e12}


def parse_quantity(q) -> float:
    """Return the plain numeric value (cores for cpu, bytes for memory, count for gpus)."""
    if q is None:
        return 0.0
    if isinstance(q, (int, float)):
        return float(q)
    s = str(q).strip()
    for suf, mult in _BIN.items():
        if s.endswith(suf):
            return float(s[: -len(suf)]) * mult
    if s and s[-1] in _DEC:
        return float(s[:-1]) * _DEC[s[-1]]
    return float(s)


def cpu_millis(q) -> int:
    return int(round(parse_quantity(q) * 1000))
