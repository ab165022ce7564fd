"""Stable pod index allocation with hole filling.

Parity role: reference internal/index/tracker.go (GetAvailableIndices) — pod hostnames are
'<pclq>-<index>' and must stay dense/stable: deleted pods free their index for reuse.
"""
from __future__ import annotations

from typing import Iterable, List


def available_indices(in_use: Iterable[int], count: int) -> List[int]:
    """Return `count` smallest non-negative indices not present in `in_use`."""
    used = set(in_use)
    out: List[int] = []
    i = 0
    while len(out) < count:
        if i not in used:
            out.append(i)
        i += 1
    return out
