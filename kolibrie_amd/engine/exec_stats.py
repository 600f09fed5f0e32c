"""Execution work counters (ref: execution/exec_stats.rs:17-64 —
SCAN_PROBES / QUADS_EXAMINED / ROWS_EMITTED, reset/snapshot; documented as
one-query-at-a-time).  Device-side kernel counters aggregate into these."""
from __future__ import annotations

from typing import Dict

_COUNTERS: Dict[str, int] = {
    "SCAN_PROBES": 0,
    "QUADS_EXAMINED": 0,
    "ROWS_EMITTED": 0,
}

enabled = True


def bump(name: str, amount: int = 1):
    if enabled:
        _COUNTERS[name] = _COUNTERS.get(name, 0) + int(amount)


def reset():
    for k in _COUNTERS:
        _COUNTERS[k] = 0


def snapshot() -> Dict[str, int]:
    return dict(_COUNTERS)
