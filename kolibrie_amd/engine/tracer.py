"""Per-operator device timing (SURVEY §5 tracing: per-kernel timing via
hipEvents).

Opt-in tracer: `trace.enable()` (or KOLIBRIE_TRACE=1) wraps every physical
operator the executor dispatches with a hipEvent pair on the current
stream (torch.cuda.Event IS hipEvent on ROCm), so per-op DEVICE time is
measured without a global synchronize per op — events are resolved
lazily at snapshot().  On CPU it falls back to wall-clock.  Like the
reference's exec-stats counters (exec_stats.rs:17-64) this is documented
one-query-at-a-time; rocprofv3 remains the tool for kernel-level stats
(profiles/).
"""
from __future__ import annotations

import os
import time
from collections import defaultdict
from typing import Dict, List, Optional, Tuple

import torch

_enabled = bool(os.environ.get("KOLIBRIE_TRACE"))
# (op label, start, end) — entries are cuda.Event pairs or float seconds
_records: List[Tuple[str, object, object]] = []


def enable():
    global _enabled
    _enabled = True


def disable():
    global _enabled
    _enabled = False


def is_enabled() -> bool:
    return _enabled


def reset():
    _records.clear()


class _Span:
    __slots__ = ("label", "dev", "t0", "e0")

    def __init__(self, label: str, device):
        self.label = label
        self.dev = device
        if device.type == "cuda":
            self.e0 = torch.cuda.Event(enable_timing=True)
            self.e0.record()
            self.t0 = None
        else:
            self.e0 = None
            self.t0 = time.perf_counter()

    def close(self):
        if self.e0 is not None:
            e1 = torch.cuda.Event(enable_timing=True)
            e1.record()
            _records.append((self.label, self.e0, e1))
        else:
            _records.append(
                (self.label, self.t0, time.perf_counter()))


def span(label: str, device) -> Optional[_Span]:
    if not _enabled:
        return None
    return _Span(label, device)


def snapshot() -> Dict[str, Tuple[int, float]]:
    """{op label: (calls, total_ms)} — synchronizes once to resolve the
    pending event pairs."""
    if any(isinstance(s, torch.cuda.Event) for _, s, _ in _records):
        torch.cuda.synchronize()
    agg: Dict[str, List[float]] = defaultdict(lambda: [0, 0.0])
    for label, a, b in _records:
        if isinstance(a, torch.cuda.Event):
            ms = a.elapsed_time(b)
        else:
            ms = (b - a) * 1000.0
        slot = agg[label]
        slot[0] += 1
        slot[1] += ms
    return {k: (int(v[0]), v[1]) for k, v in agg.items()}


def report() -> str:
    rows = sorted(snapshot().items(), key=lambda kv: -kv[1][1])
    lines = [f"{'operator':<28} {'calls':>6} {'total ms':>10}"]
    for label, (calls, ms) in rows:
        lines.append(f"{label:<28} {calls:>6} {ms:>10.3f}")
    return "\n".join(lines)
