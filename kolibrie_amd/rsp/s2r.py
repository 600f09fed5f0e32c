"""S2R: stream-to-relation operators — CSPARQL time-based sliding windows.

Ref parity: kolibrie/src/rsp/s2r.rs (497 LoC) — CSPARQLWindow with
scope/add_to_window/eviction, ReportStrategy {NonEmptyContent,
OnContentChange, OnWindowClose, Periodic}, Tick {Time,Tuple,Batch}Driven,
ContentContainer multiset with per-item last-ts + probabilistic
occurrences, consumer channel/callback registration, flush.

The engine is logical-time driven (timestamps supplied by the caller), so
stream tests are deterministic.  Window contents are item->ts
maps on the host control plane; each firing converts to device columns for
the R2R store (the K7 device ring buffer handles the bulk-ingest path).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from queue import Queue
from typing import Callable, Dict, Hashable, List, Optional, Tuple


class ReportStrategy:
    NON_EMPTY_CONTENT = "NonEmptyContent"
    ON_CONTENT_CHANGE = "OnContentChange"
    ON_WINDOW_CLOSE = "OnWindowClose"
    PERIODIC = "Periodic"


class Tick:
    TIME_DRIVEN = "TimeDriven"
    TUPLE_DRIVEN = "TupleDriven"
    BATCH_DRIVEN = "BatchDriven"


@dataclass(frozen=True)
class Window:
    open: int
    close: int


@dataclass(frozen=True)
class EventKey:
    stream_iri: str
    event_time: int
    payload_hash: int = 0


@dataclass(frozen=True)
class ProbabilisticOccurrence:
    item: Hashable
    event: EventKey
    seed_id: str


class ContentContainer:
    """Multiset window content (ref s2r.rs:100-173)."""

    def __init__(self, origin: str = ""):
        self.elements: Dict[Hashable, int] = {}
        self.deterministic_items: set = set()
        self.probabilistic_occurrences: List[ProbabilisticOccurrence] = []
        self.last_timestamp_changed = 0
        self.origin = origin

    def __len__(self):
        return len(self.elements)

    def __eq__(self, other):
        return (isinstance(other, ContentContainer)
                and self.elements == other.elements)

    def add(self, item: Hashable, ts: int):
        self.deterministic_items.add(item)
        self._add_element(item, ts)

    def _add_element(self, item: Hashable, ts: int):
        prev = self.elements.get(item)
        self.elements[item] = ts if prev is None else max(prev, ts)
        self.last_timestamp_changed = ts

    def add_probabilistic(self, occ: ProbabilisticOccurrence):
        self._add_element(occ.item, occ.event.event_time)
        if not any(x.seed_id == occ.seed_id
                   for x in self.probabilistic_occurrences):
            self.probabilistic_occurrences.append(occ)

    def is_deterministic(self, item: Hashable) -> bool:
        return item in self.deterministic_items

    def items(self):
        return self.elements.keys()

    def iter_with_timestamps(self):
        return self.elements.items()

    def clone(self) -> "ContentContainer":
        c = ContentContainer(self.origin)
        c.elements = dict(self.elements)
        c.deterministic_items = set(self.deterministic_items)
        c.probabilistic_occurrences = list(self.probabilistic_occurrences)
        c.last_timestamp_changed = self.last_timestamp_changed
        return c


class Report:
    """Conjunction of report strategies (ref s2r.rs:51-84)."""

    def __init__(self):
        self.strategies: List[Tuple[str, Optional[int]]] = []
        self._last_change = ContentContainer()

    def add(self, strategy: str, period: Optional[int] = None):
        self.strategies.append((strategy, period))

    def report(self, window: Window, content: ContentContainer, ts: int) -> bool:
        ok = True
        for strat, period in self.strategies:
            if strat == ReportStrategy.NON_EMPTY_CONTENT:
                ok &= len(content) > 0
            elif strat == ReportStrategy.ON_CONTENT_CHANGE:
                comp = content == self._last_change
                self._last_change = content.clone()
                ok &= comp
            elif strat == ReportStrategy.ON_WINDOW_CLOSE:
                ok &= window.close <= ts
            elif strat == ReportStrategy.PERIODIC:
                ok &= (ts % (period or 1)) == 0
            if not ok:
                return False
        return ok


class CSPARQLWindow:
    """Time-based sliding window operator (ref s2r.rs:175-360)."""

    def __init__(self, width: int, slide: int, report: Optional[Report] = None,
                 tick: str = Tick.TIME_DRIVEN, uri: str = ""):
        self.width = width
        self.slide = slide
        self.t_0 = 0
        self.active_windows: Dict[Window, ContentContainer] = {}
        self.report = report if report is not None else _default_report()
        self.tick = tick
        self.app_time = 0
        self.consumer: Optional[Queue] = None
        self.callback: Optional[Callable[[ContentContainer], None]] = None
        self.uri = uri

    # ------------------------------------------------------------ ingestion
    def scope(self, event_time: int):
        """Open all windows covering `event_time` (ref s2r.rs:298-330)."""
        c_sup = math.ceil(abs(event_time - self.t_0) / self.slide) * self.slide
        o_i = c_sup - self.width
        while True:
            w = Window(max(0, int(o_i)), max(0, int(o_i + self.width)))
            if w not in self.active_windows:
                self.active_windows[w] = ContentContainer(self.uri)
            o_i += self.slide
            if o_i > event_time:
                break

    def add_to_window(self, item: Hashable, ts: int):
        self.scope(ts)
        surviving: Dict[Window, ContentContainer] = {}
        for w, content in self.active_windows.items():
            if w.open <= ts < w.close:
                content.add(item, ts)
                surviving[w] = content
        self._fire(ts)
        self.active_windows = surviving

    def add_probabilistic_to_window(self, occ: ProbabilisticOccurrence):
        ts = occ.event.event_time
        self.scope(ts)
        surviving: Dict[Window, ContentContainer] = {}
        for w, content in self.active_windows.items():
            if w.open <= ts < w.close:
                content.add_probabilistic(occ)
                surviving[w] = content
        self._fire(ts)
        self.active_windows = surviving

    def _fire(self, ts: int):
        """Fire the max-close reporting window (ref s2r.rs:240-266)."""
        reporting = [(w, c) for w, c in self.active_windows.items()
                     if self.report.report(w, c, ts)]
        if not reporting:
            return
        w, content = max(reporting, key=lambda wc: wc[0].close)
        if self.tick == Tick.TIME_DRIVEN:
            if ts > self.app_time:
                self.app_time = ts
                self._emit(content.clone())

    def _emit(self, content: ContentContainer):
        if self.consumer is not None:
            self.consumer.put(content)
        if self.callback is not None:
            self.callback(content)

    # ----------------------------------------------------------- consumers
    def register(self) -> Queue:
        """Channel-style consumer (ref s2r.rs:331)."""
        self.consumer = Queue()
        return self.consumer

    def register_callback(self, fn: Callable[[ContentContainer], None]):
        self.callback = fn

    def flush(self):
        """Merge all active windows and emit (ref s2r.rs:342)."""
        merged = ContentContainer(self.uri)
        for content in self.active_windows.values():
            for occ in content.probabilistic_occurrences:
                merged.add_probabilistic(occ)
            for item, ts in content.iter_with_timestamps():
                if content.is_deterministic(item):
                    merged.add(item, ts)
        if len(merged) > 0:
            self._emit(merged)


def _default_report() -> Report:
    r = Report()
    r.add(ReportStrategy.ON_WINDOW_CLOSE)
    return r
