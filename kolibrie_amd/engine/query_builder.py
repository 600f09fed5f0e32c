"""Fluent QueryBuilder (ref: kolibrie/src/query_builder.rs:20-758 and the
PyO3 surface python/src/py_query_builder.rs:441-740).

Surface parity: with_subject/predicate/object (+ _like/_starting/_ending
variants), filter closures, cross-DB join on s/p/o or custom key fns,
distinct / order_by / asc / desc / limit / offset / group_by (+
group_by_subject/predicate/object dictionaries, count); streaming:
window(width, slide), report/tick strategy (+ with_* aliases, periodic
reports with periods, stream operator), as_stream, add_stream_triple,
get_stream_results / get_all_stream_results / clear_stream_results,
stop_stream, is_streaming and the get_*_config introspection surface.
"""
from __future__ import annotations

from typing import Callable, List, Optional, Tuple

TripleStr = Tuple[str, str, str]


class QueryBuilder:
    def __init__(self, db):
        self.db = db
        self._preds: List[Callable[[TripleStr], bool]] = []
        self._distinct = False
        self._order_by: Optional[Callable[[TripleStr], object]] = None
        self._order_desc = False
        self._limit: Optional[int] = None
        self._offset = 0
        self._group_by: Optional[Callable[[TripleStr], object]] = None
        # streaming state
        self._window: Optional[Tuple[int, int]] = None
        self._reports: List[Tuple[str, Optional[int]]] = []
        self._tick: Optional[str] = None
        self._stream_operator: Optional[str] = None
        self._stream_window = None
        self._stream_results: List[List[TripleStr]] = []

    # ------------------------------------------------------------ filtering
    def with_subject(self, s: str) -> "QueryBuilder":
        return self._add(lambda t: t[0] == s)

    def with_predicate(self, p: str) -> "QueryBuilder":
        return self._add(lambda t: t[1] == p)

    def with_object(self, o: str) -> "QueryBuilder":
        return self._add(lambda t: t[2] == o)

    def with_subject_like(self, frag: str) -> "QueryBuilder":
        return self._add(lambda t: frag in t[0])

    def with_predicate_like(self, frag: str) -> "QueryBuilder":
        return self._add(lambda t: frag in t[1])

    def with_predicate_starting(self, pre: str) -> "QueryBuilder":
        return self._add(lambda t: t[1].startswith(pre))

    def with_predicate_ending(self, suf: str) -> "QueryBuilder":
        return self._add(lambda t: t[1].endswith(suf))

    def with_object_like(self, frag: str) -> "QueryBuilder":
        return self._add(lambda t: frag in t[2])

    def with_subject_starting(self, pre: str) -> "QueryBuilder":
        return self._add(lambda t: t[0].startswith(pre))

    def with_subject_ending(self, suf: str) -> "QueryBuilder":
        return self._add(lambda t: t[0].endswith(suf))

    def with_object_starting(self, pre: str) -> "QueryBuilder":
        return self._add(lambda t: t[2].startswith(pre))

    def with_object_ending(self, suf: str) -> "QueryBuilder":
        return self._add(lambda t: t[2].endswith(suf))

    def filter(self, fn: Callable[[TripleStr], bool]) -> "QueryBuilder":
        return self._add(fn)

    def _add(self, fn) -> "QueryBuilder":
        self._preds.append(fn)
        return self

    # ------------------------------------------------------------ modifiers
    def distinct(self) -> "QueryBuilder":
        self._distinct = True
        return self

    def order_by(self, key: Callable[[TripleStr], object],
                 descending: bool = False) -> "QueryBuilder":
        self._order_by = key
        self._order_desc = descending
        return self

    def limit(self, n: int) -> "QueryBuilder":
        self._limit = n
        return self

    def offset(self, n: int) -> "QueryBuilder":
        self._offset = n
        return self

    def group_by(self, key: Callable[[TripleStr], object]) -> "QueryBuilder":
        self._group_by = key
        return self

    def asc(self) -> "QueryBuilder":
        """Sort ascending (by full triple unless order_by set a key)."""
        if self._order_by is None:
            self._order_by = lambda t: t
        self._order_desc = False
        return self

    def desc(self) -> "QueryBuilder":
        if self._order_by is None:
            self._order_by = lambda t: t
        self._order_desc = True
        return self

    # ------------------------------------------------- terminal shortcuts
    def count(self) -> int:
        rows = self.execute()
        return len(rows) if not isinstance(rows, dict) else sum(
            len(v) for v in rows.values())

    def get_decoded_triples(self) -> List[TripleStr]:
        rows = self.execute()
        if isinstance(rows, dict):
            return [t for v in rows.values() for t in v]
        return rows

    def _grouped_by_pos(self, pos: int):
        groups = {}
        for t in self.get_decoded_triples():
            groups.setdefault(t[pos], []).append(t)
        return groups

    def _project(self, pos: int) -> List[str]:
        out = [t[pos] for t in self.get_decoded_triples()]
        if self._distinct:
            out = sorted(set(out))
        return out

    def get_subjects(self) -> List[str]:
        """Decoded subjects of the matches (sorted-unique under
        distinct(), ref query_builder.rs:365)."""
        return self._project(0)

    def get_predicates(self) -> List[str]:
        return self._project(1)

    def get_objects(self) -> List[str]:
        return self._project(2)

    def group_by_subject(self):
        """Dict subject -> matching triples (ref py_query_builder.rs:709)."""
        return self._grouped_by_pos(0)

    def group_by_predicate(self):
        return self._grouped_by_pos(1)

    def group_by_object(self):
        return self._grouped_by_pos(2)

    # ------------------------------------------------------------ execution
    def _all_triples(self) -> List[TripleStr]:
        return [t for t in self.db.triples_as_strings()]

    def execute(self):
        rows = [t for t in self._all_triples()
                if all(p(t) for p in self._preds)]
        if self._distinct:
            seen = set()
            uniq = []
            for t in rows:
                if t not in seen:
                    seen.add(t)
                    uniq.append(t)
            rows = uniq
        if self._order_by is not None:
            rows.sort(key=self._order_by, reverse=self._order_desc)
        if self._offset:
            rows = rows[self._offset:]
        if self._limit is not None:
            rows = rows[:self._limit]
        if self._group_by is not None:
            groups = {}
            for t in rows:
                groups.setdefault(self._group_by(t), []).append(t)
            return groups
        return rows

    # ------------------------------------------------------- cross-DB joins
    def join(self, other_db, on: str = "s",
             key_fns: Optional[Tuple[Callable, Callable]] = None
             ) -> List[Tuple[TripleStr, TripleStr]]:
        """Join this builder's rows with another DB's triples on s/p/o or
        custom key functions (ref query_builder.rs cross-DB join)."""
        pos = {"s": 0, "p": 1, "o": 2}.get(on)
        lk = key_fns[0] if key_fns else (lambda t: t[pos])
        rk = key_fns[1] if key_fns else (lambda t: t[pos])
        left = self.execute()
        right = other_db.triples_as_strings()
        index = {}
        for t in right:
            index.setdefault(rk(t), []).append(t)
        out = []
        for lt in left:
            for rt in index.get(lk(lt), []):
                out.append((lt, rt))
        return out

    # ------------------------------------------------------------ streaming
    def window(self, width: int, slide: int) -> "QueryBuilder":
        self._window = (width, slide)
        return self

    def report_strategy(self, strategy: str,
                        period: Optional[int] = None) -> "QueryBuilder":
        self._reports.append((strategy, period))
        return self

    with_report_strategy = report_strategy

    def with_periodic_report(self, period: int) -> "QueryBuilder":
        from ..rsp.s2r import ReportStrategy
        return self.report_strategy(ReportStrategy.PERIODIC, period)

    def tick_strategy(self, tick: str) -> "QueryBuilder":
        self._tick = tick
        return self

    with_tick_strategy = tick_strategy

    def with_stream_operator(self, operator: str) -> "QueryBuilder":
        self._stream_operator = operator
        return self

    # ----------------------------------------------- stream introspection
    def get_window_config(self) -> Optional[Tuple[int, int]]:
        return self._window

    def get_report_strategies(self) -> List[str]:
        return [r[0] for r in self._reports]

    def get_periodic_periods(self) -> List[int]:
        return [r[1] for r in self._reports if r[1] is not None]

    def get_tick_strategy(self) -> Optional[str]:
        from ..rsp.s2r import Tick
        return self._tick or Tick.TIME_DRIVEN

    def get_stream_operator(self) -> Optional[str]:
        return self._stream_operator

    def is_streaming(self) -> bool:
        return self._stream_window is not None

    def stop_stream(self) -> "QueryBuilder":
        self._stream_window = None
        return self

    def clear_stream_results(self) -> "QueryBuilder":
        self._stream_results.clear()
        return self

    def get_all_stream_results(self) -> List[List[TripleStr]]:
        return self._stream_results

    def as_stream(self) -> "QueryBuilder":
        from ..rsp.s2r import CSPARQLWindow, Report, ReportStrategy, Tick
        if self._window is None:
            raise ValueError("as_stream requires window(width, slide)")
        rep = Report()
        if self._reports:
            for strat, period in self._reports:
                rep.add(strat, period)
        else:
            rep.add(ReportStrategy.ON_WINDOW_CLOSE)
        self._stream_window = CSPARQLWindow(
            self._window[0], self._window[1], rep,
            self._tick or Tick.TIME_DRIVEN, "querybuilder")
        self._stream_window.register_callback(self._on_window)
        return self

    def _on_window(self, content):
        rows = [t for t in content.items()
                if all(p(t) for p in self._preds)]
        self._stream_results.append(sorted(rows))

    def add_stream_triple(self, triple: TripleStr, ts: int) -> "QueryBuilder":
        if self._stream_window is None:
            raise ValueError("call as_stream() first")
        self._stream_window.add_to_window(tuple(triple), ts)
        return self

    def get_stream_results(self) -> List[List[TripleStr]]:
        return self._stream_results
