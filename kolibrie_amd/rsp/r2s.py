"""R2S: relation-to-stream operators (ref: kolibrie/src/rsp/r2s.rs:15-63).

StreamOperator::{RSTREAM, ISTREAM, DSTREAM}; Relation2StreamOperator::eval
diffs consecutive result sets: RSTREAM emits everything, ISTREAM emits the
rows new since the previous evaluation, DSTREAM the rows deleted.  Set
semantics over result rows (device K10 rows_diff for large sets).
"""
from __future__ import annotations

from typing import List, Sequence, Tuple


class StreamOperator:
    RSTREAM = "RSTREAM"
    ISTREAM = "ISTREAM"
    DSTREAM = "DSTREAM"


class Relation2StreamOperator:
    def __init__(self, operator: str = StreamOperator.RSTREAM):
        self.operator = operator
        self.previous: set = set()

    def eval(self, new_result: Sequence[Tuple], ts: int = 0) -> List[Tuple]:
        current = set(map(tuple, new_result))
        if self.operator == StreamOperator.RSTREAM:
            out = list(map(tuple, new_result))
        elif self.operator == StreamOperator.ISTREAM:
            out = sorted(current - self.previous)
        elif self.operator == StreamOperator.DSTREAM:
            out = sorted(self.previous - current)
        else:
            raise ValueError(f"unknown stream operator {self.operator}")
        self.previous = current
        return out
