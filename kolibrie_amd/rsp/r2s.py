"""R2S: relation-to-stream operators (ref: kolibrie/src/rsp/r2s.rs:15-63).

StreamOperator::{RSTREAM, ISTREAM, DSTREAM}; Relation2StreamOperator::eval
diffs consecutive result sets: RSTREAM emits everything, ISTREAM emits the
rows new since the previous evaluation, DSTREAM the rows deleted.

Two equivalent paths:
  - `eval`: host set semantics over decoded row tuples (multi-window /
    static-join emissions, which are already host-side);
  - `eval_columns`: the K10 device path — window results stay int32
    columns, consecutive sets diff via tensor_utils.rows_diff, and only
    the (usually tiny) Δ is decoded at the consumer boundary.  The RSP
    engine uses it structurally for single-window emissions, so the
    operator's state representation never flips mid-stream.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple


class StreamOperator:
    RSTREAM = "RSTREAM"
    ISTREAM = "ISTREAM"
    DSTREAM = "DSTREAM"


class Relation2StreamOperator:
    def __init__(self, operator: str = StreamOperator.RSTREAM):
        self.operator = operator
        self.previous: set = set()
        self.previous_cols: Optional[list] = None  # device columns

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

    def eval_columns(self, cols: list, ts: int = 0) -> list:
        """Device/columnar evaluation (K10 rows_diff): `cols` is a list of
        same-length int32 tensors (one per variable); returns the emitted
        set as columns of the same shape."""
        from ..engine.tensor_utils import rows_diff, unique_rows
        import torch
        if not cols:
            return []
        current = unique_rows(list(cols)) if cols[0].numel() else list(cols)
        prev = self.previous_cols
        if self.operator == StreamOperator.RSTREAM:
            out = list(cols)
        elif self.operator == StreamOperator.ISTREAM:
            out = current if prev is None else rows_diff(current, prev)
        elif self.operator == StreamOperator.DSTREAM:
            if prev is None:
                out = [c[:0] for c in current]
            else:
                out = rows_diff(prev, current)
        else:
            raise ValueError(f"unknown stream operator {self.operator}")
        self.previous_cols = current
        return out
