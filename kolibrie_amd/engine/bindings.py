"""Columnar solution sequences.

The reference's `Bindings = Vec<HashMap<String,u32>>` (shared/src/terms.rs:42)
becomes a struct-of-arrays row table: one int32 device tensor per variable.
Multiset semantics (UNION/VALUES multiplicity) are preserved — rows are never
implicitly deduplicated.  UNBOUND cells hold -1 (0xFFFFFFFF).
"""
from __future__ import annotations

from typing import Dict, Iterable, List, Sequence

import torch

from ..storage.terms import UNBOUND


class Bindings:
    __slots__ = ("cols", "n", "device", "maybe_unbound")

    def __init__(self, cols: Dict[str, torch.Tensor], n: int, device,
                 maybe_unbound: bool = False):
        self.cols = cols
        self.n = n
        self.device = torch.device(device)
        # conservative taint: True if any column MAY contain UNBOUND cells
        # (VALUES UNDEF, UNION fills, SUBJECT/... of non-triples).  False
        # lets hot paths skip per-row boundness scans entirely.
        self.maybe_unbound = maybe_unbound

    # ------------------------------------------------------------- factories
    @staticmethod
    def unit(device) -> "Bindings":
        """The 1-row, 0-column identity (incoming seed for plan roots)."""
        return Bindings({}, 1, device)

    @staticmethod
    def empty(device, vars_: Iterable[str] = ()) -> "Bindings":
        dev = torch.device(device)
        return Bindings(
            {v: torch.empty(0, dtype=torch.int32, device=dev) for v in vars_}, 0, dev
        )

    @staticmethod
    def from_dicts(rows: List[Dict[str, int]], device) -> "Bindings":
        """Host helper (tests/small paths): list of {var: u32 id}."""
        dev = torch.device(device)
        vars_: List[str] = []
        for r in rows:
            for k in r:
                if k not in vars_:
                    vars_.append(k)
        n = len(rows)
        cols = {}
        any_unbound = False
        for v in vars_:
            raw = [r.get(v) for r in rows]
            any_unbound |= any(x is None for x in raw)
            data = [(-1 if x is None else (x & 0xFFFFFFFF)) for x in raw]
            data = [x - 0x1_0000_0000 if x >= 0x8000_0000 else x for x in data]
            cols[v] = torch.tensor(data, dtype=torch.int32, device=dev)
        return Bindings(cols, n, dev, maybe_unbound=any_unbound)

    # ------------------------------------------------------------- accessors
    @property
    def variables(self) -> List[str]:
        return list(self.cols.keys())

    def has(self, var: str) -> bool:
        return var in self.cols

    def col(self, var: str) -> torch.Tensor:
        return self.cols[var]

    def is_empty(self) -> bool:
        return self.n == 0

    def to_dicts(self) -> List[Dict[str, int]]:
        out: List[Dict[str, int]] = []
        host = {v: c.cpu().tolist() for v, c in self.cols.items()}
        for i in range(self.n):
            row = {}
            for v in self.cols:
                x = host[v][i]
                if x != UNBOUND:
                    row[v] = x & 0xFFFFFFFF
            out.append(row)
        return out

    # ------------------------------------------------------------ operations
    def select(self, mask: torch.Tensor) -> "Bindings":
        n = int(mask.sum().item())
        return Bindings({v: c[mask] for v, c in self.cols.items()}, n,
                        self.device, self.maybe_unbound)

    def gather(self, idx: torch.Tensor) -> "Bindings":
        return Bindings({v: c[idx] for v, c in self.cols.items()},
                        idx.numel(), self.device, self.maybe_unbound)

    def with_col(self, var: str, col: torch.Tensor,
                 col_maybe_unbound: bool = False) -> "Bindings":
        cols = dict(self.cols)
        cols[var] = col
        return Bindings(cols, self.n, self.device,
                        self.maybe_unbound or col_maybe_unbound)

    def project(self, vars_: Sequence[str]) -> "Bindings":
        cols = {}
        filled = False
        for v in vars_:
            if v in self.cols:
                cols[v] = self.cols[v]
            else:
                filled = True
                cols[v] = torch.full((self.n,), UNBOUND, dtype=torch.int32, device=self.device)
        return Bindings(cols, self.n, self.device,
                        self.maybe_unbound or (filled and self.n > 0))

    def drop_cols(self, vars_: Sequence[str]) -> "Bindings":
        cols = {v: c for v, c in self.cols.items() if v not in vars_}
        return Bindings(cols, self.n, self.device, self.maybe_unbound)

    @staticmethod
    def concat(parts: List["Bindings"], device) -> "Bindings":
        """Multiset union; missing columns fill with UNBOUND."""
        parts = [p for p in parts if p is not None]
        if not parts:
            return Bindings.empty(device)
        vars_: List[str] = []
        for p in parts:
            for v in p.cols:
                if v not in vars_:
                    vars_.append(v)
        n = sum(p.n for p in parts)
        dev = torch.device(device)
        cols = {}
        filled = False
        for v in vars_:
            pieces = []
            for p in parts:
                if v in p.cols:
                    pieces.append(p.cols[v])
                else:
                    if p.n > 0:
                        filled = True
                    pieces.append(torch.full((p.n,), UNBOUND, dtype=torch.int32, device=dev))
            cols[v] = torch.cat(pieces) if pieces else torch.empty(0, dtype=torch.int32, device=dev)
        tainted = filled or any(p.maybe_unbound for p in parts)
        return Bindings(cols, n, dev, tainted)

    def repeat_rows(self, k: int) -> "Bindings":
        """Each row repeated k times consecutively."""
        return Bindings(
            {v: torch.repeat_interleave(c, k) for v, c in self.cols.items()},
            self.n * k,
            self.device,
            self.maybe_unbound,
        )

    def tile_rows(self, k: int) -> "Bindings":
        """Whole table repeated k times."""
        return Bindings({v: c.repeat(k) for v, c in self.cols.items()},
                        self.n * k, self.device, self.maybe_unbound)
