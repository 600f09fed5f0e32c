"""Index scan / probe over sorted column permutations.

This is the engine's torch mirror of HIP kernel K1 (`scan_probe` — SURVEY
§2.9): for each incoming binding row, probe the best-matching (s,p,o)-bound
index permutation with a binary-search range, emit extended rows.  The torch
path vectorizes the probe as a batched searchsorted + repeat_interleave
expansion — identical math to the kernel's per-wave binary search + two-pass
emit, so it serves as the differential-test oracle.

Ref parity: engine.rs:1018-1251 (execute_quad_scan_with_ids / match_quad),
dataset_index.rs:223-344 (8-way bound-pattern dispatch).
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from ..storage.dataset import OSP, POS, PSO, SPO, GraphIndex
from ..storage.terms import UNBOUND
from .tensor_utils import pack2, unpack2

# order -> (leading col, second col, trailing col) position indices (0=s,1=p,2=o)
_ORDER_POS = {SPO: (0, 1, 2), POS: (1, 2, 0), OSP: (2, 0, 1), PSO: (1, 0, 2)}


def choose_order(bound: set, consts: set = frozenset(),
                 prefer_second: int = None) -> Tuple[int, int]:
    """Pick the order with the deepest bound prefix.

    Tie-break: prefer an order whose LEADING component is a constant — the
    binary search then converges into that predicate's contiguous region,
    which stays cache-resident (measured 2.5x on MI355X vs probes spread
    over the whole array).

    Returns (order_code, prefix_len in {0,1,2}).
    """
    best, best_len, best_const, best_pref = SPO, 0, False, False
    for code, (a, b, _c) in _ORDER_POS.items():
        ln = 0
        if a in bound:
            ln = 1
            if b in bound:
                ln = 2
        lead_const = a in consts
        # a plen-1 slice is sorted by its SECOND component: honoring a
        # sort hint makes downstream joins merge joins
        pref = (prefer_second is not None and ln == 1 and b == prefer_second)
        better = (ln > best_len
                  or (ln == best_len and lead_const and not best_const)
                  or (ln == best_len and lead_const == best_const
                      and pref and not best_pref))
        if better:
            best, best_len, best_const, best_pref = code, ln, lead_const, pref
    return best, best_len


def _cols_from_order(idx: GraphIndex, code: int, sel: torch.Tensor
                     ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Materialize (s,p,o) for selected row indices of an order."""
    key12, z = idx.orders[code]
    a, b = unpack2(key12[sel] if sel is not None else key12)
    c = z[sel] if sel is not None else z
    pos = _ORDER_POS[code]
    out = [None, None, None]
    out[pos[0]] = a
    out[pos[1]] = b
    out[pos[2]] = c
    return out[0], out[1], out[2]  # type: ignore


def scan_unit(idx: GraphIndex, consts: Dict[int, int],
              sort_hint: int = None
              ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Scan with only constant bounds: one contiguous range slice.

    consts: {position: i32 value}; returns (s,p,o) int32 columns.
    """
    dev = idx.device
    if idx.n == 0:
        e = torch.empty(0, dtype=torch.int32, device=dev)
        return e, e.clone(), e.clone()
    code, plen = choose_order(set(consts.keys()), set(consts.keys()),
                              prefer_second=sort_hint)
    pos = _ORDER_POS[code]
    key12, z = idx.orders[code]
    if plen == 0:
        s, p, o = _cols_from_order(idx, SPO, None)
    else:
        if plen == 2:
            k = pack2(
                torch.tensor([consts[pos[0]]], dtype=torch.int32, device=dev),
                torch.tensor([consts[pos[1]]], dtype=torch.int32, device=dev),
            )
            lo = int(torch.searchsorted(key12, k, side="left").item())
            hi = int(torch.searchsorted(key12, k, side="right").item())
        else:
            v = consts[pos[0]]
            klo = pack2(torch.tensor([v], dtype=torch.int32, device=dev),
                        torch.tensor([0], dtype=torch.int32, device=dev))
            khi = pack2(torch.tensor([v], dtype=torch.int32, device=dev),
                        torch.tensor([-1], dtype=torch.int32, device=dev))
            lo = int(torch.searchsorted(key12, klo, side="left").item())
            hi = int(torch.searchsorted(key12, khi, side="right").item())
        sel = torch.arange(lo, hi, dtype=torch.long, device=dev)
        s, p, o = _cols_from_order(idx, code, sel)
    # post-filter remaining constants not covered by the prefix
    mask = None
    for position, val in consts.items():
        col = (s, p, o)[position]
        m = col == val
        mask = m if mask is None else (mask & m)
    if mask is not None and plen < len(consts):
        s, p, o = s[mask], p[mask], o[mask]
    return s, p, o


def scan_probe(
    idx: GraphIndex,
    consts: Dict[int, int],
    probes: Dict[int, torch.Tensor],
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Per-row index probe (K1).  probes: {position: int32 value per row}.

    Returns (row_idx, s, p, o): row_idx indexes the probe rows; each probe
    row expands to its matching triples.  Constants and probe positions not
    covered by the chosen order's prefix are post-filtered.
    """
    dev = idx.device
    n_rows = next(iter(probes.values())).numel()
    e = torch.empty(0, dtype=torch.int32, device=dev)
    el = torch.empty(0, dtype=torch.long, device=dev)
    if idx.n == 0 or n_rows == 0:
        return el, e, e.clone(), e.clone()
    bound = set(consts.keys()) | set(probes.keys())
    code, plen = choose_order(bound, set(consts.keys()))
    pos = _ORDER_POS[code]
    key12, z = idx.orders[code]

    def col_for(position: int) -> torch.Tensor:
        if position in probes:
            return probes[position]
        return torch.full((n_rows,), consts[position], dtype=torch.int32, device=dev)

    # ---- native K1 path on device --------------------------------------
    from ..ops import native_for
    native = native_for(key12)
    if native is not None and plen > 0:
        if plen == 2:
            keys = pack2(col_for(pos[0]).contiguous(), col_for(pos[1]).contiguous())
            li, b_col, z_col = native.probe_exact(key12, z, keys.contiguous())
        else:
            li, b_col, z_col = native.probe_range(key12, z,
                                                  col_for(pos[0]).contiguous())
        out = [None, None, None]
        out[pos[0]] = col_for(pos[0])[li]
        out[pos[1]] = b_col
        out[pos[2]] = z_col
        s, p, o = out  # type: ignore[assignment]
        covered = set(pos[:plen])
        mask = None
        for position in bound - covered:
            col = (s, p, o)[position]
            want = consts[position] if position in consts else probes[position][li]
            m = col == want
            mask = m if mask is None else (mask & m)
        if mask is not None:
            li, s, p, o = li[mask], s[mask], p[mask], o[mask]
        return li, s, p, o

    if plen == 2:
        keys = pack2(col_for(pos[0]), col_for(pos[1]))
        lo = torch.searchsorted(key12, keys, side="left")
        hi = torch.searchsorted(key12, keys, side="right")
    elif plen == 1:
        v = col_for(pos[0])
        klo = pack2(v, torch.zeros_like(v))
        khi = pack2(v, torch.full_like(v, -1))
        lo = torch.searchsorted(key12, klo, side="left")
        hi = torch.searchsorted(key12, khi, side="right")
    else:
        # no probe position matches any order prefix (cannot happen: probes
        # non-empty means at least one position is bound)
        lo = torch.zeros(n_rows, dtype=torch.long, device=dev)
        hi = torch.full((n_rows,), idx.n, dtype=torch.long, device=dev)
    cnt = hi - lo
    total = int(cnt.sum().item())
    if total == 0:
        return el, e, e.clone(), e.clone()
    li = torch.repeat_interleave(
        torch.arange(n_rows, dtype=torch.long, device=dev), cnt)
    starts = torch.cumsum(cnt, 0) - cnt
    offs = torch.arange(total, dtype=torch.long, device=dev) - starts[li]
    sel = lo[li] + offs
    s, p, o = _cols_from_order(idx, code, sel)
    # post-filter positions not in the prefix
    covered = set(pos[:plen])
    mask = None
    for position in bound - covered:
        col = (s, p, o)[position]
        want = consts[position] if position in consts else probes[position][li]
        m = col == want
        mask = m if mask is None else (mask & m)
    if mask is not None:
        li, s, p, o = li[mask], s[mask], p[mask], o[mask]
    return li, s, p, o
