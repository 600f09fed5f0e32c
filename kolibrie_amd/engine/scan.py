"""Index scan / probe over sorted column permutations.

This is the engine's torch mirror of HIP kernel K1 (`scan_probe` — SURVEY
§2.9): for each incoming binding row, probe the best-matching (s,p,o)-bound
index permutation with a binary-search range, emit extended rows.  The torch
path vectorizes the probe as a batched searchsorted + repeat_interleave
expansion — identical math to the kernel's per-wave binary search + two-pass
emit, so it serves as the differential-test oracle.

Column materialization is LAZY: `need` names the (s,p,o) positions the
caller will actually read (projection pushdown reaches the kernels) — an
unneeded position is returned as None and never unpacked from HBM.

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

ALL_POSITIONS = frozenset((0, 1, 2))


def choose_order(bound: set, consts: set = frozenset(),
                 prefer_second: int = None) -> Tuple[int, int]:
    """Pick the order with the deepest bound prefix.

    Tie-breaks: (1) prefer an order whose LEADING component is a constant —
    the binary search then converges into that predicate's contiguous
    region, which stays cache-resident; (2) prefer the order whose SECOND
    component matches `prefer_second` — a plen-1 slice is sorted by it, and
    subject-sorted (PSO) slices make downstream joins merge joins.

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
        pref = (prefer_second is not None and ln == 1 and b == prefer_second)
        better = (ln > best_len
                  or (ln == best_len and lead_const and not best_const)
                  or (ln == best_len and lead_const == best_const
                      and pref and not best_pref))
        if better:
            best, best_len, best_const, best_pref = code, ln, lead_const, pref
    return best, best_len


def _cols_from_order(idx: GraphIndex, code: int, sel: Optional[torch.Tensor],
                     need: frozenset = ALL_POSITIONS):
    """Materialize the requested (s,p,o) positions for selected rows."""
    key12, z = idx.orders[code]
    pos = _ORDER_POS[code]
    out = [None, None, None]
    k = key12[sel] if sel is not None else key12
    if pos[0] in need and pos[1] in need:
        a, b = unpack2(k)
        out[pos[0]], out[pos[1]] = a, b
    elif pos[0] in need:
        out[pos[0]] = (k >> 32).to(torch.int32)
    elif pos[1] in need:
        out[pos[1]] = (k & 0xFFFFFFFF).to(torch.int32)
    if pos[2] in need:
        out[pos[2]] = z[sel] if sel is not None else z
    return out[0], out[1], out[2]


def _cols_from_order_slice(idx: GraphIndex, code: int, lo: int, hi: int,
                           need: frozenset = ALL_POSITIONS):
    """Contiguous-range variant: views + unpack, no index gather."""
    key12, z = idx.orders[code]
    pos = _ORDER_POS[code]
    out = [None, None, None]
    k = key12[lo:hi]
    if pos[0] in need and pos[1] in need:
        a, b = unpack2(k)
        out[pos[0]], out[pos[1]] = a, b
    elif pos[0] in need:
        out[pos[0]] = (k >> 32).to(torch.int32)
    elif pos[1] in need:
        out[pos[1]] = (k & 0xFFFFFFFF).to(torch.int32)
    if pos[2] in need:
        out[pos[2]] = z[lo:hi]
    return out[0], out[1], out[2]


def scan_unit(idx: GraphIndex, consts: Dict[int, int],
              sort_hint: int = None, need=None):
    """Scan with only constant bounds: one contiguous range slice.

    consts: {position: i32 value}; returns (s,p,o) int32 columns (positions
    outside `need` are None).
    """
    dev = idx.device
    need_set = ALL_POSITIONS if need is None else frozenset(need)
    if idx.n == 0:
        e = torch.empty(0, dtype=torch.int32, device=dev)
        return tuple(e.clone() if i in need_set else None for i in range(3))
    code, plen = choose_order(set(consts.keys()), set(consts.keys()),
                              prefer_second=sort_hint)
    pos = _ORDER_POS[code]
    key12, z = idx.orders[code]
    uncovered = set(consts.keys()) - set(pos[:plen])
    mat_need = frozenset(need_set | uncovered)
    if plen == 0:
        s, p, o = _cols_from_order(idx, SPO, None, mat_need)
        n_rows = idx.n
    else:
        # packed bounds computed as PYTHON ints (pack2 convention:
        # sign-extended hi << 32 | unsigned lo) and both resolved in ONE
        # searchsorted + ONE device sync — point queries do 5+ of these
        # per request, so the saved tensor creates/syncs are measurable
        if plen == 2:
            k = (consts[pos[0]] << 32) | (consts[pos[1]] & 0xFFFFFFFF)
            k2 = k + 1  # side='right' of k == side='left' of k+1 (int keys)
        else:
            k = consts[pos[0]] << 32
            k2 = k + 0x1_0000_0000
        cached = idx.rcache.get((code, k))
        if cached is not None:
            lo, hi = cached
        else:
            if k2 > 0x7FFF_FFFF_FFFF_FFFF:
                probe = torch.tensor([k], dtype=torch.int64, device=dev)
                lo = int(torch.searchsorted(key12, probe, side="left").item())
                hi = idx.n
            else:
                probe = torch.tensor([k, k2], dtype=torch.int64, device=dev)
                lo, hi = torch.searchsorted(key12, probe, side="left").tolist()
            if len(idx.rcache) < 65536:
                idx.rcache[(code, k)] = (lo, hi)
        s, p, o = _cols_from_order_slice(idx, code, lo, hi, mat_need)
        n_rows = hi - lo
    # post-filter constants not covered by the prefix
    if uncovered:
        mask = None
        for position in uncovered:
            col = (s, p, o)[position]
            m = col == consts[position]
            mask = m if mask is None else (mask & m)
        cols = tuple(c[mask] if c is not None else None for c in (s, p, o))
        s, p, o = cols
    # drop post-filter-only columns the caller did not ask for
    return tuple(
        c if (i in need_set) else None for i, c in enumerate((s, p, o))
    )


def scan_probe(
    idx: GraphIndex,
    consts: Dict[int, int],
    probes: Dict[int, torch.Tensor],
    need=None,
):
    """Per-row index probe (K1).  probes: {position: int32 value per row}.

    Returns (row_idx, s, p, o): row_idx indexes the probe rows; each probe
    row expands to its matching triples.  Positions outside `need` are None.
    Constants and probe positions not covered by the chosen order's prefix
    are post-filtered.
    """
    dev = idx.device
    need_set = ALL_POSITIONS if need is None else frozenset(need)
    n_rows = next(iter(probes.values())).numel()
    e = torch.empty(0, dtype=torch.int32, device=dev)
    el = torch.empty(0, dtype=torch.long, device=dev)
    if idx.n == 0 or n_rows == 0:
        return (el,) + tuple(e.clone() if i in need_set else None
                             for i in range(3))
    bound = set(consts.keys()) | set(probes.keys())
    code, plen = choose_order(bound, set(consts.keys()))
    pos = _ORDER_POS[code]
    key12, z = idx.orders[code]
    covered = set(pos[:plen])
    post_positions = bound - covered
    mat_need = frozenset(need_set | post_positions)

    def col_for(position: int) -> torch.Tensor:
        if position in probes:
            return probes[position]
        return torch.full((n_rows,), consts[position], dtype=torch.int32, device=dev)

    # ---- native K1 path on device --------------------------------------
    from ..ops import native_for
    native = native_for(key12)
    if native is not None and plen > 0:
        if plen == 2:
            a_col = probes[pos[0]].contiguous() if pos[0] in probes else None
            a_const = consts.get(pos[0], 0)
            b_colp = probes[pos[1]].contiguous() if pos[1] in probes else None
            b_const = consts.get(pos[1], 0)
            li, b_col, z_col = native.probe_fused(
                key12, z, a_col, a_const, b_colp, b_const, [], True)
        else:
            li, b_col, z_col = native.probe_range(key12, z,
                                                  col_for(pos[0]).contiguous())
        out = [None, None, None]
        if pos[0] in mat_need:
            out[pos[0]] = col_for(pos[0])[li]
        if pos[1] in mat_need:
            out[pos[1]] = b_col
        if pos[2] in mat_need:
            out[pos[2]] = z_col
        s, p, o = out
        mask = None
        for position in post_positions:
            col = (s, p, o)[position]
            want = consts[position] if position in consts else probes[position][li]
            m = col == want
            mask = m if mask is None else (mask & m)
        if mask is not None:
            li = li[mask]
            s, p, o = (c[mask] if c is not None else None for c in (s, p, o))
        return (li,) + tuple(
            c if i in need_set else None for i, c in enumerate((s, p, o)))

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
        lo = torch.zeros(n_rows, dtype=torch.long, device=dev)
        hi = torch.full((n_rows,), idx.n, dtype=torch.long, device=dev)
    cnt = hi - lo
    total = int(cnt.sum().item())
    if total == 0:
        return (el,) + tuple(e.clone() if i in need_set else None
                             for i in range(3))
    li = torch.repeat_interleave(
        torch.arange(n_rows, dtype=torch.long, device=dev), cnt)
    starts = torch.cumsum(cnt, 0) - cnt
    offs = torch.arange(total, dtype=torch.long, device=dev) - starts[li]
    sel = lo[li] + offs
    s, p, o = _cols_from_order(idx, code, sel, mat_need)
    mask = None
    for position in post_positions:
        col = (s, p, o)[position]
        want = consts[position] if position in consts else probes[position][li]
        m = col == want
        mask = m if mask is None else (mask & m)
    if mask is not None:
        li = li[mask]
        s, p, o = (c[mask] if c is not None else None for c in (s, p, o))
    return (li,) + tuple(
        c if i in need_set else None for i, c in enumerate((s, p, o)))


def scan_probe_count(idx: GraphIndex, consts: Dict[int, int],
                     probes: Dict[int, torch.Tensor]) -> Optional[int]:
    """COUNT(*)-only probe: total match count with no emit pass.

    Returns None when a post-filter position would be required (caller
    falls back to the emitting path)."""
    n_rows = next(iter(probes.values())).numel()
    if idx.n == 0 or n_rows == 0:
        return 0
    bound = set(consts.keys()) | set(probes.keys())
    code, plen = choose_order(bound, set(consts.keys()))
    pos = _ORDER_POS[code]
    if plen == 0 or (bound - set(pos[:plen])):
        return None
    key12, _z = idx.orders[code]

    def col_for(position: int) -> torch.Tensor:
        if position in probes:
            return probes[position]
        return torch.full((n_rows,), consts[position], dtype=torch.int32,
                          device=idx.device)

    from ..ops import native_for
    native = native_for(key12)
    if native is not None:
        if plen == 2:
            keys = pack2(col_for(pos[0]).contiguous(),
                         col_for(pos[1]).contiguous())
            cnt = native.probe_exact_counts(key12, keys.contiguous())
        else:
            cnt = native.probe_range_counts(key12, col_for(pos[0]).contiguous())
        return int(cnt.sum().item())
    if plen == 2:
        keys = pack2(col_for(pos[0]), col_for(pos[1]))
        lo = torch.searchsorted(key12, keys, side="left")
        hi = torch.searchsorted(key12, keys, side="right")
    else:
        v = col_for(pos[0])
        lo = torch.searchsorted(key12, pack2(v, torch.zeros_like(v)), side="left")
        hi = torch.searchsorted(key12, pack2(v, torch.full_like(v, -1)), side="right")
    return int((hi - lo).sum().item())


def scan_unit_count(idx: GraphIndex, consts: Dict[int, int]) -> Optional[int]:
    """Row count of a constant-bound scan without materializing columns
    (COUNT(*) pushdown).  Returns None when post-filtering would be needed
    (caller falls back to the materializing path)."""
    if idx.n == 0:
        return 0
    code, plen = choose_order(set(consts.keys()), set(consts.keys()))
    pos = _ORDER_POS[code]
    if set(consts.keys()) - set(pos[:plen]):
        return None
    if plen == 0:
        return idx.n
    dev = idx.device
    key12, _z = idx.orders[code]
    if plen == 2:
        k = pack2(torch.tensor([consts[pos[0]]], dtype=torch.int32, device=dev),
                  torch.tensor([consts[pos[1]]], dtype=torch.int32, device=dev))
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
    return hi - lo


def scan_probe_carry(idx: GraphIndex, consts: Dict[int, int],
                     probes: Dict[int, torch.Tensor], need,
                     carry: Dict[str, torch.Tensor]):
    """K1 probe with carry columns: the emit pass gathers the incoming
    rows' surviving columns directly (fused kernel — no separate pack2
    pass, no separate per-column gathers).

    Returns (li, s, p, o, carried: {name: tensor}).
    """
    dev = idx.device
    need_set = ALL_POSITIONS if need is None else frozenset(need)
    bound = set(consts.keys()) | set(probes.keys())
    code, plen = choose_order(bound, set(consts.keys()))
    pos = _ORDER_POS[code]
    post_positions = bound - set(pos[:plen])
    from ..ops import native_for
    key12, z = idx.orders[code]
    native = native_for(key12) if idx.n else None
    n_rows = next(iter(probes.values())).numel()
    if (native is None or plen != 2 or post_positions or len(carry) > 4
            or n_rows == 0 or idx.n == 0):
        li, s, p, o = scan_probe(idx, consts, probes, need)
        carried = {k: v[li] for k, v in carry.items()}
        return li, s, p, o, carried

    def col_or_const(position):
        if position in probes:
            return probes[position].contiguous(), 0
        return None, consts[position]

    a_col, a_const = col_or_const(pos[0])
    b_col, b_const = col_or_const(pos[1])
    names = list(carry.keys())
    emit_b = pos[1] in need_set
    out = native.probe_fused(key12, z, a_col, a_const, b_col, b_const,
                             [carry[k].contiguous() for k in names], emit_b)
    li, b_out, z_out = out[0], out[1], out[2]
    carried = {k: t for k, t in zip(names, out[3:])}
    cols = [None, None, None]
    if pos[0] in need_set:
        src0 = probes[pos[0]] if pos[0] in probes else None
        if src0 is not None:
            cols[pos[0]] = src0[li]
        else:
            cols[pos[0]] = torch.full((li.numel(),), consts[pos[0]],
                                      dtype=torch.int32, device=dev)
    if emit_b:
        cols[pos[1]] = b_out
    if pos[2] in need_set:
        cols[pos[2]] = z_out
    return li, cols[0], cols[1], cols[2], carried
