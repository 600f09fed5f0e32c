"""Vectorized multi-column row-set primitives over torch tensors.

These run identically on CPU (test oracle / plumbing path) and on MI355X
device tensors (where the hot ones are superseded by the HIP kernels in
kolibrie_amd/ops).  All row tables are struct-of-arrays: a list of equal
length 1-D tensors.

Conventions: IDs are int32 two's-complement views of u32 dictionary IDs.
Sort order is the *signed* int order of each column — semantically arbitrary
but used consistently everywhere (build and probe), which is all that range
lookups and equality grouping need.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import torch

Rows = Sequence[torch.Tensor]


def lexsort(cols: Rows) -> torch.Tensor:
    """Return a permutation sorting rows lexicographically by cols[0..k).

    Implemented as chained stable sorts from the least-significant key up —
    the classic radix-style lexsort that maps directly onto device sort
    passes.
    """
    n = cols[0].numel()
    if n == 0:
        return torch.empty(0, dtype=torch.long, device=cols[0].device)
    perm = torch.arange(n, dtype=torch.long, device=cols[0].device)
    for c in reversed(list(cols)):
        perm = perm[torch.argsort(c[perm], stable=True)]
    return perm


def rows_equal_prev(cols: Rows) -> torch.Tensor:
    """Bool mask m[i]=True iff row i equals row i-1 (m[0]=False).

    Rows must already be sorted so equal rows are adjacent.
    """
    n = cols[0].numel()
    dev = cols[0].device
    if n == 0:
        return torch.empty(0, dtype=torch.bool, device=dev)
    eq = torch.ones(n, dtype=torch.bool, device=dev)
    eq[0] = False
    for c in cols:
        eq[1:] &= c[1:] == c[:-1]
    return eq


def unique_rows(cols: Rows) -> List[torch.Tensor]:
    """Sort rows lexicographically and drop duplicates (set semantics)."""
    perm = lexsort(cols)
    sorted_cols = [c[perm] for c in cols]
    dup = rows_equal_prev(sorted_cols)
    keep = ~dup
    return [c[keep] for c in sorted_cols]


def sort_rows(cols: Rows) -> List[torch.Tensor]:
    perm = lexsort(cols)
    return [c[perm] for c in cols]


def _run_ids(sorted_cols: Rows) -> torch.Tensor:
    """Dense group index per row for lexicographically sorted rows."""
    eq = rows_equal_prev(sorted_cols)
    return torch.cumsum(~eq, dim=0) - 1


def group_index(cols: Rows) -> Tuple[torch.Tensor, int]:
    """Map each row to a dense group id in [0, n_groups); order-insensitive.

    Returns (gid per original row, n_groups).  This is the generic
    key-compression step that turns a k-column equality key into one int64,
    feeding single-key joins/aggregates.
    """
    n = cols[0].numel()
    if n == 0:
        return torch.empty(0, dtype=torch.long, device=cols[0].device), 0
    perm = lexsort(cols)
    sorted_cols = [c[perm] for c in cols]
    rid = _run_ids(sorted_cols)
    gid = torch.empty(n, dtype=torch.long, device=cols[0].device)
    gid[perm] = rid
    ng = int(rid[-1].item()) + 1 if n else 0
    return gid, ng


def rows_diff(a_cols: Rows, b_cols: Rows) -> List[torch.Tensor]:
    """Set difference A \\ B over row sets (dedups A).  K10-class op."""
    a_cols = unique_rows(a_cols)
    n_a = a_cols[0].numel()
    if n_a == 0 or b_cols[0].numel() == 0:
        return list(a_cols)
    b_cols = unique_rows(b_cols)
    mask = ~membership_mask(a_cols, b_cols)
    return [c[mask] for c in a_cols]


def membership_mask(a_cols: Rows, b_sorted_unique: Rows) -> torch.Tensor:
    """For each row of A, True iff it occurs in B.

    B must be lexicographically sorted (dups allowed).  Vectorized via the
    combined-sort run trick: works for any column count without key packing.
    """
    n_a = a_cols[0].numel()
    dev = a_cols[0].device
    if n_a == 0:
        return torch.empty(0, dtype=torch.bool, device=dev)
    n_b = b_sorted_unique[0].numel()
    if n_b == 0:
        return torch.zeros(n_a, dtype=torch.bool, device=dev)
    comb = [torch.cat([a, b]) for a, b in zip(a_cols, b_sorted_unique)]
    flag = torch.cat([
        torch.zeros(n_a, dtype=torch.int8, device=dev),
        torch.ones(n_b, dtype=torch.int8, device=dev),
    ])
    perm = lexsort(comb)
    sorted_cols = [c[perm] for c in comb]
    rid = _run_ids(sorted_cols)
    ng = int(rid[-1].item()) + 1
    has_b = torch.zeros(ng, dtype=torch.int8, device=dev)
    has_b.scatter_reduce_(0, rid, flag[perm], reduce="amax")
    hit_sorted = has_b[rid].to(torch.bool) & (flag[perm] == 0)
    mask = torch.zeros(n_a + n_b, dtype=torch.bool, device=dev)
    mask[perm] = hit_sorted
    return mask[:n_a]


def merge_join_indices(
    left_key: torch.Tensor, right_key: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Equi-join two single-column int64 keys, returning (li, ri) index pairs.

    Torch-composite sort-merge join — the portable fallback for the HIP hash
    join (K2).  Multiset semantics: every matching pair is emitted.
    """
    dev = left_key.device
    nl, nr = left_key.numel(), right_key.numel()
    if nl == 0 or nr == 0:
        e = torch.empty(0, dtype=torch.long, device=dev)
        return e, e
    r_perm = torch.argsort(right_key)
    r_sorted = right_key[r_perm]
    lo = torch.searchsorted(r_sorted, left_key, side="left")
    hi = torch.searchsorted(r_sorted, left_key, side="right")
    cnt = hi - lo
    li = torch.repeat_interleave(torch.arange(nl, dtype=torch.long, device=dev), cnt)
    total = int(cnt.sum().item())
    if total == 0:
        e = torch.empty(0, dtype=torch.long, device=dev)
        return e, e
    # offsets within each left row's match range
    starts = torch.cumsum(cnt, 0) - cnt
    pos = torch.arange(total, dtype=torch.long, device=dev) - starts[li]
    ri = r_perm[lo[li] + pos]
    return li, ri


def pack2(hi: torch.Tensor, lo: torch.Tensor) -> torch.Tensor:
    """Pack two int32 (u32-view) columns into one int64 key.

    Injective on (hi, lo); ordering is the consistent-but-arbitrary signed
    order (fine for grouping and range lookups done with the same packing).
    """
    return (hi.to(torch.int64) << 32) | (lo.to(torch.int64) & 0xFFFFFFFF)


def unpack2(key: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    hi = (key >> 32).to(torch.int32)
    lo = (key & 0xFFFFFFFF).to(torch.int32)
    return hi, lo


def values_for_ids(vc, ids_u):
    """f64 values for u32-view id tensor; ids outside the interned
    vocabulary -> 0.0 (same guard as the K5 kernel's `u < value_n`)."""
    import torch
    if vc.numel() == 0:
        return torch.zeros(ids_u.numel(), dtype=torch.float64,
                           device=ids_u.device)
    in_range = ids_u < vc.numel()
    vals = vc[torch.clamp(ids_u, max=vc.numel() - 1)]
    return torch.where(in_range, vals,
                       torch.zeros((), dtype=torch.float64,
                                   device=ids_u.device))
