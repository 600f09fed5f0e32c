"""SELECT finalization: GROUP BY aggregation, ORDER BY, DISTINCT,
LIMIT/OFFSET, projection.

Ref parity: execute_query.rs:404-475 (aggregate_rows: COUNT/SUM/AVG/MIN/MAX),
:477 (numeric-aware ORDER BY), engine.rs:785-905 (subquery finalize).

MI355X-native: aggregation runs on the device value column (float64 parsed
once at encode time — K4 class segmented reduce) instead of per-row string
parsing; only final aggregate *results* are encoded back into the host
dictionary.
"""
from __future__ import annotations

from typing import Dict, List

import torch

from ..parsing.ast import Projection, SelectQuery
from ..storage.terms import UNBOUND
from .bindings import Bindings
from .tensor_utils import group_index, lexsort, unique_rows


def _fmt_num(v: float) -> str:
    if v != v:  # NaN
        return "0"
    if v == int(v) and abs(v) < 1e15:
        return str(int(v))
    return repr(v)


def _encode_i32(db, s: str) -> int:
    x = db.dictionary.encode(s) & 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


def finalize_select_bindings(select: SelectQuery, rows: Bindings, db
                             ) -> Bindings:
    """ID-level finalize (used for subqueries and as the core of the
    top-level path)."""
    dev = rows.device
    has_agg = any(p.aggregate for p in select.variables)

    if has_agg or select.group_by:
        rows = _aggregate(select, rows, db)
        having = getattr(select, "having", None)
        if having is not None and rows.n > 0:
            # HAVING (engine extension): filter the aggregated rows; the
            # aggregate columns are interned numerals, so the normal
            # FILTER machinery (value column) evaluates them
            from .filters import CompiledExpr
            mask = CompiledExpr(having, db, dict(db.prefixes)).eval_mask(
                rows, db)
            rows = rows.select(mask)
    # ORDER BY first — it may sort on non-projected variables
    # (numeric-aware, ref execute_query.rs:477)
    if select.order_by and rows.n > 1:
        perm = _order_perm(select, rows, db)
        rows = rows.gather(perm)
    # projection
    if select.select_star or not select.variables:
        proj_names = rows.variables
    else:
        proj_names = [p.output_name() for p in select.variables]
        rows = rows.project(proj_names)
    if select.distinct and rows.n > 1:
        cols = [rows.col(v) for v in proj_names]
        if select.order_by:
            # stable distinct preserving order: keep first occurrence
            gid, _ = group_index(cols)
            first = torch.zeros(rows.n, dtype=torch.bool, device=dev)
            seen: Dict[int, bool] = {}
            gl = gid.cpu().tolist()
            keep_idx = []
            for i, g in enumerate(gl):
                if g not in seen:
                    seen[g] = True
                    keep_idx.append(i)
            rows = rows.gather(torch.tensor(keep_idx, dtype=torch.long, device=dev))
        else:
            uc = unique_rows(cols)
            rows = Bindings(dict(zip(proj_names, uc)),
                            uc[0].numel() if uc else 0, dev)
    # OFFSET / LIMIT
    off = select.offset or 0
    if off or select.limit is not None:
        end = rows.n if select.limit is None else min(rows.n, off + select.limit)
        idx = torch.arange(off, max(off, end), dtype=torch.long, device=dev)
        rows = rows.gather(idx)
    return rows


def _order_perm(select: SelectQuery, rows: Bindings, db) -> torch.Tensor:
    dev = rows.device
    perm = torch.arange(rows.n, dtype=torch.long, device=dev)
    for cond in reversed(select.order_by):
        if not rows.has(cond.var):
            continue
        ids = rows.col(cond.var)[perm]
        ids_u = ids.to(torch.int64) & 0xFFFFFFFF
        strs = [db.dictionary.decode(int(x)) or "" for x in ids_u.cpu().tolist()]
        all_numeric = all(_is_num(s) for s in strs) and len(strs) > 0
        if all_numeric:
            vc = db.value_column()
            from .tensor_utils import values_for_ids
            vals = values_for_ids(vc, ids_u)
            key = torch.argsort(vals, stable=True, descending=cond.descending)
        else:
            import numpy as np
            order = sorted(range(len(strs)), key=lambda i: strs[i],
                           reverse=cond.descending)
            key = torch.tensor(order, dtype=torch.long, device=dev)
        perm = perm[key]
    return perm


def _is_num(s: str) -> bool:
    try:
        float(s)
        return True
    except ValueError:
        return False


def _aggregate(select: SelectQuery, rows: Bindings, db) -> Bindings:
    """GROUP BY + aggregates on the device value column (K4 class)."""
    dev = rows.device
    group_vars = [v for v in select.group_by if rows.has(v)]
    n = rows.n
    if n == 0:
        out_cols = {}
        for v in group_vars:
            out_cols[v] = torch.empty(0, dtype=torch.int32, device=dev)
        for p in select.variables:
            if p.aggregate:
                out_cols[p.output_name()] = torch.empty(0, dtype=torch.int32, device=dev)
        if not group_vars:
            # aggregates over empty input: COUNT=0, others empty-string
            vals = {}
            for p in select.variables:
                if p.aggregate == "COUNT":
                    vals[p.output_name()] = _encode_i32(db, "0")
                elif p.aggregate:
                    vals[p.output_name()] = _encode_i32(db, "")
            return Bindings(
                {k: torch.tensor([v], dtype=torch.int32, device=dev)
                 for k, v in vals.items()},
                1, dev)
        return Bindings(out_cols, 0, dev)

    if not group_vars:
        # single-group fast path: no scatter needed (COUNT/SUM/... over all)
        out_cols: Dict[str, torch.Tensor] = {}
        vc = db.value_column()
        for p in select.variables:
            if not p.aggregate:
                continue
            name = p.output_name()
            if p.aggregate == "COUNT":
                if p.agg_arg is None:
                    c = n
                else:
                    col = rows.col(p.agg_arg) if rows.has(p.agg_arg) else None
                    if col is None:
                        c = 0
                    elif p.distinct:
                        bound = col[col != UNBOUND]
                        c = int(torch.unique(bound).numel())
                    else:
                        c = int((col != UNBOUND).sum().item())
                out_cols[name] = torch.tensor([_encode_i32(db, str(c))],
                                              dtype=torch.int32, device=dev)
                continue
            col = rows.col(p.agg_arg) if p.agg_arg and rows.has(p.agg_arg) else None
            if col is None:
                out_cols[name] = torch.tensor([_encode_i32(db, "")],
                                              dtype=torch.int32, device=dev)
                continue
            ids_u = col.to(torch.int64) & 0xFFFFFFFF
            from .tensor_utils import values_for_ids
            vals = values_for_ids(vc, ids_u)
            vals = vals[col != UNBOUND]
            if p.aggregate == "SUM":
                r = float(vals.sum().item()) if vals.numel() else 0.0
            elif p.aggregate == "AVG":
                r = float(vals.mean().item()) if vals.numel() else 0.0
            elif p.aggregate == "MIN":
                r = float(vals.min().item()) if vals.numel() else 0.0
            elif p.aggregate == "MAX":
                r = float(vals.max().item()) if vals.numel() else 0.0
            else:
                raise ValueError(f"unknown aggregate {p.aggregate}")
            out_cols[name] = torch.tensor([_encode_i32(db, _fmt_num(r))],
                                          dtype=torch.int32, device=dev)
        return Bindings(out_cols, 1, dev)

    native_out = _native_group_aggregate(select, rows, db, group_vars)
    if native_out is not None:
        return native_out

    gid, ng = group_index([rows.col(v) for v in group_vars])
    # representative row per group (first occurrence)
    rep = torch.full((ng,), -1, dtype=torch.long, device=dev)
    rev = torch.arange(n - 1, -1, -1, dtype=torch.long, device=dev)
    rep.scatter_(0, gid.flip(0), rev)  # last write wins => first row index
    out_cols: Dict[str, torch.Tensor] = {}
    for v in group_vars:
        out_cols[v] = rows.col(v)[rep]
    vc = db.value_column()
    for p in select.variables:
        if not p.aggregate:
            if p.var and p.var not in out_cols and rows.has(p.var):
                out_cols[p.var] = rows.col(p.var)[rep]
            continue
        name = p.output_name()
        if p.aggregate == "COUNT":
            if p.agg_arg is None:
                cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
                cnt.scatter_add_(0, gid, torch.ones(n, dtype=torch.int64, device=dev))
            else:
                col = rows.col(p.agg_arg) if rows.has(p.agg_arg) else None
                if col is None:
                    cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
                elif p.distinct:
                    pair = unique_rows([gid.to(torch.int32), col])
                    bound = pair[1] != UNBOUND
                    cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
                    cnt.scatter_add_(0, pair[0][bound].to(torch.int64),
                                     torch.ones(int(bound.sum()), dtype=torch.int64, device=dev))
                else:
                    bound = (col != UNBOUND).to(torch.int64)
                    cnt = torch.zeros(ng, dtype=torch.int64, device=dev)
                    cnt.scatter_add_(0, gid, bound)
            out_cols[name] = _encode_numbers(db, cnt.to(torch.float64), dev, integral=True)
            continue
        col = rows.col(p.agg_arg) if p.agg_arg and rows.has(p.agg_arg) else None
        if col is None:
            out_cols[name] = torch.full((ng,), _encode_i32(db, ""),
                                        dtype=torch.int32, device=dev)
            continue
        ids_u = col.to(torch.int64) & 0xFFFFFFFF
        from .tensor_utils import values_for_ids
        vals = values_for_ids(vc, ids_u)
        bound = col != UNBOUND
        if p.aggregate in ("SUM", "AVG"):
            acc = torch.zeros(ng, dtype=torch.float64, device=dev)
            acc.scatter_add_(0, gid[bound], vals[bound])
            if p.aggregate == "AVG":
                cnt = torch.zeros(ng, dtype=torch.float64, device=dev)
                cnt.scatter_add_(0, gid[bound],
                                 torch.ones(int(bound.sum()), dtype=torch.float64, device=dev))
                acc = torch.where(cnt > 0, acc / torch.clamp(cnt, min=1), acc)
            out_cols[name] = _encode_numbers(db, acc, dev)
        elif p.aggregate in ("MIN", "MAX"):
            init = float("inf") if p.aggregate == "MIN" else float("-inf")
            acc = torch.full((ng,), init, dtype=torch.float64, device=dev)
            acc.scatter_reduce_(0, gid[bound], vals[bound],
                                reduce="amin" if p.aggregate == "MIN" else "amax")
            acc = torch.where(torch.isinf(acc), torch.zeros_like(acc), acc)
            out_cols[name] = _encode_numbers(db, acc, dev)
        else:
            raise ValueError(f"unknown aggregate {p.aggregate}")
    return Bindings(out_cols, ng, dev)


def _native_group_aggregate(select: SelectQuery, rows: Bindings, db,
                            group_vars) -> "Bindings | None":
    """K4 hand-written LDS-staged hash aggregate (ops/csrc/kernels.hip
    group_aggregate; ref semantics execute_query.rs:404-475).

    Eligible shape: device rows, 1-2 fully-bound group keys, aggregates
    drawn from COUNT(*)/COUNT(x)/SUM/AVG/MIN/MAX over at most ONE distinct
    argument, no DISTINCT aggregates, no non-group plain projections.
    Returns None to fall back to the torch composite path."""
    from ..ops import native_for
    dev = rows.device
    if dev.type != "cuda" or rows.maybe_unbound:
        return None
    if not (1 <= len(group_vars) <= 2) or len(group_vars) != len(select.group_by):
        return None
    arg = None
    want_sum = want_min = want_max = False
    for p in select.variables:
        if not p.aggregate:
            if p.var not in group_vars:
                return None  # representative-row projection: torch path
            continue
        if p.distinct:
            return None
        if p.aggregate == "COUNT":
            if p.agg_arg is not None:
                if not rows.has(p.agg_arg):
                    return None
                # all rows bound (maybe_unbound False) => COUNT(x)==COUNT(*)
            continue
        if p.aggregate not in ("SUM", "AVG", "MIN", "MAX"):
            return None
        if p.agg_arg is None or not rows.has(p.agg_arg):
            return None
        if arg is None:
            arg = p.agg_arg
        elif arg != p.agg_arg:
            return None  # two different value columns: torch path
        if p.aggregate in ("SUM", "AVG"):
            want_sum = True
        if p.aggregate == "MIN":
            want_min = True
        if p.aggregate == "MAX":
            want_max = True
    native = native_for(rows.col(group_vars[0]))
    if native is None:
        return None
    if len(group_vars) == 1:
        keys = rows.col(group_vars[0]).to(torch.int64) & 0xFFFFFFFF
    else:
        a = rows.col(group_vars[0]).to(torch.int64) & 0xFFFFFFFF
        b = rows.col(group_vars[1]).to(torch.int64) & 0xFFFFFFFF
        keys = (a << 32) | b
    # sampled cardinality estimate: sizes the hash table in one shot and
    # routes mostly-unique keys (where a sort beats hashing) to the torch
    # path.  Uniform-groups model: E[u] = g(1 - exp(-s/g)) inverted by a
    # few fixed-point steps.
    n = keys.numel()
    groups_hint = 0
    if n > 300_000:
        stride = max(1, n // 65536)
        sample = keys[::stride]
        s_n = sample.numel()
        u = int(torch.unique(sample).numel())
        r = u / max(1, s_n)
        if r > 0.9:
            return None  # ~all-unique: sort-based composite wins
        import math
        g = float(u)
        for _ in range(20):
            g = u / max(1e-9, 1.0 - math.exp(-s_n / max(g, 1.0)))
        groups_hint = max(2048, min(n, int(2.0 * g)))
    vals = None
    if arg is not None:
        from .tensor_utils import values_for_ids
        ids_u = rows.col(arg).to(torch.int64) & 0xFFFFFFFF
        vals = values_for_ids(db.value_column(), ids_u).to(torch.float64)
    gkeys, cnt, gsum, gmn, gmx = native.group_aggregate(
        keys, vals, want_sum, want_min, want_max, -1, groups_hint)
    out_cols: Dict[str, torch.Tensor] = {}
    if len(group_vars) == 1:
        out_cols[group_vars[0]] = gkeys.to(torch.int32)
    else:
        out_cols[group_vars[0]] = (gkeys >> 32).to(torch.int32)
        out_cols[group_vars[1]] = (gkeys & 0xFFFFFFFF).to(torch.int32)
    for p in select.variables:
        if not p.aggregate:
            continue
        name = p.output_name()
        if p.aggregate == "COUNT":
            out_cols[name] = _encode_numbers(db, cnt.to(torch.float64), dev,
                                             integral=True)
        elif p.aggregate == "SUM":
            out_cols[name] = _encode_numbers(db, gsum, dev)
        elif p.aggregate == "AVG":
            out_cols[name] = _encode_numbers(
                db, gsum / cnt.to(torch.float64).clamp(min=1), dev)
        elif p.aggregate == "MIN":
            out_cols[name] = _encode_numbers(db, gmn, dev)
        else:
            out_cols[name] = _encode_numbers(db, gmx, dev)
    return Bindings(out_cols, gkeys.numel(), dev)


def _encode_numbers(db, vals: torch.Tensor, dev, integral: bool = False
                    ) -> torch.Tensor:
    """Encode aggregate results back into the dictionary.

    Vectorized: only the UNIQUE values cross to the host for string
    interning (aggregate outputs cluster heavily — a 1M-group COUNT has a
    few thousand distinct counts), then a device gather rebuilds the full
    column.  The former per-group host loop was the VERDICT r1 item 3
    bottleneck."""
    n = vals.numel()
    if n == 0:
        return torch.empty(0, dtype=torch.int32, device=dev)
    if n > 64:
        uniq, inv = torch.unique(vals, return_inverse=True)
        if uniq.numel() < n:
            ids = _encode_numbers(db, uniq, dev, integral)
            return ids[inv]
    out = []
    for v in vals.cpu().tolist():
        s = str(int(v)) if integral else _fmt_num(v)
        out.append(_encode_i32(db, s))
    return torch.tensor(out, dtype=torch.int32, device=dev)


from contextlib import contextmanager


@contextmanager
def _gc_paused():
    """Pause the cyclic GC around bulk Python-container construction:
    building tens of thousands of row lists triggers generational
    collections that traverse torch's module graph — measured 7.9 ms vs
    0.7 ms for a 14k-row pivot with the GC paused."""
    import gc
    was = gc.isenabled()
    if was:
        gc.disable()
    try:
        yield
    finally:
        if was:
            gc.enable()


def _host_ids(rows: Bindings, names) -> Dict[str, "object"]:
    """ONE device->host transfer for every requested column: the
    per-column .cpu() each cost a full device sync (~15 us on ROCm), and
    point queries decode 5-6 columns — batching them was worth ~30% of a
    small SELECT's end-to-end time."""
    present = [v for v in names if rows.has(v)]
    if not present:
        return {}
    if len(present) == 1:
        v = present[0]
        return {v: (rows.col(v).to(torch.int64) & 0xFFFFFFFF).cpu().numpy()}
    m = torch.stack([rows.col(v).to(torch.int64) & 0xFFFFFFFF
                     for v in present]).cpu().numpy()
    return {v: m[i] for i, v in enumerate(present)}


def _decode_column(rows: Bindings, v: str, db, ids=None) -> List[str]:
    """Batch-decode one result column: single D2H copy + numpy gather over
    the cached dictionary mirror; quoted triples (rare) decode per cell."""
    import numpy as np
    n = rows.n
    if not rows.has(v):
        return [""] * n
    table = db.dictionary.np_table()
    n_plain = len(table)
    if ids is None:
        ids = (rows.col(v).to(torch.int64) & 0xFFFFFFFF).cpu().numpy()
    # default "": UNBOUND and out-of-vocabulary plain ids (synthetic dense
    # id blocks have no string form) both decode to ""
    out = np.full(n, "", dtype=object)
    quoted = (ids & 0x8000_0000) != 0
    ok = (ids < n_plain) & ~quoted
    out[ok] = table[ids[ok]]
    annex = db.dictionary.annex
    if annex is not None:
        # bulk-vocabulary annex ids batch-decode natively (only the
        # REQUESTED ids materialize as Python strings)
        amask = ~quoted & (ids >= n_plain) & (ids != 0xFFFFFFFF)
        if amask.any():
            idxs = np.nonzero(amask)[0]
            lst = annex[0].vocab_decode_batch(
                annex[1], torch.from_numpy(ids[idxs].astype(np.int64)))
            arr = np.array(lst, dtype=object)
            arr[np.equal(arr, None)] = ""
            out[idxs] = arr
    for i in np.nonzero(quoted & (ids != 0xFFFFFFFF))[0]:
        out[i] = db.decode_term(int(ids[i])) or ""
    return out.tolist()


def decode_columns(select: SelectQuery, rows: Bindings, db
                   ) -> Dict[str, List[str]]:
    """Columnar result decode: {var: [values...]} with NO per-row Python
    list construction — the MI355X-native result shape (sub-ms at 14k
    rows where the row pivot alone costs ~1.6 ms)."""
    if select.select_star or not select.variables:
        names = rows.variables
    else:
        names = [p.output_name() for p in select.variables]
    host = _host_ids(rows, names)
    return {v: _decode_column(rows, v, db, host.get(v)) for v in names}


def decode_rows(select: SelectQuery, rows: Bindings, db) -> List[List[str]]:
    """Final string decode (only at the top, ref engine.rs:347-373).

    Row-heavy results decode in BATCH: one D2H copy per column, then a
    numpy object-array gather over the cached dictionary mirror — the
    per-cell Python loop this replaces dominated materializing queries
    (L5-class: 14k rows took ~3 ms of host decode)."""
    if select.select_star or not select.variables:
        names = rows.variables
    else:
        names = [p.output_name() for p in select.variables]
    n = rows.n
    if n >= 64:
        with _gc_paused():
            host_ids = _host_ids(rows, names)
            cols_dec = [_decode_column(rows, v, db, host_ids.get(v))
                        for v in names]
            # row pivot: map/zip beats any numpy object-array reshape
            # (object stack+tolist measured 9.5 ms at 14k rows); the
            # remaining cost IS building the row lists — decode_columns
            # below avoids it entirely for columnar consumers
            return list(map(list, zip(*cols_dec)))
    host_np = _host_ids(rows, names)
    host = {}
    for v in names:
        if v in host_np:
            host[v] = host_np[v].tolist()
        else:
            host[v] = [None] * rows.n
    # plain (non-quoted, interned) ids decode by direct list index — the
    # per-cell function call dominates row-heavy results otherwise
    id2s = db.dictionary.id_to_str
    n_plain = len(id2s)
    out: List[List[str]] = []
    for i in range(rows.n):
        row = []
        for v in names:
            x = host[v][i]
            if x is None or x == 0xFFFFFFFF:
                row.append("")
            elif x < n_plain and not (x & 0x8000_0000):
                row.append(id2s[x] or "")
            else:
                row.append(db.decode_term(x) or "")
        out.append(row)
    return out
