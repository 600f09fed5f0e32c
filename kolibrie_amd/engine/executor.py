"""The execution engine — recursive physical-plan interpreter.

Ref parity: streamertail_optimizer/execution/engine.rs (1 618 LoC).  The
universal contract is preserved:

    execute(op, ctx, incoming: Bindings) -> Bindings     (engine.rs:407)

but Bindings are columnar device tables and every operator is a vectorized
device op (torch fallback here; the HIP kernels in kolibrie_amd/ops take
over the hot paths on gfx950).  GRAPH semantics, merged-FROM dedup default,
bind-join dependency, hash-join multiset semantics and quoted-triple
matching follow engine.rs:946-1403.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..plan.physical import (
    PBind, PBindJoin, PExchange, PFilter, PHashJoin, PIndexScan,
    PInMemoryBuffer,
    PConstStar, PLeftJoin, PMLPredict, PMinus, PNestedLoopJoin, PStarJoin,
    PSubquery, PTableScan, PUnion, PUnit, PValues, PhysicalOp,
)
from ..storage.dataset import DEFAULT_GRAPH, GraphIndex
from ..storage.terms import (
    Constant, QuotedTriplePattern, TriplePattern, UNBOUND, Variable,
)
from .bindings import Bindings
from .scan import scan_unit
from .tensor_utils import group_index, merge_join_indices
from . import exec_stats
from . import tracer

import logging

_log = logging.getLogger("kolibrie_amd.engine")


@dataclass
class DatasetView:
    """FROM / FROM NAMED dataset view (ref execute_query.rs:228)."""
    default_graphs: List[int] = field(default_factory=lambda: [DEFAULT_GRAPH])
    named_graphs: Optional[List[int]] = None  # None = all named graphs


@dataclass
class ExecutionContext:
    db: object
    view: DatasetView = field(default_factory=DatasetView)
    # distributed execution (planner-emitted PExchange ops): world size and
    # this process's rank; 1/0 = single-process (exchanges are identity)
    world: int = 1
    rank: int = 0

    def default_index(self) -> GraphIndex:
        gs = self.view.default_graphs
        if len(gs) == 1:
            return self.db.store.graph_index(gs[0])
        return self.db.store.merged_index(gs)

    def named_graph_ids(self) -> List[int]:
        if self.view.named_graphs is not None:
            return self.view.named_graphs
        return self.db.store.named_graph_ids()


def _subtree_has_exchange(op: PhysicalOp) -> bool:
    """True if the subtree contains a PExchange.  Distributed collectives
    must run on EVERY rank, so empty-input short-circuits may not skip a
    subtree containing one (a rank-local empty intermediate would
    deadlock the other ranks).  Memoized on the plan node."""
    cached = getattr(op, "_has_exch", None)
    if cached is not None:
        return cached
    found = isinstance(op, PExchange)
    if not found:
        for name in ("input", "left", "right"):
            child = getattr(op, name, None)
            if isinstance(child, PhysicalOp) and _subtree_has_exchange(child):
                found = True
                break
    try:
        op._has_exch = found
    except AttributeError:
        pass
    return found


class ExecutionEngine:
    def __init__(self, ctx: ExecutionContext):
        self.ctx = ctx
        self.db = ctx.db
        self.device = ctx.db.device

    # ----------------------------------------------------------- dispatch --
    def execute(self, op: PhysicalOp, incoming: Bindings) -> Bindings:
        if tracer.is_enabled():
            sp = tracer.span(type(op).__name__, self.device)
            try:
                return self._execute(op, incoming)
            finally:
                sp.close()
        return self._execute(op, incoming)

    def _execute(self, op: PhysicalOp, incoming: Bindings) -> Bindings:
        if isinstance(op, PUnit):
            return incoming
        needed = getattr(op, "needed", None)
        if (needed is not None and len(needed) == 0
                and isinstance(op, (PBindJoin, PHashJoin))
                and incoming.n == 1 and not incoming.cols):
            fused = self._try_chain_count(op)
            if fused is not None:
                return Bindings({}, fused, self.device)
        if isinstance(op, (PTableScan, PIndexScan)):
            return self._exec_scan(op.pattern, op.graph, incoming, needed,
                                   sort_hint=getattr(op, "sort_hint", None))
        if isinstance(op, PConstStar):
            return self._exec_const_star(op, incoming, needed)
        if isinstance(op, PStarJoin):
            out = incoming
            for i, pat in enumerate(op.patterns):
                # keep vars required by later patterns of the chain
                step_needed = None
                if needed is not None:
                    step_needed = set(needed) | {op.join_var}
                    for later in op.patterns[i + 1:]:
                        step_needed.update(later.variables())
                out = self._exec_scan(pat, op.graph, out, step_needed,
                                      sort_hint=0 if i == 0 else None)
                if out.is_empty():
                    break
            return out
        if isinstance(op, PHashJoin):
            left = self.execute(op.left, incoming)
            if left.is_empty() and not _subtree_has_exchange(op.right):
                return left
            right = self.execute(op.right, Bindings.unit(self.device))
            return join_bindings(left, right, needed)
        if isinstance(op, PBindJoin):
            left = self.execute(op.left, incoming)
            if left.is_empty() and not _subtree_has_exchange(op.right):
                return left
            return self.execute(op.right, left)
        if isinstance(op, PNestedLoopJoin):
            left = self.execute(op.left, incoming)
            if left.is_empty() and not _subtree_has_exchange(op.right):
                return left
            right = self.execute(op.right, Bindings.unit(self.device))
            return join_bindings(left, right, needed)
        if isinstance(op, PExchange):
            rows = self.execute(op.input, incoming)
            return self._exec_exchange(op, rows)
        if isinstance(op, PUnion):
            l = self.execute(op.left, incoming)
            r = self.execute(op.right, incoming)
            return Bindings.concat([l, r], self.device)
        if isinstance(op, PFilter):
            rows = self.execute(op.input, incoming)
            if rows.is_empty():
                return rows
            mask = op.condition.eval_mask(rows, self.db)
            return _prune(rows.select(mask), needed)
        if isinstance(op, PBind):
            rows = self.execute(op.input, incoming)
            if rows.is_empty():
                return rows.project(rows.variables + [op.var])
            col = op.expr.eval_ids(rows, self.db)
            taint = getattr(op.expr, "may_produce_unbound", False)
            return _prune(rows.with_col(op.var, col, taint), needed)
        if isinstance(op, PValues):
            rows = self.execute(op.input, incoming)
            vals = self._values_bindings(op.variables, op.rows)
            return join_bindings(rows, vals, needed)
        if isinstance(op, PSubquery):
            rows = self.execute(op.input, incoming)
            sub = self._exec_subquery(op.select)
            return join_bindings(rows, sub)
        if isinstance(op, PLeftJoin):
            left = self.execute(op.left, incoming)
            if left.is_empty() and not _subtree_has_exchange(op.right):
                return left
            right = self.execute(op.right, Bindings.unit(self.device))
            if left.is_empty():
                return left
            return left_outer_join(left, right, needed)
        if isinstance(op, PMinus):
            left = self.execute(op.left, incoming)
            if left.is_empty() and not _subtree_has_exchange(op.right):
                return left
            from ..plan.physical import op_certain_vars, op_possible_vars
            possible = op_possible_vars(op.right)
            if (possible is not None
                    and not _subtree_has_exchange(op.right)
                    and not any(v in left.cols for v in possible)):
                # SPARQL MINUS with statically disjoint domains removes
                # nothing — skip evaluating the right side entirely
                return left
            certain = op_certain_vars(op.right)
            if (self.ctx.world == 1 and not left.maybe_unbound
                    and left.n <= 5_000_000
                    and any(v in left.cols for v in certain)):
                # semi-join pushdown: evaluate the right side PROBED by the
                # left rows instead of scanning it in full — the probed
                # result carries every left column, so anti-join
                # compatibility reduces to "this left row found a match".
                # Sound only because (a) every left shared var is bound
                # (maybe_unbound is False) and (b) every right solution
                # binds >=1 var shared with the left (certain-vars check),
                # so the spec's dom-intersection requirement always holds.
                right = self.execute(op.right, left)
            else:
                right = self.execute(op.right, Bindings.unit(self.device))
            return anti_join(left, right)
        if isinstance(op, PInMemoryBuffer):
            return join_bindings(incoming, op.bindings)
        if isinstance(op, PMLPredict):
            rows = self.execute(op.input, incoming)
            from ..ml.predict import execute_ml_predict
            return execute_ml_predict(op.info, rows, self.db)
        raise ValueError(f"cannot execute {type(op).__name__}")

    def _exec_exchange(self, op: PExchange, rows: Bindings) -> Bindings:
        """Planner-emitted re-partition.

        hash:      row i -> rank (row[var] & 0xFFFFFFFF) % world — the same
                   partition function the subject-hash loader uses, so an
                   exchanged intermediate is co-located with the shard rows
                   of any pattern whose subject is `var`.
        broadcast: replicate the (small) table on every rank (broadcast
                   join build side / MINUS-OPTIONAL right sides).
        rank0:     keep rows only on rank 0 (turns a replicated source into
                   a valid partitioned one, e.g. a VALUES branch of UNION).
        """
        world = self.ctx.world
        if world <= 1:
            return rows
        from ..parallel import dist as D
        if not D.is_dist():
            return rows
        if op.mode == "rank0":
            if self.ctx.rank == 0:
                return rows
            return Bindings.empty(self.device, rows.variables) \
                if rows.cols else Bindings({}, 0, self.device)
        # column-set synchronization: empty intermediates can carry fewer
        # columns than non-empty ones on other ranks (pruned empty-join
        # paths, probed scans on empty incoming, OPTIONAL early returns) —
        # the per-column collectives below would then desynchronize and
        # deadlock.  Exchange the column-name lists, use the stable union,
        # and fill locally-missing columns with UNBOUND.
        import torch.distributed as dist
        meta = [None] * world
        dist.all_gather_object(meta, (list(rows.variables),
                                      bool(rows.maybe_unbound)))
        names = list(dict.fromkeys(v for lst, _mu in meta for v in lst))
        tainted = any(mu for _lst, mu in meta) or \
            any(set(lst) != set(names) for lst, _mu in meta)
        def _col(v):
            if rows.has(v):
                return rows.col(v)
            return torch.full((rows.n,), UNBOUND, dtype=torch.int32,
                              device=self.device)
        if op.mode == "broadcast":
            if not names:
                n = D.allreduce_sum_scalar(rows.n, self.device)
                return Bindings({}, n, self.device)
            cols = D.all_gather_rows([_col(v) for v in names])
            return Bindings(dict(zip(names, cols)), cols[0].numel(),
                            self.device, maybe_unbound=tainted)
        # hash re-partition on op.var
        if op.var not in names:
            raise ValueError(
                f"PExchange(hash) key ?{op.var} missing from row table "
                f"columns {names} — planner/runtime mismatch")
        key = _col(op.var).to(torch.int64) & 0xFFFFFFFF
        dest = key % world
        cols = D.all_to_all_rows([_col(v) for v in names], dest)
        return Bindings(dict(zip(names, cols)),
                        cols[0].numel() if cols else 0, self.device,
                        maybe_unbound=tainted)

    def _exec_const_star(self, op, incoming: Bindings, needed) -> Bindings:
        """Bound-subject star: ONE SPO-region fetch + host-side pattern
        evaluation (the region is the subject's handful of triples).  One
        device sync total instead of two per pattern."""
        idx = self.ctx.default_index()
        sid = op.subject_id - 0x1_0000_0000 if op.subject_id >= 0x8000_0000 \
            else op.subject_id
        _s, p_col, o_col = scan_unit(idx, {0: sid}, need={1, 2})
        n = p_col.numel()
        vars_ = [v for _pid, v in op.items]
        if n == 0 or n > 8192:
            if n == 0:
                star = Bindings.empty(self.device, vars_)
                return star
            # hub subject: fall back to per-pattern device masks
            star = None
            for pid, var in op.items:
                pm = p_col == pid
                b = Bindings({var: o_col[pm]}, int(pm.sum().item()),
                             self.device)
                star = b if star is None else join_bindings(star, b)
            return join_bindings(incoming, _prune(star, needed), needed)
        pl = p_col.cpu().tolist()
        ol = o_col.cpu().tolist()
        per = []
        for pid, var in op.items:
            vals = [o for pp, o in zip(pl, ol) if pp == pid]
            if not vals:
                return Bindings.empty(self.device, vars_)
            per.append((var, vals))
        # cartesian across patterns (typically 1 value each); repeated
        # output vars constrain equality
        rows = [{}]
        for var, vals in per:
            nxt = []
            for r in rows:
                for v in vals:
                    if var in r:
                        if r[var] == v:
                            nxt.append(r)
                    else:
                        r2 = dict(r)
                        r2[var] = v
                        nxt.append(r2)
            rows = nxt
            if not rows:
                return Bindings.empty(self.device, vars_)
        uniq_vars = list(dict.fromkeys(vars_))
        # ONE H2D upload for all result columns (torch.tensor-from-list
        # costs ~17 µs per call on ROCm; this path runs per query)
        import numpy as np
        arr = np.array([[r[v] for v in uniq_vars] for r in rows],
                       dtype=np.int32)
        t = torch.from_numpy(arr).to(self.device)
        cols = {v: t[:, j].contiguous() for j, v in enumerate(uniq_vars)}
        star = Bindings(cols, len(rows), self.device)
        return join_bindings(incoming, _prune(star, needed), needed)

    # --------------------------------------------- fused chain-count (K1+K4)
    def _try_chain_count(self, op) -> Optional[int]:
        """COUNT(*) over a probe chain keyed ONLY off the seed scan fuses
        into one kernel: count = sum_rows prod_hops |matches(key(row))|.

        Shape requirements (bail -> None, generic path runs):
          - left-deep PBindJoin/PHashJoin tree of plain default-graph scans,
          - seed pattern: (?s  P  ?o) with P constant, ?s != ?o,
          - every hop: (?v  P_h  ?free) or (?free  P_h  ?v) with P_h
            constant, ?v in {seed ?s, seed ?o}, and ?free occurring nowhere
            else (no cross-hop constraints).
        """
        from ..storage.dataset import POS, PSO
        from .tensor_utils import pack2
        # regions are store-version-stable: cache them on the plan node.
        # Checked BEFORE the plan walk — the cached path is the per-query
        # hot loop and skips every isinstance traversal below.
        cache = getattr(op, "_chain_cache", None)
        if cache is not None and cache[0] == self.db.store.version:
            seed_key12, seed_z, hop_regions, native = cache[1:5]
            serve = getattr(op, "_chain_serve", None)
            if isinstance(serve, tuple):
                # C++ serving path: one pybind call = launches + pinned
                # readback (no torch dispatch, no graph-replay floor)
                return serve[0].serve_chain_count(serve[1])
            graph = getattr(op, "_chain_graph", None)
            if isinstance(graph, tuple):
                graph[0].replay()
                return int(graph[1].item())
            if native is not None:
                cnt = int(native.chain_count(cache[6], seed_z, *cache[5]))
                self._maybe_capture_chain_graph(op, native, cache[6],
                                                seed_z, cache[5])
                return cnt
            return self._chain_count_torch(seed_key12, seed_z, hop_regions)
        scans: List = []

        def flatten(x) -> bool:
            if isinstance(x, (PTableScan, PIndexScan)):
                scans.append(x)
                return x.graph is None
            if isinstance(x, (PBindJoin, PHashJoin)):
                return flatten(x.left) and flatten(x.right)
            return False

        if not flatten(op) or len(scans) < 2:
            return None
        if self.ctx.view.default_graphs != [DEFAULT_GRAPH] \
                or self.ctx.view.named_graphs not in (None, []):
            pass  # merged views still resolve through default_index below
        base = scans[0].pattern
        if not (isinstance(base.p, Constant) and isinstance(base.s, Variable)
                and isinstance(base.o, Variable)
                and base.s.name != base.o.name):
            return None
        var_counts: Dict[str, int] = {}
        for sc in scans:
            for v in sc.pattern.variables():
                var_counts[v] = var_counts.get(v, 0) + 1
        hops = []
        for sc in scans[1:]:
            pat = sc.pattern
            if not isinstance(pat.p, Constant):
                return None
            s_t, o_t = pat.s, pat.o
            if not (isinstance(s_t, Variable) and isinstance(o_t, Variable)):
                return None
            seed_vars = (base.s.name, base.o.name)
            if s_t.name in seed_vars and var_counts.get(o_t.name, 0) == 1:
                probe_pos, probe_var = 0, s_t.name
            elif o_t.name in seed_vars and var_counts.get(s_t.name, 0) == 1:
                probe_pos, probe_var = 2, o_t.name
            else:
                return None
            hops.append((pat.p.id, probe_pos,
                         0 if probe_var == base.s.name else 1))
        idx = self.ctx.default_index()
        if idx.n == 0:
            return 0
        seed_key12, seed_z, lo, hi = _pso_region(idx, base.p.id)
        if hi <= lo:
            return 0
        hop_regions = []
        for (pid, probe_pos, src) in hops:
            code = PSO if probe_pos == 0 else POS
            key12, _z = idx.orders[code]
            import torch as _t
            klo = pack2(_t.tensor([pid], dtype=_t.int32),
                        _t.tensor([0], dtype=_t.int32)).to(idx.device)
            khi = pack2(_t.tensor([pid], dtype=_t.int32),
                        _t.tensor([-1], dtype=_t.int32)).to(idx.device)
            rlo = int(_t.searchsorted(key12, klo, side="left").item())
            rhi = int(_t.searchsorted(key12, khi, side="right").item())
            hop_regions.append((key12[rlo:rhi], pid, src))
        from ..ops import native_for
        import torch as _t
        native = native_for(seed_key12)
        seed_key12 = seed_key12.contiguous()
        seed_z = seed_z.contiguous()
        hop_regions = [
            (r[0].contiguous(), r[1], r[2],
             *self._chain_hop_table(native, r[0], seed_key12.numel()))
            for r in hop_regions]
        # native kernels stream 32-bit columns: the region high words are
        # each hop's constant predicate and the seed key's low word is the
        # probe component — dropping the high halves halves the HBM bytes
        # and doubles the LDS window capacity (u32-order preserved)
        seed_b = (seed_key12 & 0xFFFFFFFF).to(_t.int32).contiguous()
        hop_args = ([(r[0] & 0xFFFFFFFF).to(_t.int32).contiguous()
                     if r[3].numel() == 0 else
                     _t.empty(0, dtype=_t.int32, device=seed_b.device)
                     for r in hop_regions],
                    [r[2] for r in hop_regions],
                    [r[3] for r in hop_regions],
                    [r[4] for r in hop_regions])
        op._chain_cache = (self.db.store.version, seed_key12, seed_z,
                           hop_regions, native, hop_args, seed_b)
        if hasattr(op, "_chain_graph"):
            del op._chain_graph  # stale capture from a previous store version
        if native is not None:
            return int(native.chain_count(seed_b, seed_z, *hop_args))
        return self._chain_count_torch(seed_key12, seed_z, hop_regions)

    def _maybe_capture_chain_graph(self, op, native, seed_b, seed_z,
                                   hop_args):
        """Record the tile_bounds + chain_count launch pair into a hipGraph
        (torch.cuda.CUDAGraph IS hipGraph on ROCm): the cached path becomes
        one graph replay + one 8-byte read-back.  Capture failures fall
        back permanently to plain launches."""
        if not seed_b.is_cuda or getattr(op, "_chain_graph", False) is None:
            return
        import torch as _t
        # preferred: C++ registry serving (direct launches beat a graph
        # replay at this kernel count — MI355X guide §graph-replay-floor)
        try:
            from ..ops import native_for
            native2 = native_for(seed_b)
            old = getattr(op, "_chain_serve", None)
            if isinstance(old, tuple):
                old[0].release_chain_serve(old[1])
            sid = native2.register_chain_serve(seed_b, seed_z, *hop_args)
            op._chain_serve = (native2, sid)
            return
        except Exception:
            _log.warning("chain-serve registration failed; falling back "
                         "to hipGraph capture", exc_info=True)
            op._chain_serve = None
        try:
            k = len(hop_args[0])
            tile = getattr(native, "chain_tile", lambda: 256)()
            n_tiles = (seed_b.numel() + tile - 1) // tile
            win = _t.empty(max(1, n_tiles * k * 2), dtype=_t.int64,
                           device=seed_b.device)
            total = _t.zeros(1, dtype=_t.int64, device=seed_b.device)
            side = _t.cuda.Stream()
            side.wait_stream(_t.cuda.current_stream())
            with _t.cuda.stream(side):   # warmup outside capture
                native.chain_count_into(seed_b, seed_z, *hop_args,
                                        win, total)
            _t.cuda.current_stream().wait_stream(side)
            g = _t.cuda.CUDAGraph()
            with _t.cuda.graph(g):
                native.chain_count_into(seed_b, seed_z, *hop_args,
                                        win, total)
            op._chain_graph = (g, total, win)
        except Exception:
            _log.warning("hipGraph capture failed; cached COUNT path will "
                         "use plain launches", exc_info=True)
            op._chain_graph = None  # permanent fallback marker

    # hop regions much smaller than the seed count pay log2(n) L2 lines per
    # seed in the fused kernel; collapse them ONCE (per store version, the
    # cache above) to an open-addressing (value -> count) table so the hop
    # is a single 8-byte load.  Only worthwhile when the table stays
    # L2-resident and the seeds dominate the build cost.
    _HOP_TABLE_MAX_ROWS = 1_000_000

    def _chain_hop_table(self, native, region, n_seeds):
        """Returns (table, dmin): dmin >= 0 marks a DIRECT-indexed dense
        table counts[v - dmin] (coalesced probes); dmin == -1 marks the
        hashed open-addressing forms; empty table = binary-search hop."""
        import os
        import torch as _t
        n = region.numel()
        cap = int(os.environ.get("KOLIBRIE_HOP_TABLE_MAX",
                                 self._HOP_TABLE_MAX_ROWS))
        if (native is None or not region.is_cuda or n == 0
                or n > cap or n * 4 > n_seeds):
            return _t.empty(0, dtype=_t.int64, device=region.device), -1
        vals, counts = _t.unique_consecutive(region & 0xFFFFFFFF,
                                             return_counts=True)
        vmin = int(vals.min().item()) if vals.numel() else 0
        vmax = int(vals.max().item()) if vals.numel() else 0
        span = vmax - vmin + 1
        if (vals.numel() and span <= (1 << 22)
                and span <= max(4 * vals.numel(), 1024)
                and os.environ.get("KOLIBRIE_DIRECT_HOP", "1") != "0"):
            # dense value space (e.g. a contiguous id block): direct
            # index beats hashing — consecutive seed values read
            # consecutive table slots, so the wave's loads coalesce
            tbl = _t.zeros(span, dtype=_t.int32, device=region.device)
            tbl[(vals - vmin).to(_t.long)] = counts.to(_t.int32)
            return tbl, vmin
        if vals.numel() and vmax < 0x1FFFFFF \
                and int(counts.max().item()) < 128:
            # packed (val<<7)|count u32 table: half the per-probe bytes
            # and table footprint (L2 residency beside the seed stream)
            return native.build_count_table32(
                ((vals << 7) | counts).to(_t.int32)), -1
        return native.build_count_table((vals << 32) | counts), -1

    def _chain_count_torch(self, seed_key12, seed_z, hop_regions) -> int:
        # torch fallback (CPU oracle): vectorized per-hop count product
        from .tensor_utils import pack2
        import torch as _t
        b_comp = (seed_key12 & 0xFFFFFFFF).to(_t.int32)
        z_comp = seed_z
        prod = _t.ones(seed_key12.numel(), dtype=_t.int64,
                       device=seed_key12.device)
        for (region, pid, src, *_tbl) in hop_regions:
            comp = b_comp if src == 0 else z_comp
            keys = pack2(_t.full_like(comp, pid), comp)
            lo_t = _t.searchsorted(region, keys, side="left")
            hi_t = _t.searchsorted(region, keys, side="right")
            prod *= (hi_t - lo_t)
        return int(prod.sum().item())

    # --------------------------------------------------------------- values --
    def _values_bindings(self, variables: List[str],
                         rows: List[List[Optional[int]]]) -> Bindings:
        dev = self.device
        n = len(rows)
        cols = {}
        any_undef = False
        for j, v in enumerate(variables):
            any_undef |= any(r[j] is None for r in rows)
            data = [(UNBOUND if r[j] is None else r[j]) for r in rows]
            cols[v] = torch.tensor(data, dtype=torch.int32, device=dev)
        return Bindings(cols, n, dev, maybe_unbound=any_undef)

    # ----------------------------------------------------------------- scan --
    def _exec_scan(self, pattern: TriplePattern, scope, incoming: Bindings,
                   needed=None, sort_hint=None) -> Bindings:
        if scope is None:
            idx = self.ctx.default_index()
            return self._scan_index(idx, pattern, incoming, extra=None,
                                    needed=needed, sort_hint=sort_hint)
        if scope[0] == "const":
            idx = self.db.store.graph_index(scope[1] & 0xFFFFFFFF)
            return self._scan_index(idx, pattern, incoming, extra=None,
                                    needed=needed, sort_hint=sort_hint)
        if scope[0] == "closure":
            idx = self._closure_index(scope[1], scope[2], scope[3])
            return self._scan_index(idx, pattern, incoming, extra=None,
                                    needed=needed, sort_hint=sort_hint)
        # GRAPH ?g — iterate named graphs, bind the graph variable
        gvar = scope[1]
        if needed is not None:
            needed = set(needed) | {gvar}
        parts = []
        for gid in self.ctx.named_graph_ids():
            idx = self.db.store.graph_index(gid)
            gid_i32 = gid - 0x1_0000_0000 if gid >= 0x8000_0000 else gid
            res = self._scan_index(idx, pattern, incoming, extra=(gvar, gid_i32),
                                   needed=needed)
            parts.append(res)
        if not parts:
            all_vars = pattern.variables() + [gvar]
            return Bindings.empty(self.device, all_vars)
        return Bindings.concat(parts, self.device)

    def _closure_index(self, pid: int, reflexive: bool,
                       gid: Optional[int] = None) -> GraphIndex:
        """Transitive closure of predicate `pid` as an auxiliary
        GraphIndex, materialized by log-doubling joins on device and
        cached per store version (p+ / p* property paths).  `*` adds the
        zero-length pairs over the predicate's node set."""
        cache = getattr(self.db, "_closures", None)
        if cache is None:
            cache = self.db._closures = {}
        key = (pid, reflexive, gid)
        hit = cache.get(key)
        if hit is not None and hit[0] == self.db.store.version:
            return hit[1]
        from .scan import scan_unit
        from .tensor_utils import pack2, unique_rows
        pid_i32 = pid - 0x1_0000_0000 if pid >= 0x8000_0000 else pid
        base = self.ctx.default_index() if gid is None \
            else self.db.store.graph_index(gid)
        s, _p, o = scan_unit(base, {1: pid_i32})
        dev = self.device
        cs, co = s, o
        # squaring: C_{k+1} = C_k ∪ (C_k ∘ C_k) reaches length 2^k in k
        # rounds (log-depth for chains; joins are the same sorted
        # searchsorted machinery as the fixpoint kernels drive)
        for _ in range(64):
            cs, co = unique_rows([cs, co])
            # join C.o == C.s
            left_o = co.to(torch.int64) & 0xFFFFFFFF
            right_s = (cs.to(torch.int64) & 0xFFFFFFFF)
            order = torch.argsort(right_s)
            rs_sorted = right_s[order]
            lo = torch.searchsorted(rs_sorted, left_o, side="left")
            hi = torch.searchsorted(rs_sorted, left_o, side="right")
            cnt = hi - lo
            li = torch.repeat_interleave(
                torch.arange(cs.numel(), dtype=torch.long, device=dev), cnt)
            starts = torch.cumsum(cnt, 0) - cnt
            pos = torch.arange(int(cnt.sum().item()), dtype=torch.long,
                               device=dev) - starts[li]
            ri = order[lo[li] + pos]
            new_s, new_o = cs[li], co[ri]
            all_s = torch.cat([cs, new_s])
            all_o = torch.cat([co, new_o])
            u_s, u_o = unique_rows([all_s, all_o])
            if u_s.numel() == cs.numel():
                cs, co = u_s, u_o
                break
            cs, co = u_s, u_o
        if reflexive:
            nodes = torch.unique(torch.cat([s, o]))
            cs = torch.cat([cs, nodes])
            co = torch.cat([co, nodes])
        pcol = torch.full((cs.numel(),), pid_i32, dtype=torch.int32,
                          device=dev)
        gi = GraphIndex.from_columns(cs, pcol, co, device=str(dev))
        cache[key] = (self.db.store.version, gi)
        return gi

    def _scan_index(self, idx: GraphIndex, pattern: TriplePattern,
                    incoming: Bindings, extra: Optional[Tuple[str, int]],
                    needed=None, sort_hint=None) -> Bindings:
        """Scan one index, extending each incoming row (engine.rs:1018)."""
        dev = self.device
        # graph-variable consistency: if ?g already bound, pre-filter rows
        inc = incoming
        if extra is not None and inc.has(extra[0]):
            m = (inc.col(extra[0]) == extra[1]) | (inc.col(extra[0]) == UNBOUND)
            inc = inc.select(m)
            if inc.is_empty():
                return inc
        consts: Dict[int, int] = {}
        var_pos: Dict[int, str] = {}
        qt_pos: Dict[int, QuotedTriplePattern] = {}
        for i, t in enumerate(pattern.terms()):
            if isinstance(t, Constant):
                consts[i] = t.id
            elif isinstance(t, Variable):
                var_pos[i] = t.name
            else:
                qt_pos[i] = t
        is_unit = inc.n == 1 and not inc.cols
        probe_vars = {i: v for i, v in var_pos.items()
                      if not is_unit and inc.has(v)}
        exec_stats.bump("SCAN_PROBES", max(1, inc.n))

        # lazy materialization: which (s,p,o) positions must the scan emit?
        from collections import Counter
        name_counts = Counter(var_pos.values())
        scan_need = set(qt_pos.keys())
        for pos_i, name in var_pos.items():
            if name_counts[name] > 1 or needed is None or name in needed:
                scan_need.add(pos_i)

        if not probe_vars:
            if (needed is not None and not scan_need and not qt_pos
                    and extra is None and is_unit):
                # COUNT(*) over a bare scan: range size only
                from .scan import scan_unit_count
                cnt = scan_unit_count(idx, consts)
                if cnt is not None:
                    exec_stats.bump("ROWS_EMITTED", cnt)
                    return Bindings({}, cnt, dev)
            s, p, o = scan_unit(idx, consts, sort_hint=sort_hint,
                                need=scan_need)
            n_sc = next((c.numel() for c in (s, p, o) if c is not None), -1)
            if n_sc < 0:
                # no column materialized (all pattern vars pruned): the row
                # count still matters for multiplicity / graph binding
                from .scan import scan_unit_count
                n_sc = scan_unit_count(idx, consts)
                if n_sc is None:
                    s, p, o = scan_unit(idx, consts, sort_hint=sort_hint)
                    n_sc = next((c.numel() for c in (s, p, o)
                                 if c is not None), 0)
            exec_stats.bump("QUADS_EXAMINED", n_sc)
            cand = self._build_candidate(s, p, o, var_pos, qt_pos, None,
                                         needed, n_sc)
            if is_unit:
                exec_stats.bump("ROWS_EMITTED", cand.n)
                return self._add_extra(cand, extra)
            out = join_bindings(_prune(inc, needed), cand, needed)
            exec_stats.bump("ROWS_EMITTED", out.n)
            return self._add_extra(out, extra)

        # group rows by which probe vars are actually bound (UNDEF handling);
        # untainted inputs (no UNBOUND anywhere) skip the grouping scans
        if not inc.maybe_unbound:
            if needed is not None and not needed and not scan_need \
                    and not qt_pos and extra is None:
                from .scan import scan_probe_count
                cnt = scan_probe_count(
                    idx, consts, {i: inc.col(v)
                                  for i, v in probe_vars.items()})
                if cnt is not None:
                    exec_stats.bump("ROWS_EMITTED", cnt)
                    return Bindings({}, cnt, dev)
            probes = {i: inc.col(v) for i, v in probe_vars.items()}
            if needed is None:
                carry_src = inc
            else:
                pattern_names = set(var_pos.values())
                carry_src = _prune(inc, set(needed) - pattern_names)
            from .scan import scan_probe_carry
            li, s, p, o, carried = scan_probe_carry(
                idx, consts, probes, scan_need, dict(carry_src.cols))
            exec_stats.bump("QUADS_EXAMINED", li.numel())
            base = Bindings(carried, li.numel(), dev)
            out = self._build_candidate(s, p, o, var_pos, qt_pos, base,
                                        needed, li.numel())
            exec_stats.bump("ROWS_EMITTED", out.n)
            return self._add_extra(out, extra)
        masks = {i: (inc.col(v) != UNBOUND) for i, v in probe_vars.items()}
        sig = torch.zeros(inc.n, dtype=torch.int64, device=dev)
        for k, i in enumerate(sorted(masks)):
            sig |= masks[i].to(torch.int64) << k
        parts: List[Bindings] = []
        for sval in torch.unique(sig).tolist():
            rows_mask = sig == sval
            sub = inc.select(rows_mask)
            active = {i: v for k, (i, v) in
                      enumerate(sorted(probe_vars.items()))
                      if (sval >> k) & 1}
            if not active:
                s, p, o = scan_unit(idx, consts)
                cand = self._build_candidate(s, p, o, var_pos, qt_pos, None, needed)
                # cartesian: these rows have the vars unbound -> binder
                sub2 = sub.drop_cols(list(probe_vars.values()))
                parts.append(join_bindings(sub2, cand, needed))
                continue
            probes = {i: sub.col(v) for i, v in active.items()}
            if (needed is not None and not needed and not scan_need
                    and not qt_pos and extra is None):
                # COUNT(*) pushdown: no columns needed — count matches only
                from .scan import scan_probe_count
                cnt = scan_probe_count(idx, consts, probes)
                if cnt is not None:
                    parts.append(Bindings({}, cnt, dev))
                    continue
            # incoming columns the scan does not itself supply are carried
            # through the fused emit (projection pushdown + fused gather)
            if needed is None:
                carry_src = sub
            else:
                pattern_names = set(var_pos.values())
                carry_src = _prune(sub, set(needed) - pattern_names)
            from .scan import scan_probe_carry
            li, s, p, o, carried = scan_probe_carry(
                idx, consts, probes, scan_need, dict(carry_src.cols))
            exec_stats.bump("QUADS_EXAMINED", li.numel())
            base = Bindings(carried, li.numel(), dev)
            cand = self._build_candidate(s, p, o, var_pos, qt_pos, base,
                                         needed, li.numel())
            parts.append(cand)
        out = Bindings.concat(parts, dev) if len(parts) != 1 else parts[0]
        exec_stats.bump("ROWS_EMITTED", out.n)
        return self._add_extra(out, extra)

    def _add_extra(self, b: Bindings, extra: Optional[Tuple[str, int]]
                   ) -> Bindings:
        if extra is None:
            return b
        gvar, gid = extra
        if b.has(gvar):
            mask = (b.col(gvar) == gid) | (b.col(gvar) == UNBOUND)
            b = b.select(mask)
            return b.with_col(gvar, torch.full((b.n,), gid, dtype=torch.int32,
                                               device=b.device))
        return b.with_col(gvar, torch.full((b.n,), gid, dtype=torch.int32,
                                           device=b.device))

    def _build_candidate(self, s, p, o, var_pos: Dict[int, str],
                         qt_pos: Dict[int, QuotedTriplePattern],
                         base: Optional[Bindings], needed=None,
                         n: int = None) -> Bindings:
        """Assemble bindings from scanned triple columns: bind variables,
        enforce repeated-variable equality (engine.rs:1223-1239) and match
        quoted-triple sub-patterns (engine.rs:1253).  Columns may be None
        when projection pushdown proved them unneeded."""
        dev = self.device
        cols_all = (s, p, o)
        if n is None:
            n = next((c.numel() for c in cols_all if c is not None), 0)
        out_cols: Dict[str, torch.Tensor] = {} if base is None else dict(base.cols)
        base_vars = set() if base is None else set(base.cols.keys())
        mask = torch.ones(n, dtype=torch.bool, device=dev)
        seen_here: set = set()
        for i, name in var_pos.items():
            col = cols_all[i]
            if col is None:
                continue  # pruned: not needed downstream
            if name in base_vars and name not in seen_here:
                # probed position: equality enforced by scan_probe already
                out_cols[name] = col
                seen_here.add(name)
            elif name in out_cols:
                # repeated variable within the pattern -> equality constraint
                mask &= out_cols[name] == col
            else:
                out_cols[name] = col
                seen_here.add(name)
        for i, qtp in qt_pos.items():
            m, qt_cols = self._match_quoted(cols_all[i], qtp, out_cols)
            mask &= m
            out_cols.update(qt_cols)
        res = Bindings(out_cols, n, dev)
        if not bool(mask.all()):
            res = res.select(mask)
        return _prune(res, needed)

    def _match_quoted(self, ids: torch.Tensor, qtp: QuotedTriplePattern,
                      bound_cols: Dict[str, torch.Tensor]
                      ) -> Tuple[torch.Tensor, Dict[str, torch.Tensor]]:
        """Match a quoted-triple pattern position against a column of ids."""
        dev = self.device
        qs, qp, qo = self.db.quoted_columns()
        n = ids.numel()
        is_q = (ids < 0) & (ids != UNBOUND)
        qidx = (ids.to(torch.int64) & 0x7FFFFFFF).clamp(max=max(0, qs.numel() - 1))
        mask = is_q.clone()
        new_cols: Dict[str, torch.Tensor] = {}
        for comp_term, comp_col in zip(
            (qtp.s, qtp.p, qtp.o),
            (qs[qidx] if qs.numel() else torch.zeros(n, dtype=torch.int32, device=dev),
             qp[qidx] if qp.numel() else torch.zeros(n, dtype=torch.int32, device=dev),
             qo[qidx] if qo.numel() else torch.zeros(n, dtype=torch.int32, device=dev)),
        ):
            if isinstance(comp_term, Constant):
                mask &= comp_col == comp_term.id
            elif isinstance(comp_term, Variable):
                nm = comp_term.name
                if nm in bound_cols:
                    mask &= bound_cols[nm] == comp_col
                elif nm in new_cols:
                    mask &= new_cols[nm] == comp_col
                else:
                    new_cols[nm] = comp_col
            else:  # nested quoted pattern
                m2, c2 = self._match_quoted(comp_col, comp_term,
                                            {**bound_cols, **new_cols})
                mask &= m2
                new_cols.update(c2)
        return mask, new_cols

    # ------------------------------------------------------------- subquery --
    def _exec_subquery(self, sub) -> Bindings:
        """Run a compiled subquery plan and finalize its modifiers at ID
        level (ref engine.rs:785-905)."""
        inner = self.execute(sub.physical, Bindings.unit(self.device))
        from .finalize import finalize_select_bindings
        return finalize_select_bindings(sub.select, inner, self.db)


# ------------------------------------------------------------------- joins --
def _prune(b: Bindings, needed) -> Bindings:
    """Drop columns outside the needed set (projection pushdown)."""
    if needed is None:
        return b
    cols = {v: c for v, c in b.cols.items() if v in needed}
    if len(cols) == len(b.cols):
        return b
    return Bindings(cols, b.n, b.device)


def join_bindings(left: Bindings, right: Bindings, needed=None) -> Bindings:
    """SPARQL-compatible natural join (multiset).

    Keyed rows (all shared vars bound) go through the sort-merge equi-join
    (K2's torch mirror); rows with unbound shared vars fall back to the
    compatibility nested loop (ref engine.rs:1389-1395 unkeyed fallback).
    """
    dev = left.device
    if left.n == 1 and not left.cols:
        return _prune(right, needed)
    if right.n == 1 and not right.cols:
        return _prune(left, needed)
    if left.is_empty() or right.is_empty():
        vars_ = list(dict.fromkeys(left.variables + right.variables))
        if needed is not None:
            # match the non-empty path's pruning: a divergent column set on
            # an empty intermediate would desynchronize distributed
            # exchanges (per-column collectives) across ranks
            vars_ = [v for v in vars_ if v in needed]
        return Bindings.empty(dev, vars_)
    shared = [v for v in left.variables if v in right.cols]
    if not shared:
        # cartesian product
        li = torch.arange(left.n, dtype=torch.long, device=dev).repeat_interleave(right.n)
        ri = torch.arange(right.n, dtype=torch.long, device=dev).repeat(left.n)
        return _merge_pairs(left, right, li, ri, shared, needed)
    if (not left.maybe_unbound and not right.maybe_unbound
            and left.n * right.n <= 262_144):
        # tiny-tables fast path: selective point queries produce 1-row
        # intermediates where the merge/hash machinery is pure per-op
        # overhead — one broadcast compare replaces ~20 tensor ops
        mask = left.col(shared[0]).unsqueeze(1) == \
            right.col(shared[0]).unsqueeze(0)
        for v in shared[1:]:
            mask = mask & (left.col(v).unsqueeze(1) ==
                           right.col(v).unsqueeze(0))
        li, ri = mask.nonzero(as_tuple=True)
        if needed is not None and len(needed) == 0:
            return Bindings({}, int(li.numel()), dev)
        return _merge_pairs(left, right, li, ri, shared, needed)
    parts: List[Bindings] = []
    if not left.maybe_unbound and not right.maybe_unbound:
        l_keyed, l_unkeyed = left, Bindings.empty(dev, left.variables)
        r_keyed, r_unkeyed = right, Bindings.empty(dev, right.variables)
    else:
        lb = torch.ones(left.n, dtype=torch.bool, device=dev)
        for v in shared:
            lb &= left.col(v) != UNBOUND
        rb = torch.ones(right.n, dtype=torch.bool, device=dev)
        for v in shared:
            rb &= right.col(v) != UNBOUND
        l_keyed, l_unkeyed = left.select(lb), left.select(~lb)
        r_keyed, r_unkeyed = right.select(rb), right.select(~rb)
    count_only = needed is not None and len(needed) == 0
    if count_only and l_keyed.n and r_keyed.n:
        # COUNT(*) pushdown: match counts without emitting pairs
        n_match = _join_count(l_keyed, r_keyed, shared)
        parts.append(Bindings({}, n_match, dev))
    elif l_keyed.n and r_keyed.n:
        from ..ops import native_for
        native = native_for(l_keyed.col(shared[0])) if len(shared) <= 4 else None
        merged = None
        if len(shared) == 1:
            # sortedness-aware merge join: a PSO-slice scan output is already
            # sorted by subject — searchsorted beats a hash build (no table,
            # sequential probe locality).  Detecting monotonicity is one
            # cheap pass.
            lk = l_keyed.col(shared[0])
            rk = r_keyed.col(shared[0])
            if _is_sorted(rk):
                lo = torch.searchsorted(rk, lk, side="left")
                hi = torch.searchsorted(rk, lk, side="right")
                li, ri = _expand_ranges(lo, hi)
                merged = _merge_pairs(l_keyed, r_keyed, li, ri, shared, needed)
            elif _is_sorted(lk):
                lo = torch.searchsorted(lk, rk, side="left")
                hi = torch.searchsorted(lk, rk, side="right")
                ri, li = _expand_ranges(lo, hi)
                merged = _merge_pairs(l_keyed, r_keyed, li, ri, shared, needed)
        if merged is not None:
            parts.append(merged)
        elif native is not None:
            # K2: chained open-addressing hash join on device.
            # Build on the smaller side, probe with the larger.
            if l_keyed.n >= r_keyed.n:
                li, ri = native.hash_join(
                    [l_keyed.col(v).contiguous() for v in shared],
                    [r_keyed.col(v).contiguous() for v in shared],
                )
            else:
                ri, li = native.hash_join(
                    [r_keyed.col(v).contiguous() for v in shared],
                    [l_keyed.col(v).contiguous() for v in shared],
                )
        else:
            key_cols = [torch.cat([l_keyed.col(v), r_keyed.col(v)]) for v in shared]
            gid, _ = group_index(key_cols)
            lkey, rkey = gid[:l_keyed.n], gid[l_keyed.n:]
            li, ri = merge_join_indices(lkey, rkey)
        if merged is None:
            parts.append(_merge_pairs(l_keyed, r_keyed, li, ri, shared, needed))
    if l_unkeyed.n and right.n:
        parts.append(_compat_nlj(l_unkeyed, right, shared, needed))
    if l_keyed.n and r_unkeyed.n:
        parts.append(_compat_nlj(l_keyed, r_unkeyed, shared, needed))
    if not parts:
        vars_ = list(dict.fromkeys(left.variables + right.variables))
        return Bindings.empty(dev, vars_)
    return Bindings.concat(parts, dev) if len(parts) > 1 else parts[0]


def _join_count(l_keyed: Bindings, r_keyed: Bindings,
                shared: Sequence[str]) -> int:
    dev = l_keyed.device
    if len(shared) == 1:
        lk = l_keyed.col(shared[0])
        rk = r_keyed.col(shared[0])
        if _is_sorted(rk):
            lo = torch.searchsorted(rk, lk, side="left")
            hi = torch.searchsorted(rk, lk, side="right")
            return int((hi - lo).sum().item())
        if _is_sorted(lk):
            lo = torch.searchsorted(lk, rk, side="left")
            hi = torch.searchsorted(lk, rk, side="right")
            return int((hi - lo).sum().item())
    from ..ops import native_for
    native = native_for(l_keyed.col(shared[0])) if len(shared) <= 4 else None
    if native is not None:
        cnt = native.hash_join_counts(
            [l_keyed.col(v).contiguous() for v in shared],
            [r_keyed.col(v).contiguous() for v in shared])
        return int(cnt.to(torch.int64).sum().item())
    key_cols = [torch.cat([l_keyed.col(v), r_keyed.col(v)]) for v in shared]
    gid, _ = group_index(key_cols)
    li, _ri = merge_join_indices(gid[:l_keyed.n], gid[l_keyed.n:])
    return int(li.numel())


def _is_sorted(col: torch.Tensor) -> bool:
    if col.numel() <= 1:
        return True
    return bool((col[1:] >= col[:-1]).all())


def _expand_ranges(lo: torch.Tensor, hi: torch.Tensor):
    """Expand per-row [lo, hi) ranges into (row_idx, range_pos) pairs."""
    dev = lo.device
    cnt = hi - lo
    total = int(cnt.sum().item())
    n = lo.numel()
    if total == 0:
        e = torch.empty(0, dtype=torch.long, device=dev)
        return e, e.clone()
    li = torch.repeat_interleave(torch.arange(n, dtype=torch.long, device=dev), cnt)
    starts = torch.cumsum(cnt, 0) - cnt
    pos = torch.arange(total, dtype=torch.long, device=dev) - starts[li]
    return li, lo[li] + pos


def _merge_pairs(left: Bindings, right: Bindings, li, ri,
                 shared: Sequence[str], needed=None) -> Bindings:
    dev = left.device
    cols: Dict[str, torch.Tensor] = {}
    for v, c in left.cols.items():
        if needed is not None and v not in needed:
            continue
        cols[v] = c[li]
    for v, c in right.cols.items():
        if needed is not None and v not in needed:
            continue
        if v in cols:
            if v in shared:
                # take bound value (left may be UNBOUND in compat path)
                lvals = cols[v]
                rvals = c[ri]
                cols[v] = torch.where(lvals != UNBOUND, lvals, rvals)
        elif v in shared and left.has(v):
            lvals = left.col(v)[li]
            rvals = c[ri]
            cols[v] = torch.where(lvals != UNBOUND, lvals, rvals)
        else:
            cols[v] = c[ri]
    return Bindings(cols, li.numel(), dev)


def _compat_nlj(left: Bindings, right: Bindings, shared: Sequence[str],
                needed=None) -> Bindings:
    """Cartesian + compatibility mask (UNBOUND matches anything)."""
    dev = left.device
    li = torch.arange(left.n, dtype=torch.long, device=dev).repeat_interleave(right.n)
    ri = torch.arange(right.n, dtype=torch.long, device=dev).repeat(left.n)
    mask = torch.ones(li.numel(), dtype=torch.bool, device=dev)
    for v in shared:
        lv = left.col(v)[li]
        rv = right.col(v)[ri]
        mask &= (lv == rv) | (lv == UNBOUND) | (rv == UNBOUND)
    li, ri = li[mask], ri[mask]
    return _merge_pairs(left, right, li, ri, shared, needed)


_LOJ_ROWVAR = "\x00loj_row"  # internal row-id column (never a user var)


def left_outer_join(left: Bindings, right: Bindings,
                    needed=None) -> Bindings:
    """OPTIONAL: every left row survives; matched rows extend with right
    columns, unmatched rows pad the right-only columns UNBOUND (SPARQL
    left outer join; engine extension beyond the reference subset).

    Matched-ness is tracked by a hidden per-row id column carried through
    the inner join, so a left row is padded iff it produced NO joined row —
    correct even when the shared-variable set is empty (cartesian: every
    left row matches whenever right is non-empty) or left shared vars are
    UNBOUND (compat semantics), where MINUS-style anti_join would
    disagree."""
    if right.is_empty():
        return left
    dev = left.device
    rowid = torch.arange(left.n, dtype=torch.int32, device=dev)
    inner = join_bindings(left.with_col(_LOJ_ROWVAR, rowid), right, None)
    if inner.is_empty():
        return left
    matched = torch.zeros(left.n, dtype=torch.bool, device=dev)
    matched[inner.col(_LOJ_ROWVAR).long()] = True
    inner = inner.project([v for v in inner.variables if v != _LOJ_ROWVAR])
    if needed is not None:
        inner = _prune(inner, needed)
    if bool(matched.all().item()):
        return inner
    unmatched = left.select(~matched)
    if needed is not None:
        unmatched = _prune(unmatched, needed)
    return Bindings.concat([inner, unmatched], dev)


def anti_join(left: Bindings, right: Bindings) -> Bindings:
    """Keep left rows with no compatible right row sharing >=1 bound var."""
    dev = left.device
    shared = [v for v in left.variables if v in right.cols]
    if not shared or right.is_empty():
        return left
    if (not left.maybe_unbound and not right.maybe_unbound
            and left.n * right.n <= 262_144):
        # tiny-tables fast path (same shape as join_bindings'): one
        # broadcast compare instead of unique+membership machinery
        # (262k boolean pairs is noise on-device; the former 4096 cap
        # pushed 100x100 MINUS shapes onto the multi-op membership path)
        hit = left.col(shared[0]).unsqueeze(1) == \
            right.col(shared[0]).unsqueeze(0)
        for v in shared[1:]:
            hit = hit & (left.col(v).unsqueeze(1) ==
                         right.col(v).unsqueeze(0))
        return left.select(~hit.any(dim=1))
    keep = torch.ones(left.n, dtype=torch.bool, device=dev)
    # vectorize by left boundness signature; fall back to NLJ for
    # right-side unbound rows
    rb = torch.ones(right.n, dtype=torch.bool, device=dev)
    for v in shared:
        rb &= right.col(v) != UNBOUND
    r_keyed = right.select(rb)
    if r_keyed.n:
        from .tensor_utils import membership_mask, unique_rows
        if not left.maybe_unbound:
            # common case: every left shared var bound -> one membership
            rcols = unique_rows([r_keyed.col(v) for v in shared])
            hit = membership_mask([left.col(v) for v in shared], rcols)
            keep &= ~hit
        else:
            # group left rows by which shared vars they bind: a row with a
            # partially-unbound key is still removed when a keyed right
            # row agrees on its BOUND shared vars (>=1 required — SPARQL
            # dom-intersection rule); the membership key is restricted to
            # that signature's bound vars
            sig = torch.zeros(left.n, dtype=torch.long, device=dev)
            for i, v in enumerate(shared):
                sig |= (left.col(v) != UNBOUND).long() << i
            for s in torch.unique(sig).tolist():
                if s == 0:
                    continue  # no bound shared var -> spec removes nothing
                vars_b = [v for i, v in enumerate(shared) if s >> i & 1]
                rows = (sig == s).nonzero(as_tuple=True)[0]
                rcols = unique_rows([r_keyed.col(v) for v in vars_b])
                hit = membership_mask(
                    [left.col(v)[rows] for v in vars_b], rcols)
                keep[rows[hit]] = False
    r_unkeyed = right.select(~rb)
    if r_unkeyed.n:
        for j in range(r_unkeyed.n):
            m = torch.ones(left.n, dtype=torch.bool, device=dev)
            any_shared = torch.zeros(left.n, dtype=torch.bool, device=dev)
            for v in shared:
                rv = int(r_unkeyed.col(v)[j].item())
                lv = left.col(v)
                if rv == UNBOUND:
                    continue
                m &= (lv == rv) | (lv == UNBOUND)
                any_shared |= lv != UNBOUND
            keep &= ~(m & any_shared)
    return left.select(keep)


def _pso_region(idx: GraphIndex, pid: int):
    """(key12 slice, z slice, lo, hi) of one predicate's PSO region."""
    from ..storage.dataset import PSO
    from .tensor_utils import pack2
    key12, z = idx.orders[PSO]
    klo = pack2(torch.tensor([pid], dtype=torch.int32),
                torch.tensor([0], dtype=torch.int32)).to(idx.device)
    khi = pack2(torch.tensor([pid], dtype=torch.int32),
                torch.tensor([-1], dtype=torch.int32)).to(idx.device)
    lo = int(torch.searchsorted(key12, klo, side="left").item())
    hi = int(torch.searchsorted(key12, khi, side="right").item())
    return key12[lo:hi], z[lo:hi], lo, hi
