"""Distributed SPARQL execution over subject-partitioned shards.

BASELINE config 3: 100M triples hash-partitioned across 8x MI355X, with the
join shuffle as an RCCL all-to-all over xGMI.  Round-2 design (VERDICT r1
item 1): the shuffle is PLANNER-DRIVEN — `distplan.distribute_plan`
rewrites the single-GPU Volcano plan, inserting PExchange ops wherever a
join key is not the partition key, and the finalize stage merges
partitioned results with collectives:

  COUNT(*)            -> local count + all-reduce
  GROUP BY aggregates -> hash-exchange rows on the group key, complete
                         local aggregation per rank, all-gather of the
                         decoded group rows
  plain SELECT        -> all-gather of the binding columns, identical
                         finalize everywhere

Invariants the loader enforces (and tests verify):
  - quad (g,s,p,o) lives on rank (s & 0xFFFFFFFF) % world;
  - the dictionary is REPLICATED: every rank interns the same strings in
    the same order (load through `add_triples_partitioned`, which encodes
    every triple's terms but stores only local rows), so int32 IDs are
    globally consistent and can cross ranks raw;
  - stats are all-reduced so every rank plans the identical plan.
"""
from __future__ import annotations

from dataclasses import replace
from typing import List, Optional

import torch

from ..engine.bindings import Bindings
from ..engine.executor import DatasetView, ExecutionContext, ExecutionEngine
from ..engine.finalize import decode_rows, finalize_select_bindings
from ..storage.database import SparqlDatabase
from . import dist as D
from .distplan import REPLICATED, distribute_plan


def allreduce_stats(st, device):
    """Merge per-rank DatabaseStats into identical global stats.

    Counts sum exactly.  Per-predicate distinct SUBJECTS sum exactly too
    (a subject's rows all live on one rank); distinct objects sum to an
    upper bound (the same object can appear on several ranks) — fine for
    cardinality estimation, noted here for honesty."""
    if not D.is_dist():
        return st
    import torch.distributed as dist
    world = dist.get_world_size()
    parts = [None] * world
    dist.all_gather_object(parts, st)
    from ..plan.stats import DatabaseStats
    out = DatabaseStats()
    for p in parts:
        out.total += p.total
        out.quoted_count = max(out.quoted_count, p.quoted_count)
        for d_out, d_in in ((out.pred_count, p.pred_count),
                            (out.pred_distinct_subj, p.pred_distinct_subj),
                            (out.pred_distinct_obj, p.pred_distinct_obj),
                            (out.graph_counts, p.graph_counts)):
            for k, v in d_in.items():
                d_out[k] = d_out.get(k, 0) + v
    out.distinct_subjects = sum(p.distinct_subjects for p in parts)
    out.distinct_objects = min(sum(p.distinct_objects for p in parts),
                               max(p.distinct_objects for p in parts) * world)
    return out


class DistributedDatabase:
    """A rank-local shard plus the collective query driver."""

    def __init__(self, rank: int, world: int, device):
        self.rank = rank
        self.world = world
        self.device = torch.device(device)
        self.db = SparqlDatabase(device=str(device))
        self._global_stats = None
        # predicate IDs whose relations the loader replicated on every
        # rank (load-time broadcast-table layout); the planner treats
        # their scans as part=REPLICATED
        self.replicated_preds = set()

    def declare_replicated(self, pred_iris):
        """Record that the caller loaded these predicates' triples on
        EVERY rank (small build-side relations)."""
        for iri in pred_iris:
            self.replicated_preds.add(
                self.db.dictionary.encode(iri) & 0xFFFFFFFF)
        self._global_stats = None

    # ------------------------------------------------------------- loading --
    def load_shard_columns(self, s, p, o):
        """Columns already partitioned by the caller (subject-hash)."""
        self.db.load_columns(s, p, o)
        self._global_stats = None

    def load_columns_partitioned(self, s: torch.Tensor, p: torch.Tensor,
                                 o: torch.Tensor):
        """Full column set on every rank: keep the local subject slice."""
        if self.world > 1:
            mine = (s.to(torch.int64) & 0xFFFFFFFF) % self.world == self.rank
            s, p, o = s[mine], p[mine], o[mine]
        self.db.load_columns(s, p, o)
        self._global_stats = None

    def add_triples_partitioned(self, triples):
        """Replicated-dictionary string ingest: every rank encodes EVERY
        triple (keeping IDs globally consistent) but stores only rows whose
        subject hashes here."""
        from ..storage.dataset import DEFAULT_GRAPH
        db = self.db
        for s, p, o in triples:
            sid = db.encode_term_star(s)
            pid = db.encode_term_star(p)
            oid = db.encode_term_star(o)
            if (sid & 0xFFFFFFFF) % self.world == self.rank:
                db.store.insert_quad(DEFAULT_GRAPH, sid, pid, oid)
        db.store.commit_all()
        self._global_stats = None

    # -------------------------------------------------------------- stats --
    def global_stats(self):
        if self._global_stats is None:
            local = self.db.get_or_build_stats()
            g = allreduce_stats(local, self.device)
            # replicated relations were summed world x by the allreduce;
            # every rank holds the identical full copy, so the local
            # counts ARE the global ones
            for pid in self.replicated_preds:
                for d_g, d_l in ((g.pred_count, local.pred_count),
                                 (g.pred_distinct_subj,
                                  local.pred_distinct_subj),
                                 (g.pred_distinct_obj,
                                  local.pred_distinct_obj)):
                    if pid in d_l:
                        d_g[pid] = d_l[pid]
            self._global_stats = g
        return self._global_stats

    # -------------------------------------------------------------- query --
    def prepare(self, sparql: str):
        """Parse + plan + distribute (deterministic across ranks)."""
        from ..parsing.sparql import parse_combined_query
        from ..plan.lower import build_logical_plan
        from ..plan.optimizer import Streamertail, annotate_needed
        from ..engine.query import _top_needed
        db = self.db
        cq = parse_combined_query(sparql)
        prefixes = dict(db.prefixes)
        prefixes.update(cq.prefixes)
        sel = cq.select
        if sel is None:
            raise ValueError("DistributedDatabase.query supports SELECT/ASK")
        stats = self.global_stats()
        logical = build_logical_plan(sel.where, db, prefixes)
        physical = Streamertail(stats).find_best_plan(logical)
        physical, part = distribute_plan(physical, stats, self.world,
                                         self.replicated_preds)
        annotate_needed(physical, _top_needed(sel))
        return sel, physical, part

    def query(self, sparql: str) -> List[List[str]]:
        """Distributed SELECT/ASK: identical decoded rows on every rank."""
        sel, physical, part = self.prepare(sparql)
        return self.execute_prepared(sel, physical, part)

    def execute_prepared(self, sel, physical, part) -> List[List[str]]:
        eng = getattr(self, "_engine_cache", None)
        if eng is None:
            ctx = ExecutionContext(self.db, DatasetView(),
                                   world=self.world, rank=self.rank)
            eng = self._engine_cache = ExecutionEngine(ctx)
        rows = eng.execute(physical, Bindings.unit(self.device))
        return self._finalize(sel, rows, part)

    # ----------------------------------------------------------- finalize --
    def _finalize(self, sel, rows: Bindings, part) -> List[List[str]]:
        db = self.db
        if self.world <= 1 or part == REPLICATED:
            if getattr(sel, "ask", False):
                return [["true" if rows.n > 0 else "false"]]
            final = finalize_select_bindings(sel, rows, db)
            return decode_rows(sel, final, db)
        if getattr(sel, "ask", False):
            total = D.allreduce_sum_scalar(rows.n, self.device)
            return [["true" if total > 0 else "false"]]
        if self._count_star_only(sel):
            total = D.allreduce_sum_scalar(rows.n, self.device)
            if sel.offset not in (None, 0) or sel.limit == 0:
                return []
            return [[str(total)] * len(sel.variables)]
        has_agg = any(p.aggregate for p in sel.variables)
        if sel.group_by and has_agg:
            return self._finalize_grouped(sel, rows)
        # global aggregates / plain rows: gather the needed columns so the
        # standard finalize runs identically on every rank
        rows = self._pre_shrink(sel, rows)
        names = rows.variables
        if names:
            cols = D.all_gather_rows([rows.col(v) for v in names])
            rows = Bindings(dict(zip(names, cols)),
                            cols[0].numel() if cols else 0, self.device,
                            maybe_unbound=rows.maybe_unbound)
        else:
            n = D.allreduce_sum_scalar(rows.n, self.device)
            rows = Bindings({}, n, self.device)
        final = finalize_select_bindings(sel, rows, db)
        return decode_rows(sel, final, db)

    @staticmethod
    def _count_star_only(sel) -> bool:
        return (not sel.select_star and not sel.group_by
                and not sel.order_by and not sel.distinct
                and getattr(sel, "having", None) is None
                and bool(sel.variables)
                and all(p.aggregate == "COUNT" and p.agg_arg is None
                        and not p.distinct for p in sel.variables))

    def _pre_shrink(self, sel, rows: Bindings) -> Bindings:
        """Cut gather traffic where legal: local DISTINCT before a global
        DISTINCT; local head() when LIMIT without ORDER BY/aggregates."""
        has_agg = any(p.aggregate for p in sel.variables)
        if has_agg or sel.group_by:
            return rows
        if sel.distinct and not sel.order_by and rows.n > 1:
            from ..engine.tensor_utils import unique_rows
            names = rows.variables
            uc = unique_rows([rows.col(v) for v in names])
            rows = Bindings(dict(zip(names, uc)),
                            uc[0].numel() if uc else 0, self.device,
                            maybe_unbound=rows.maybe_unbound)
        if (sel.limit is not None and not sel.order_by and not sel.distinct):
            keep = (sel.offset or 0) + sel.limit
            if rows.n > keep:
                idx = torch.arange(keep, dtype=torch.long, device=rows.device)
                rows = rows.gather(idx)
        return rows

    def _finalize_grouped(self, sel, rows: Bindings) -> List[List[str]]:
        """Distributed GROUP BY: hash-exchange rows on
        the first group key so each rank owns complete groups, aggregate
        locally (HAVING included), decode, then all-gather the decoded rows
        and apply ORDER BY / OFFSET / LIMIT globally."""
        import torch.distributed as dist
        db = self.db
        gv = next((v for v in sel.group_by if rows.has(v)), None)
        if gv is not None:
            key = rows.col(gv).to(torch.int64) & 0xFFFFFFFF
            dest = key % self.world
            names = rows.variables
            cols = D.all_to_all_rows([rows.col(v) for v in names], dest)
            rows = Bindings(dict(zip(names, cols)),
                            cols[0].numel() if cols else 0, self.device,
                            maybe_unbound=rows.maybe_unbound)
        local_sel = replace(sel, order_by=[], limit=None, offset=None)
        final = finalize_select_bindings(local_sel, rows, db)
        decoded = decode_rows(local_sel, final, db)
        parts: List[Optional[list]] = [None] * self.world
        dist.all_gather_object(parts, decoded)
        merged: List[List[str]] = []
        for p in parts:
            merged.extend(p)
        merged = _order_decoded(sel, merged)
        off = sel.offset or 0
        if off or sel.limit is not None:
            end = len(merged) if sel.limit is None else off + sel.limit
            merged = merged[off:end]
        return merged


def _order_decoded(sel, rows: List[List[str]]) -> List[List[str]]:
    """ORDER BY on decoded rows, matching finalize's numeric-aware
    semantics (numeric sort when every key parses as a number, else
    lexicographic) — ref execute_query.rs:477."""
    if not sel.order_by or len(rows) <= 1:
        return rows
    names = [p.output_name() for p in sel.variables] if sel.variables else []
    for cond in reversed(sel.order_by):
        if cond.var not in names:
            continue
        j = names.index(cond.var)
        vals = [r[j] for r in rows]
        numeric = all(_is_num(v) for v in vals)
        if numeric:
            rows = sorted(rows, key=lambda r: float(r[j]),
                          reverse=cond.descending)
        else:
            rows = sorted(rows, key=lambda r: r[j], reverse=cond.descending)
    return rows


def _is_num(s: str) -> bool:
    try:
        float(s)
        return True
    except ValueError:
        return False
