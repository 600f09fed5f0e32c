"""Distributed BGP query execution over subject-partitioned shards.

BASELINE config 3: 100M triples hash-partitioned across 8x MI355X, with the
join shuffle as an RCCL all-to-all over xGMI.  The execution strategy
(SURVEY §2.10):

  1. subject-keyed patterns evaluate rank-locally (the shard IS the hash
     bucket),
  2. when the next join key is not the partition key, the intermediate row
     table is re-partitioned with all_to_all_rows keyed by hash(join var),
  3. the probe continues against the local shard of the other pattern,
  4. aggregates finish with an all-reduce.

This module drives the single-node engine per rank; `DistributedDatabase`
holds the local shard in a SparqlDatabase and executes shuffle plans.
"""
from __future__ import annotations


import torch

from ..engine.bindings import Bindings
from ..engine.executor import DatasetView, ExecutionContext, ExecutionEngine
from ..storage.database import SparqlDatabase
from . import dist as D


class DistributedDatabase:
    """A rank-local shard plus the collective query driver."""

    def __init__(self, rank: int, world: int, device):
        self.rank = rank
        self.world = world
        self.device = torch.device(device)
        self.db = SparqlDatabase(device=str(device))

    def load_shard_columns(self, s, p, o):
        self.db.load_columns(s, p, o)

    BROADCAST_MAX_ROWS = 4_000_000

    def count_query_with_shuffle(
        self,
        local_star_sparql: str,
        shuffle_var: str,
        probe_sparql: str,
    ) -> int:
        """Execute: local subquery -> exchange on `shuffle_var` -> local
        join with `probe_sparql` results -> global COUNT.

        Exchange strategy (cost-based): when the
        probe side is small it is REPLICATED with an all-gather (broadcast
        join — the xGMI traffic is the small table, not the big
        intermediate); otherwise both sides hash-repartition with the
        pairwise all-to-all row shuffle.
        """
        left = self._rows_for(local_star_sparql)
        right = self._rows_for(probe_sparql)
        from ..engine.executor import join_bindings
        if self.world > 1:
            right_total = D.allreduce_sum_scalar(right.n, self.device)
            if right_total <= self.BROADCAST_MAX_ROWS:
                names = right.variables
                cols = D.all_gather_rows([right.col(v) for v in names])
                right = Bindings(dict(zip(names, cols)),
                                 cols[0].numel() if cols else 0, self.device)
            else:
                key = left.col(shuffle_var).to(torch.int64) & 0xFFFFFFFF
                dest = key % self.world
                names = left.variables
                cols = D.all_to_all_rows([left.col(v) for v in names], dest)
                left = Bindings(dict(zip(names, cols)),
                                cols[0].numel() if cols else 0, self.device)
        joined = join_bindings(left, right, needed=set())
        local = joined.n
        return D.allreduce_sum_scalar(local, self.device)

    def _rows_for(self, sparql: str) -> Bindings:
        from ..parsing.sparql import parse_combined_query
        from ..plan.lower import build_logical_plan
        from ..plan.optimizer import Streamertail
        db = self.db
        cq = parse_combined_query(sparql)
        prefixes = dict(db.prefixes)
        prefixes.update(cq.prefixes)
        sel = cq.select
        stats = db.get_or_build_stats()
        logical = build_logical_plan(sel.where, db, prefixes)
        physical = Streamertail(stats).find_best_plan(logical)
        from ..engine.query import _top_needed
        from ..plan.optimizer import annotate_needed
        annotate_needed(physical, _top_needed(sel))
        ctx = ExecutionContext(db, DatasetView())
        rows = ExecutionEngine(ctx).execute(physical, Bindings.unit(db.device))
        # apply projection only (no aggregates here)
        if not sel.select_star and sel.variables:
            rows = rows.project([p.output_name() for p in sel.variables])
        return rows
