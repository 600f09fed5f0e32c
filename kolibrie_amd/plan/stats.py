"""Database statistics for the cost-based optimizer.

Ref: streamertail_optimizer/stats/database_stats.rs:18-158 — predicate /
subject / object cardinalities, per-predicate distinct subjects/objects,
graph cardinalities, quoted-triple count.

MI355X-native: gathered as device reductions over the committed sorted
columns (torch unique/segment ops; K8 class at 100M scale) instead of the
reference's sampled host scan — no sampling needed when the reduce runs at
HBM bandwidth.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict

import torch

from ..engine.tensor_utils import group_index
from ..storage.terms import Constant, Variable


@dataclass
class DatabaseStats:
    total: int = 0
    pred_count: Dict[int, int] = field(default_factory=dict)
    pred_distinct_subj: Dict[int, int] = field(default_factory=dict)
    pred_distinct_obj: Dict[int, int] = field(default_factory=dict)
    distinct_subjects: int = 1
    distinct_objects: int = 1
    graph_counts: Dict[int, int] = field(default_factory=dict)
    quoted_count: int = 0

    @staticmethod
    def gather(db) -> "DatabaseStats":
        st = DatabaseStats()
        store = db.store
        store.commit_all()
        s_all, p_all, o_all = [], [], []
        for g, buf in store.graphs.items():
            idx = buf.index
            if idx.n == 0:
                continue
            st.graph_counts[g] = idx.n
            s, p, o = idx.columns()
            s_all.append(s)
            p_all.append(p)
            o_all.append(o)
        if not s_all:
            return st
        s = torch.cat(s_all)
        p = torch.cat(p_all)
        o = torch.cat(o_all)
        st.total = int(s.numel())
        st.quoted_count = len(db.quoted_triples)
        # predicate histogram
        pv, pc = torch.unique(p, return_counts=True)
        for pid, cnt in zip(pv.tolist(), pc.tolist()):
            st.pred_count[pid & 0xFFFFFFFF] = cnt
        st.distinct_subjects = max(1, int(torch.unique(s).numel()))
        st.distinct_objects = max(1, int(torch.unique(o).numel()))
        # per-predicate distinct subj/obj via (p,x) pair dedup then histogram
        from ..engine.tensor_utils import unique_rows
        for col, out in ((s, st.pred_distinct_subj), (o, st.pred_distinct_obj)):
            up = unique_rows([p, col])[0]
            pv2, pc2 = torch.unique(up, return_counts=True)
            for pid, cnt in zip(pv2.tolist(), pc2.tolist()):
                out[pid & 0xFFFFFFFF] = cnt
        return st

    # ---------------------------------------------------------- estimation --
    def estimate_pattern(self, pattern, graph=None) -> float:
        """Per-pattern cardinality (ref cost/estimator.rs:248-311 9-case)."""
        s_b = isinstance(pattern.s, Constant)
        p_b = isinstance(pattern.p, Constant)
        o_b = isinstance(pattern.o, Constant)
        total = max(1, self.total)
        if graph is not None and graph[0] == "const":
            total = max(1, self.graph_counts.get(graph[1] & 0xFFFFFFFF, total))
        if p_b:
            pid = pattern.p.id & 0xFFFFFFFF
            pc = self.pred_count.get(pid, 0)
            if pc == 0:
                return 0.0
            if s_b and o_b:
                return 1.0
            if s_b:
                return max(1.0, pc / max(1, self.pred_distinct_subj.get(pid, 1)))
            if o_b:
                return max(1.0, pc / max(1, self.pred_distinct_obj.get(pid, 1)))
            return float(pc)
        if s_b and o_b:
            return max(1.0, total / max(1, self.distinct_subjects * self.distinct_objects))
        if s_b:
            return max(1.0, total / self.distinct_subjects)
        if o_b:
            return max(1.0, total / self.distinct_objects)
        return float(total)
