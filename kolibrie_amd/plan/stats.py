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
        st.quoted_count = len(db.quoted_triples)
        live = [buf.index for buf in store.graphs.values() if buf.index.n]
        if len(live) == 1:
            # common case: every statistic is an order statistic of the
            # sorted index permutations the store ALREADY maintains —
            # transition counts + segment cumsums, pure bandwidth
            st.total = live[0].n
            st._gather_from_index(live[0])
            return st
        s = torch.cat(s_all)
        p = torch.cat(p_all)
        o = torch.cat(o_all)
        st.total = int(s.numel())
        if s.is_cuda:
            from ..ops import _native
            if _native is None:
                raise RuntimeError(
                    "native extension required for device stats gather")
            st._gather_native(_native, s, p, o)
            return st
        st._gather_torch(s, p, o)
        return st

    def _gather_from_index(self, idx):
        """Stats from the sorted orders (single-graph path): distinct
        subjects/objects = leading-word transition counts of SPO/OSP;
        per-predicate row counts and distinct subjects/objects = segment
        boundaries + full-key transition cumsums over PSO/POS.  Measured
        at 100M rows: ~2 ms vs 50 ms torch-unique composite vs 111 ms
        one-pass hash-claim kernel (stats_gather, kept for the
        multi-graph union case)."""
        from ..storage.dataset import OSP, POS, PSO, SPO

        def transitions(k):
            return 1 + int((k[1:] != k[:-1]).sum().item())

        k_spo, _ = idx.orders[SPO]
        self.distinct_subjects = max(1, transitions(k_spo >> 32))
        k_osp, _ = idx.orders[OSP]
        self.distinct_objects = max(1, transitions(k_osp >> 32))
        n = idx.n
        for code, out in ((PSO, self.pred_distinct_subj),
                          (POS, self.pred_distinct_obj)):
            key, _ = idx.orders[code]
            hi = key >> 32
            change = torch.nonzero(hi[1:] != hi[:-1]).flatten() + 1
            z = torch.zeros(1, dtype=change.dtype, device=change.device)
            starts = torch.cat([z, change])
            ends = torch.cat([change, torch.full((1,), n, dtype=change.dtype,
                                                 device=change.device)])
            preds = (hi[starts] & 0xFFFFFFFF).tolist()
            # distinct full keys per segment: t[i]=1 iff key starts a new
            # run; segment starts always do (p changed), so the count in
            # [a,b) is C[b-1] - C[a-1] with C the inclusive cumsum
            t = torch.ones(n, dtype=torch.int64, device=key.device)
            t[1:] = (key[1:] != key[:-1]).to(torch.int64)
            c = torch.cumsum(t, 0)
            hi_c = c[ends - 1]
            lo_c = torch.where(starts > 0, c[(starts - 1).clamp(min=0)],
                               torch.zeros_like(hi_c))
            seg = (hi_c - lo_c).tolist()
            cnts = (ends - starts).tolist()
            if code == PSO:
                for pid, rows in zip(preds, cnts):
                    self.pred_count[pid] = rows
            for pid, d in zip(preds, seg):
                out[pid] = d

    def _gather_native(self, _native, s, p, o):
        """K8 kernel: one pass over the columns computes the predicate
        histogram, the global distinct subject/object counts and the
        per-predicate distinct subjects/objects (ops/csrc/kernels.hip
        stats_gather_kernel)."""
        pred_keys, pred_rows, pred_ds, pred_do, gc = _native.stats_gather(
            s.contiguous(), p.contiguous(), o.contiguous())
        live = (pred_keys != -1).nonzero(as_tuple=True)[0]
        keys = pred_keys[live].cpu().tolist()
        rows = pred_rows[live].cpu().tolist()
        ds = pred_ds[live].cpu().tolist()
        do_ = pred_do[live].cpu().tolist()
        for pid, cnt, a, b in zip(keys, rows, ds, do_):
            pid &= 0xFFFFFFFF
            self.pred_count[pid] = cnt
            self.pred_distinct_subj[pid] = a
            self.pred_distinct_obj[pid] = b
        g = gc.cpu().tolist()
        self.distinct_subjects = max(1, int(g[0]))
        self.distinct_objects = max(1, int(g[1]))

    def _gather_torch(self, s, p, o):
        # predicate histogram
        pv, pc = torch.unique(p, return_counts=True)
        for pid, cnt in zip(pv.tolist(), pc.tolist()):
            self.pred_count[pid & 0xFFFFFFFF] = cnt
        self.distinct_subjects = max(1, int(torch.unique(s).numel()))
        self.distinct_objects = max(1, int(torch.unique(o).numel()))
        # per-predicate distinct subj/obj via (p,x) pair dedup then histogram
        from ..engine.tensor_utils import unique_rows
        for col, out in ((s, self.pred_distinct_subj),
                         (o, self.pred_distinct_obj)):
            up = unique_rows([p, col])[0]
            pv2, pc2 = torch.unique(up, return_counts=True)
            for pid, cnt in zip(pv2.tolist(), pc2.tolist()):
                out[pid & 0xFFFFFFFF] = cnt

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
