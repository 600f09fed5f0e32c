"""Cost model (ref: streamertail_optimizer/cost/estimator.rs:54-64 constants,
:228-350 estimation).

Constants keep the reference's skeleton (scan=100/row, index=1/row with a
10^bound discount, hash join=2/row, NLJ=10/row, probe=2, ML.PREDICT
1000 + 100/feature/row) with two GPU-era terms: a fixed kernel-launch cost
(so tiny inputs prefer fused/bind joins) and an xGMI shuffle term used by the
distributed planner (per-link ~153 GB/s — SURVEY §2.10)."""
from __future__ import annotations

from typing import Set

from ..storage.terms import Constant, TriplePattern, Variable
from .stats import DatabaseStats

SCAN_COST_PER_ROW = 100.0
INDEX_COST_PER_ROW = 1.0
HASH_JOIN_COST_PER_ROW = 2.0
NLJ_COST_PER_ROW = 10.0
PROBE_COST = 2.0
KERNEL_LAUNCH_COST = 2000.0       # ~10 us launch ≈ 2000 row-equivalents
ML_PREDICT_BASE = 1000.0
ML_PREDICT_PER_FEATURE_ROW = 100.0
XGMI_BYTES_PER_ROW_COST = 12.0 / 153.0   # 3 int32 over 153 GB/s link


class CostEstimator:
    def __init__(self, stats: DatabaseStats):
        self.stats = stats

    def bound_positions(self, pattern: TriplePattern, bound_vars: Set[str]):
        def is_b(t):
            if isinstance(t, Constant):
                return True
            if isinstance(t, Variable):
                return t.name in bound_vars
            return False  # quoted-pattern: treated as unbound probe
        return is_b(pattern.s), is_b(pattern.p), is_b(pattern.o)

    def estimate_scan(self, pattern: TriplePattern, bound_vars: Set[str],
                      graph=None) -> float:
        """Cardinality of the scan result given already-bound variables
        (ref estimator.rs:314-350 estimate_bound_scan_cardinality)."""
        s_b, p_b, o_b = self.bound_positions(pattern, bound_vars)
        st = self.stats
        proxy = TriplePattern(
            Constant(0) if s_b else Variable("_s"),
            pattern.p if isinstance(pattern.p, Constant) else (
                Constant(0) if p_b else Variable("_p")),
            Constant(0) if o_b else Variable("_o"),
        )
        if isinstance(pattern.p, Constant):
            pid = pattern.p.id & 0xFFFFFFFF
            pc = st.pred_count.get(pid, 0)
            if pc == 0:
                return 0.0
            if s_b and o_b:
                return 1.0
            if s_b:
                return max(1.0, pc / max(1, st.pred_distinct_subj.get(pid, 1)))
            if o_b:
                return max(1.0, pc / max(1, st.pred_distinct_obj.get(pid, 1)))
            return float(pc)
        return st.estimate_pattern(proxy, graph)

    def scan_cost(self, pattern: TriplePattern, bound_vars: Set[str],
                  graph=None) -> float:
        s_b, p_b, o_b = self.bound_positions(pattern, bound_vars)
        n_bound = int(s_b) + int(p_b) + int(o_b)
        est = self.estimate_scan(pattern, bound_vars, graph)
        if n_bound == 0:
            return KERNEL_LAUNCH_COST + SCAN_COST_PER_ROW * max(1.0, est) / 100.0
        return KERNEL_LAUNCH_COST + INDEX_COST_PER_ROW * max(1.0, est)
