"""SDD engine: bottom-up compilation of lineage circuits to a canonical
decision diagram with exact weighted model counting.

Ref parity: shared/src/sdd.rs (1 686 LoC) — SddManager with vtree,
unique-table, apply (:468), negate (:715), one-pass WMC (:739),
exactly_one (:229), model enumeration (:780), per-var pos/neg weights +
VarKind (:130), budgeted try_* operations with node cap + deadline
(:821-1240); SddProvenance (:1241).

This manager fixes a right-linear vtree over the variable order — under a
right-linear vtree the compressed/trimmed SDD coincides with the ROBDD of
the same order, which keeps apply() canonical and WMC a single bottom-up
pass while preserving the full SddManager surface.
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, Iterator, List, Optional, Sequence, Tuple

TRUE = 1
FALSE = 0
# internal nodes are ids >= 2


class VarKind:
    STREAM = "stream"
    STATIC = "static"
    SYNTHETIC = "synthetic"


@dataclass
class SddOperationBudget:
    """Node cap + deadline (ref sdd.rs:92-128)."""
    max_nodes: Optional[int] = None
    deadline_s: Optional[float] = None
    _start: float = 0.0

    def start(self):
        self._start = time.monotonic()
        return self

    def exceeded(self, n_nodes: int) -> bool:
        if self.max_nodes is not None and n_nodes > self.max_nodes:
            return True
        if self.deadline_s is not None and \
                time.monotonic() - self._start > self.deadline_s:
            return True
        return False


class BudgetExceeded(Exception):
    pass


class SddManager:
    def __init__(self):
        # node id -> (var, hi, lo); var order = vtree order (right-linear)
        self.nodes: List[Tuple[int, int, int]] = [(-1, 0, 0), (-1, 1, 1)]
        self.unique: Dict[Tuple[int, int, int], int] = {}
        self.apply_cache: Dict[Tuple[str, int, int], int] = {}
        self.neg_cache: Dict[int, int] = {}
        self.var_order: Dict[int, int] = {}
        self.pos_weight: Dict[int, float] = {}
        self.neg_weight: Dict[int, float] = {}
        self.var_kind: Dict[int, str] = {}
        self._budget: Optional[SddOperationBudget] = None

    # -------------------------------------------------------------- vtree
    def declare_var(self, var: int, pos_weight: float = 1.0,
                    neg_weight: Optional[float] = None,
                    kind: str = VarKind.STREAM):
        if var not in self.var_order:
            self.var_order[var] = len(self.var_order)
        self.pos_weight[var] = pos_weight
        self.neg_weight[var] = (1.0 - pos_weight) if neg_weight is None \
            else neg_weight
        self.var_kind[var] = kind

    def _rank(self, var: int) -> int:
        if var not in self.var_order:
            self.declare_var(var)
        return self.var_order[var]

    # -------------------------------------------------------- construction
    def true_node(self) -> int:
        return TRUE

    def false_node(self) -> int:
        return FALSE

    def literal(self, var: int, positive: bool = True) -> int:
        self._rank(var)
        return self._mk(var, TRUE, FALSE) if positive else self._mk(var, FALSE, TRUE)

    def _mk(self, var: int, hi: int, lo: int) -> int:
        if hi == lo:
            return hi
        key = (var, hi, lo)
        nid = self.unique.get(key)
        if nid is None:
            nid = len(self.nodes)
            self.nodes.append(key)
            self.unique[key] = nid
            if self._budget is not None and self._budget.exceeded(len(self.nodes)):
                raise BudgetExceeded()
        return nid

    def _var(self, nid: int) -> int:
        return self.nodes[nid][0]

    # --------------------------------------------------------------- apply
    def apply(self, op: str, a: int, b: int) -> int:
        """op in {'and','or','xor'} (ref sdd.rs:468)."""
        if op == "and":
            if a == FALSE or b == FALSE:
                return FALSE
            if a == TRUE:
                return b
            if b == TRUE:
                return a
            if a == b:
                return a
        elif op == "or":
            if a == TRUE or b == TRUE:
                return TRUE
            if a == FALSE:
                return b
            if b == FALSE:
                return a
            if a == b:
                return a
        elif op == "xor":
            if a == b:
                return FALSE
            if a == FALSE:
                return b
            if b == FALSE:
                return a
        key = (op, min(a, b), max(a, b))
        hit = self.apply_cache.get(key)
        if hit is not None:
            return hit
        va, vb = self._var(a), self._var(b)
        ra = self._rank(va) if a > TRUE else 1 << 60
        rb = self._rank(vb) if b > TRUE else 1 << 60
        if ra <= rb:
            var = va
            a_hi, a_lo = self.nodes[a][1], self.nodes[a][2]
        else:
            var = vb
            a_hi = a_lo = a
        if rb <= ra:
            b_hi, b_lo = self.nodes[b][1], self.nodes[b][2]
        else:
            b_hi = b_lo = b
        res = self._mk(var, self.apply(op, a_hi, b_hi),
                       self.apply(op, a_lo, b_lo))
        self.apply_cache[key] = res
        return res

    def conjoin(self, a: int, b: int) -> int:
        return self.apply("and", a, b)

    def disjoin(self, a: int, b: int) -> int:
        return self.apply("or", a, b)

    def negate(self, a: int) -> int:
        """(ref sdd.rs:715)"""
        if a == TRUE:
            return FALSE
        if a == FALSE:
            return TRUE
        hit = self.neg_cache.get(a)
        if hit is not None:
            return hit
        var, hi, lo = self.nodes[a]
        res = self._mk(var, self.negate(hi), self.negate(lo))
        self.neg_cache[a] = res
        self.neg_cache[res] = a
        return res

    def exactly_one(self, vars_: Sequence[int]) -> int:
        """⊕ constraint over a set of variables (ref sdd.rs:229)."""
        total = FALSE
        for v in vars_:
            term = self.literal(v, True)
            for w in vars_:
                if w != v:
                    term = self.conjoin(term, self.literal(w, False))
            total = self.disjoin(total, term)
        return total

    # ---------------------------------------------------------------- WMC
    def wmc(self, node: int) -> float:
        """One-pass weighted model count (ref sdd.rs:739).  Unconstrained
        variables are marginalized: w+ + w- per skipped level."""
        order = sorted(self.var_order, key=self.var_order.get)
        memo: Dict[int, float] = {}

        def level_weight_span(from_rank: int, to_rank: int) -> float:
            w = 1.0
            for r in range(from_rank + 1, to_rank):
                v = order[r]
                w *= self.pos_weight.get(v, 1.0) + self.neg_weight.get(v, 0.0)
            return w

        def rank_of(nid: int) -> int:
            return len(order) if nid <= TRUE else self._rank(self._var(nid))

        def rec(nid: int) -> float:
            if nid == TRUE:
                return 1.0
            if nid == FALSE:
                return 0.0
            hit = memo.get(nid)
            if hit is not None:
                return hit
            var, hi, lo = self.nodes[nid]
            r = self._rank(var)
            whi = rec(hi) * level_weight_span(r, rank_of(hi))
            wlo = rec(lo) * level_weight_span(r, rank_of(lo))
            res = self.pos_weight.get(var, 1.0) * whi \
                + self.neg_weight.get(var, 0.0) * wlo
            memo[nid] = res
            return res

        root_span = level_weight_span(-1, rank_of(node))
        return rec(node) * root_span

    # ------------------------------------------------------ model iteration
    def models(self, node: int) -> Iterator[Dict[int, bool]]:
        """Enumerate satisfying assignments over CONSTRAINED variables
        (ref sdd.rs:780)."""
        def rec(nid: int, acc: Dict[int, bool]):
            if nid == FALSE:
                return
            if nid == TRUE:
                yield dict(acc)
                return
            var, hi, lo = self.nodes[nid]
            acc[var] = True
            yield from rec(hi, acc)
            acc[var] = False
            yield from rec(lo, acc)
            del acc[var]

        yield from rec(node, {})

    def node_count(self) -> int:
        return len(self.nodes)

    # -------------------------------------------------------- budgeted ops
    def try_apply(self, op: str, a: int, b: int,
                  budget: SddOperationBudget) -> Optional[int]:
        """(ref sdd.rs:821-1240 try_* with node cap + deadline)"""
        self._budget = budget.start()
        try:
            return self.apply(op, a, b)
        except BudgetExceeded:
            return None
        finally:
            self._budget = None

    def try_conjoin(self, a: int, b: int, budget) -> Optional[int]:
        return self.try_apply("and", a, b, budget)

    def try_disjoin(self, a: int, b: int, budget) -> Optional[int]:
        return self.try_apply("or", a, b, budget)


class SddProvenance:
    """Semiring over SDD nodes (ref sdd.rs:1241): ⊕=disjoin, ⊗=conjoin,
    negate, recover=WMC."""
    name = "sdd"

    def __init__(self, manager: Optional[SddManager] = None):
        self.manager = manager if manager is not None else SddManager()
        self._next_var = 1

    def zero(self):
        return FALSE

    def one(self):
        return TRUE

    def plus(self, a, b):
        return self.manager.disjoin(a, b)

    def times(self, a, b):
        return self.manager.conjoin(a, b)

    def negate(self, a):
        return self.manager.negate(a)

    def saturate(self, a):
        return a

    def tag_from_probability(self, p: float, seed_id: Optional[int] = None):
        var = seed_id if seed_id is not None else self._next_var
        self._next_var = max(self._next_var, var) + 1
        self.manager.declare_var(var, pos_weight=float(p))
        return self.manager.literal(var, True)

    def recover(self, tag) -> float:
        return self.manager.wmc(tag)

    def better(self, a, b) -> bool:
        return a != b and self.recover(a) > self.recover(b)
