"""Distributed plan post-pass: insert PExchange ops where a join key is not
the partition key.

Partitioning model: the store is SUBJECT-hash partitioned — quad (g,s,p,o)
lives on rank (s & 0xFFFFFFFF) % world (what `DistributedDatabase.load_*`
enforces).  Every rank builds the SAME physical plan from the SAME
(replicated) dictionary and global stats, then this pass rewrites it into a
distributed plan with these invariants per subtree:

  part = ("hash", v):  rows live on rank hash(row[v]) % world; the disjoint
                       union across ranks is the global result multiset.
  part = "scattered":  rows are SOME disjoint partition of the global
                       result (union invariant holds, key unknown).
  part = "replicated": every rank holds the FULL global result.

A local join is correct when matching rows are co-located: both sides
("hash", k) on the same shared key k, or one side replicated.  Otherwise
the pass inserts PExchange(hash k) on the non-co-located side(s), or
broadcasts a side whose estimated cardinality is below BROADCAST_ROWS
(xGMI traffic = the small table instead of the big intermediate).

Compat-join hazard: a row whose shared key is UNBOUND must meet EVERY row
of the other side, so hash exchange on k is only sound when k is bound in
every solution of both sides (op_certain_vars); otherwise the smaller side
is broadcast.

Ref semantics source: engine.rs:1367 hash_join_solution_sequences /
join_algorithm.rs:64 (the joins being distributed).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Set, Tuple, Union

from ..plan.physical import (
    PBind, PBindJoin, PConstStar, PExchange, PFilter, PHashJoin, PIndexScan,
    PInMemoryBuffer, PLeftJoin, PMLPredict, PMinus, PNestedLoopJoin,
    PProjection, PStarJoin, PSubquery, PTableScan, PUnion, PUnit, PValues,
    PhysicalOp, op_certain_vars,
)
from ..plan.optimizer import phys_out_vars
from ..storage.terms import Constant, Variable

# broadcast a side instead of hash-exchanging when its estimated row count
# is below this (the all-gathered table is then the cheaper xGMI payload);
# KOLIBRIE_BCAST_ROWS overrides (0 forces hash shuffles — used by tests)
BROADCAST_ROWS = 200_000


def _bcast_threshold() -> float:
    import os
    v = os.environ.get("KOLIBRIE_BCAST_ROWS")
    return float(v) if v else BROADCAST_ROWS

REPLICATED = "replicated"
SCATTERED = "scattered"
Part = Union[str, Tuple[str, str]]  # "replicated" | "scattered" | ("hash",v)


@dataclass
class DistPlanner:
    stats: object        # DatabaseStats (global, allreduced)
    world: int
    # predicates whose relations the loader REPLICATED on every rank (the
    # load-time broadcast-table layout for small build sides): scans of
    # these are part=REPLICATED and never need an exchange
    replicated_preds: frozenset = frozenset()

    def _is_replicated_scan(self, op: PhysicalOp) -> bool:
        if isinstance(op, (PTableScan, PIndexScan)):
            return (isinstance(op.pattern.p, Constant)
                    and (op.pattern.p.id & 0xFFFFFFFF)
                    in self.replicated_preds)
        if isinstance(op, PStarJoin):
            return all(isinstance(p.p, Constant)
                       and (p.p.id & 0xFFFFFFFF) in self.replicated_preds
                       for p in op.patterns)
        if isinstance(op, (PFilter, PBind, PProjection)):
            return self._is_replicated_scan(op.input)
        return False

    # ------------------------------------------------------------ helpers --
    def _est_rows(self, op: PhysicalOp) -> float:
        """Crude subtree cardinality for the broadcast decision."""
        from ..plan.cost import CostEstimator
        est = CostEstimator(self.stats)
        if isinstance(op, (PTableScan, PIndexScan)):
            return est.estimate_scan(op.pattern, set(), op.graph)
        if isinstance(op, PStarJoin):
            per = [est.estimate_scan(p, set(), op.graph) for p in op.patterns]
            return min(per) if per else 1.0
        if isinstance(op, PConstStar):
            return float(len(op.items))
        if isinstance(op, PFilter):
            return max(1.0, self._est_rows(op.input) / 3.0)
        if isinstance(op, (PBind, PProjection, PExchange)):
            return self._est_rows(op.input)
        if isinstance(op, PValues):
            return self._est_rows(op.input) * max(1, len(op.rows))
        if isinstance(op, PUnion):
            return self._est_rows(op.left) + self._est_rows(op.right)
        if isinstance(op, (PHashJoin, PBindJoin, PNestedLoopJoin)):
            return max(self._est_rows(op.left), self._est_rows(op.right))
        if isinstance(op, (PMinus, PLeftJoin)):
            return self._est_rows(op.left)
        if isinstance(op, PInMemoryBuffer) and op.bindings is not None:
            return float(op.bindings.n)
        return float("inf")  # unknown (subquery/ML): never auto-broadcast

    @staticmethod
    def _bcast(op: PhysicalOp, part: Part) -> Tuple[PhysicalOp, Part]:
        if part == REPLICATED:
            return op, REPLICATED
        return PExchange(op, "", "broadcast"), REPLICATED

    @staticmethod
    def _hash_to(op: PhysicalOp, part: Part, var: str
                 ) -> Tuple[PhysicalOp, Part]:
        if part == ("hash", var):
            return op, part
        return PExchange(op, var, "hash"), ("hash", var)

    # --------------------------------------------------------------- core --
    def distribute(self, op: PhysicalOp) -> Tuple[PhysicalOp, Part]:
        """Returns (rewritten plan, partition property of its output)."""
        if isinstance(op, PUnit):
            return op, REPLICATED
        if isinstance(op, (PTableScan, PIndexScan)):
            if self._is_replicated_scan(op):
                return op, REPLICATED
            if isinstance(op.pattern.s, Variable):
                return op, ("hash", op.pattern.s.name)
            return op, SCATTERED  # const subject: all rows on one rank
        if isinstance(op, PStarJoin):
            if self._is_replicated_scan(op):
                return op, REPLICATED
            # all patterns share the subject var: co-located by definition
            return op, ("hash", op.join_var)
        if isinstance(op, PConstStar):
            return op, SCATTERED
        if isinstance(op, PFilter):
            child, part = self.distribute(op.input)
            op.input = child
            return op, part
        if isinstance(op, PProjection):
            child, part = self.distribute(op.input)
            op.input = child
            if isinstance(part, tuple) and part[1] not in op.variables:
                part = SCATTERED  # key column projected away
            return op, part
        if isinstance(op, PBind):
            child, part = self.distribute(op.input)
            op.input = child
            return op, part
        if isinstance(op, PValues):
            # VALUES rows are identical on every rank (replicated table
            # joined with the child)
            child, part = self.distribute(op.input)
            op.input = child
            return op, part if not isinstance(op.input, PUnit) else REPLICATED
        if isinstance(op, PMLPredict):
            child, part = self.distribute(op.input)
            op.input = child
            return op, part
        if isinstance(op, PSubquery):
            # inner select runs per rank over the local shard: distribute
            # its plan and replicate its (finalized) result so the outer
            # join sees the global subquery answer on every rank
            sel = op.select
            if getattr(sel, "physical", None) is not None:
                inner, ipart = self.distribute(sel.physical)
                if ipart != REPLICATED:
                    inner = PExchange(inner, "", "broadcast")
                sel.physical = inner
            child, part = self.distribute(op.input)
            op.input = child
            return op, part
        if isinstance(op, PInMemoryBuffer):
            return op, REPLICATED  # RSP-injected tables are rank-local state
        if isinstance(op, PUnion):
            l, lp = self.distribute(op.left)
            r, rp = self.distribute(op.right)
            if lp == REPLICATED and rp == REPLICATED:
                op.left, op.right = l, r
                return op, REPLICATED
            # mixed: de-duplicate the replicated branch to rank 0 so the
            # union-invariant holds
            if lp == REPLICATED:
                l = PExchange(l, "", "rank0")
                lp = SCATTERED
            if rp == REPLICATED:
                r = PExchange(r, "", "rank0")
                rp = SCATTERED
            op.left, op.right = l, r
            if lp == rp and isinstance(lp, tuple):
                return op, lp
            return op, SCATTERED
        if isinstance(op, (PMinus, PLeftJoin)):
            l, lp = self.distribute(op.left)
            r, rp = self.distribute(op.right)
            shared = phys_out_vars(op.left) & phys_out_vars(op.right)
            key = self._pick_key(op.left, op.right, shared, lp, rp)
            if (key is not None and lp == ("hash", key)
                    and rp == ("hash", key)):
                pass  # co-partitioned on a certain shared key: local is exact
            else:
                # right side must see the GLOBAL right solutions
                r, rp = self._bcast(r, rp)
            op.left, op.right = l, r
            return op, lp
        if isinstance(op, (PHashJoin, PNestedLoopJoin)):
            return self._dist_join(op)
        if isinstance(op, PBindJoin):
            return self._dist_bindjoin(op)
        if isinstance(op, PExchange):
            raise ValueError("distribute() called on an already-distributed "
                             "plan (nested PExchange)")
        raise ValueError(f"cannot distribute {type(op).__name__}")

    def _pick_key(self, left: PhysicalOp, right: PhysicalOp,
                  shared: Set[str], lp: Part, rp: Part) -> Optional[str]:
        """A shared var bound in EVERY solution of both sides (hash-exchange
        soundness); prefer one a side is already partitioned on."""
        certain = op_certain_vars(left) & op_certain_vars(right) & shared
        if not certain:
            return None
        for p in (lp, rp):
            if isinstance(p, tuple) and p[1] in certain:
                return p[1]
        return sorted(certain)[0]

    def _dist_join(self, op) -> Tuple[PhysicalOp, Part]:
        l, lp = self.distribute(op.left)
        r, rp = self.distribute(op.right)
        return self._dist_join_core(op, l, lp, r, rp)

    def _dist_join_core(self, op, l, lp, r, rp) -> Tuple[PhysicalOp, Part]:
        """Join placement with ALREADY-distributed children (l, r)."""
        op.left, op.right = l, r
        shared = phys_out_vars(l) & phys_out_vars(r)
        if lp == REPLICATED and rp == REPLICATED:
            op.left, op.right = l, r
            return op, REPLICATED
        if rp == REPLICATED:
            op.left, op.right = l, r
            return op, lp
        if lp == REPLICATED:
            op.left, op.right = l, r
            return op, rp
        key = self._pick_key(op.left, op.right, shared, lp, rp)
        if key is None:
            # cartesian or unbound-capable keys: broadcast the smaller side
            if self._est_rows(op.left) <= self._est_rows(op.right):
                l, lp = self._bcast(l, lp)
                out = rp
            else:
                r, rp = self._bcast(r, rp)
                out = lp
            op.left, op.right = l, r
            return op, out
        if lp == ("hash", key) and rp == ("hash", key):
            op.left, op.right = l, r
            return op, lp
        # broadcast a small side instead of moving the big one
        le, re_ = self._est_rows(op.left), self._est_rows(op.right)
        if re_ <= _bcast_threshold() and re_ <= le and rp != ("hash", key):
            r, rp = self._bcast(r, rp)
            op.left, op.right = l, r
            return op, lp
        if le <= _bcast_threshold() and le < re_ and lp != ("hash", key):
            l, lp = self._bcast(l, lp)
            op.left, op.right = l, r
            return op, rp
        l, lp = self._hash_to(l, lp, key)
        r, rp = self._hash_to(r, rp, key)
        op.left, op.right = l, r
        return op, ("hash", key)

    def _dist_bindjoin(self, op: PBindJoin) -> Tuple[PhysicalOp, Part]:
        """Dependent join: the right side probes the LOCAL shard per left
        row, so left rows must be co-located with the shard rows they
        probe — partitioned on the probed pattern's subject var, or
        replicated (every rank probes its own shard with the full left
        table; the disjoint shard partition keeps the union invariant)."""
        l, lp = self.distribute(op.left)
        right = op.right
        probe_subject = self._probe_subject(right)
        lvars = phys_out_vars(l)
        if self._leafish(right) and self._is_replicated_scan(right):
            # probing a replicated relation is local for ANY left layout
            op.left = l
            return op, lp
        if not self._leafish(right):
            # compound right (e.g. const-star seeding a chain): its internal
            # joins need their own exchanges — distribute it, after placing
            # the left where the chain's seed scan can probe it locally
            if (probe_subject is not None and probe_subject in lvars
                    and probe_subject in op_certain_vars(l)
                    and lp != REPLICATED):
                l, lp = self._hash_to(l, lp, probe_subject)
            elif lp != REPLICATED:
                l, lp = self._bcast(l, lp)
            r, rp = self.distribute(right)
            op.left, op.right = l, r
            return op, rp
        if lp == REPLICATED:
            op.left = l
            return op, self._bindjoin_out_part(right, probe_subject)
        if (probe_subject is not None and probe_subject in lvars
                and probe_subject in op_certain_vars(l)):
            l, lp = self._hash_to(l, lp, probe_subject)
            op.left = l
            return op, lp
        # probe keyed on object/predicate (or uncertain subject binding):
        # replicate the left table, or fall back to a hash join with
        # exchanges when the left is too big to broadcast
        if self._est_rows(l) > _bcast_threshold():
            newop = PHashJoin(l, right)
            newop.needed = getattr(op, "needed", None)
            r, rp = self.distribute(right)
            return self._dist_join_core(newop, l, lp, r, rp)
        l, lp = self._bcast(l, lp)
        op.left = l
        return op, self._bindjoin_out_part(right, probe_subject)

    @staticmethod
    def _leafish(op: PhysicalOp) -> bool:
        """Right sides safe to probe without internal distribution: scans
        (and scans under filter/bind/projection) — no internal joins."""
        if isinstance(op, (PTableScan, PIndexScan, PStarJoin, PConstStar)):
            return True
        if isinstance(op, (PFilter, PBind, PProjection)):
            return DistPlanner._leafish(op.input)
        return False

    @staticmethod
    def _probe_subject(right: PhysicalOp) -> Optional[str]:
        """Subject variable of the right (probed) side's FIRST pattern —
        the var whose left-row values key the local-shard probe."""
        if isinstance(right, (PTableScan, PIndexScan)):
            if isinstance(right.pattern.s, Variable):
                return right.pattern.s.name
            return None
        if isinstance(right, PStarJoin):
            return right.join_var
        if isinstance(right, (PFilter, PBind, PProjection)):
            return DistPlanner._probe_subject(right.input)
        if isinstance(right, PBindJoin):
            return DistPlanner._probe_subject(right.left)
        return None

    @staticmethod
    def _bindjoin_out_part(right: PhysicalOp, probe_subject: Optional[str]
                           ) -> Part:
        """Output rows of a replicated-left bind join live where the shard
        rows they matched live: hash of the probed pattern's subject."""
        if probe_subject is not None:
            return ("hash", probe_subject)
        return SCATTERED


def distribute_plan(physical: PhysicalOp, stats, world: int,
                    replicated_preds=frozenset()) -> Tuple[PhysicalOp, Part]:
    """Entry point: rewrite a single-GPU physical plan for `world` ranks.
    Returns (plan, partition property of the root output)."""
    if world <= 1:
        return physical, REPLICATED
    return DistPlanner(stats, world,
                       frozenset(replicated_preds)).distribute(physical)
