"""Physical plan IR (ref: streamertail_optimizer/operators/physical.rs:18-88).

Each node names the device algorithm the executor dispatches to: index scans
probe the sorted column permutations (HIP kernel K1 on device), HashJoin is
the K2 build/probe pair, BindJoin the K3 dependent join, StarJoin a pipelined
subject-star chain, Filter the K5 predicate kernel, etc.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

from ..storage.terms import TriplePattern
from .logical import GraphScope


@dataclass
class PhysicalOp:
    pass


@dataclass
class PUnit(PhysicalOp):
    pass


@dataclass
class PTableScan(PhysicalOp):
    pattern: TriplePattern
    graph: GraphScope = None
    sort_hint: Optional[int] = None  # position (0=s,1=p,2=o) to sort output by


@dataclass
class PIndexScan(PhysicalOp):
    pattern: TriplePattern
    graph: GraphScope = None
    sort_hint: Optional[int] = None


@dataclass
class PUnion(PhysicalOp):
    left: PhysicalOp
    right: PhysicalOp


@dataclass
class PFilter(PhysicalOp):
    condition: object
    input: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PHashJoin(PhysicalOp):
    left: PhysicalOp
    right: PhysicalOp


@dataclass
class PBindJoin(PhysicalOp):
    """Dependent / index-nested-loop join: left rows drive right-side probes
    (ref engine.rs:1289 execute_bind_join, chunked >= 64)."""
    left: PhysicalOp
    right: PhysicalOp


@dataclass
class PNestedLoopJoin(PhysicalOp):
    left: PhysicalOp
    right: PhysicalOp


@dataclass
class PStarJoin(PhysicalOp):
    """Subject-star: >=3 patterns sharing a subject var; pipelined scans
    (ref optimizer.rs:293 is_star_query, engine.rs:496-512)."""
    join_var: str
    patterns: List[TriplePattern] = field(default_factory=list)
    graph: GraphScope = None


@dataclass
class PProjection(PhysicalOp):
    variables: List[str] = field(default_factory=list)
    input: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PBind(PhysicalOp):
    expr: object = None
    var: str = ""
    input: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PValues(PhysicalOp):
    variables: List[str] = field(default_factory=list)
    rows: List[List[Optional[int]]] = field(default_factory=list)
    input: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PSubquery(PhysicalOp):
    select: object = None
    input: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PMLPredict(PhysicalOp):
    info: dict = field(default_factory=dict)
    input: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PMinus(PhysicalOp):
    """Anti-join: keep left rows with no compatible right row (NAF)."""
    left: PhysicalOp = field(default_factory=PUnit)
    right: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PConstStar(PhysicalOp):
    """Star of >=2 (CONST subject, CONST predicate, ?var) patterns: fetch
    the subject's SPO region once and evaluate every pattern host-side in
    one pass (1 device sync instead of 2 per pattern — the S4-class
    point-query shape)."""
    subject_id: int = 0
    items: tuple = ()         # ((pid_i32, out_var), ...)


@dataclass
class PLeftJoin(PhysicalOp):
    """OPTIONAL: inner join plus unmatched left rows padded UNBOUND."""
    left: PhysicalOp = field(default_factory=PUnit)
    right: PhysicalOp = field(default_factory=PUnit)


@dataclass
class PInMemoryBuffer(PhysicalOp):
    """Materialized bindings injected by the RSP runtime."""
    bindings: object = None


@dataclass
class PExchange(PhysicalOp):
    """Distributed re-partition: the planner inserts
    this when a join key is not the current partition key.

    mode="hash":      rows move to rank hash(row[var]) % world via the
                      pairwise xGMI all-to-all (all_to_all_rows).
    mode="broadcast": the (small) input is replicated on every rank via
                      all-gather — the broadcast-join build side.
    Single-process execution is the identity."""
    input: PhysicalOp = field(default_factory=PUnit)
    var: str = ""            # partition key variable (hash mode)
    mode: str = "hash"       # "hash" | "broadcast"


def _graph_var(graph: GraphScope):
    if graph is not None and graph[0] == "var":
        return graph[1]
    return None


def op_certain_vars(op: PhysicalOp) -> set:
    """Variables bound in EVERY solution of the (sub)plan — the static
    domain analysis MINUS needs (SPARQL spec: MINUS removes a left row only
    when a compatible right row shares >=1 *bound* variable with it; a
    disjoint-domain MINUS removes nothing).  Conservative: returns a subset
    of the true certain set; unknown node kinds contribute nothing."""
    if isinstance(op, (PTableScan, PIndexScan)):
        vs = set(op.pattern.variables())
        gv = _graph_var(op.graph)
        if gv is not None:
            vs.add(gv)
        return vs
    if isinstance(op, PStarJoin):
        vs = {op.join_var}
        for pat in op.patterns:
            vs |= set(pat.variables())
        gv = _graph_var(op.graph)
        if gv is not None:
            vs.add(gv)
        return vs
    if isinstance(op, PConstStar):
        return {v for _p, v in op.items}
    if isinstance(op, (PHashJoin, PBindJoin, PNestedLoopJoin)):
        return op_certain_vars(op.left) | op_certain_vars(op.right)
    if isinstance(op, PUnion):
        return op_certain_vars(op.left) & op_certain_vars(op.right)
    if isinstance(op, (PLeftJoin, PMinus)):
        return op_certain_vars(op.left)
    if isinstance(op, (PFilter, PExchange)):
        return op_certain_vars(op.input)
    if isinstance(op, PBind):
        base = op_certain_vars(op.input)
        if not getattr(op.expr, "may_produce_unbound", False):
            base = base | {op.var}
        return base
    if isinstance(op, PValues):
        base = op_certain_vars(op.input)
        if op.rows:
            for j, v in enumerate(op.variables):
                if all(r[j] is not None for r in op.rows):
                    base = base | {v}
        return base
    if isinstance(op, PProjection):
        return op_certain_vars(op.input) & set(op.variables)
    if isinstance(op, PInMemoryBuffer) and op.bindings is not None:
        b = op.bindings
        if not getattr(b, "maybe_unbound", True):
            return set(b.variables)
        return set()
    return set()


def op_possible_vars(op: PhysicalOp):
    """Superset of the variables the (sub)plan can ever bind, or None when
    the node kind is opaque (subquery/ML) — callers must treat None as
    "could bind anything"."""
    if isinstance(op, (PTableScan, PIndexScan, PStarJoin, PConstStar)):
        return op_certain_vars(op)
    if isinstance(op, (PHashJoin, PBindJoin, PNestedLoopJoin, PUnion,
                       PLeftJoin)):
        l = op_possible_vars(op.left)
        r = op_possible_vars(op.right)
        if l is None or r is None:
            return None
        return l | r
    if isinstance(op, PMinus):
        return op_possible_vars(op.left)
    if isinstance(op, (PFilter, PExchange)):
        return op_possible_vars(op.input)
    if isinstance(op, PBind):
        base = op_possible_vars(op.input)
        return None if base is None else base | {op.var}
    if isinstance(op, PValues):
        base = op_possible_vars(op.input)
        return None if base is None else base | set(op.variables)
    if isinstance(op, PProjection):
        return set(op.variables)
    if isinstance(op, PInMemoryBuffer) and op.bindings is not None:
        return set(op.bindings.variables)
    if isinstance(op, PUnit):
        return set()
    return None


def plan_key(op: PhysicalOp) -> str:
    """Serialize a plan to a memo key (ref optimizer.rs:751-848)."""
    if isinstance(op, PUnit):
        return "U"
    if isinstance(op, (PTableScan, PIndexScan)):
        tag = "T" if isinstance(op, PTableScan) else "I"
        return f"{tag}({op.pattern},{op.graph})"
    if isinstance(op, PStarJoin):
        return f"S({op.join_var},{op.patterns},{op.graph})"
    if isinstance(op, (PHashJoin, PBindJoin, PNestedLoopJoin, PUnion)):
        tag = type(op).__name__[1:3]
        return f"{tag}[{plan_key(op.left)},{plan_key(op.right)}]"
    if isinstance(op, PExchange):
        return f"X({op.mode},{op.var})[{plan_key(op.input)}]"
    if hasattr(op, "input"):
        return f"{type(op).__name__}[{plan_key(op.input)}]"
    return type(op).__name__
