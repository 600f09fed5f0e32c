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
    if hasattr(op, "input"):
        return f"{type(op).__name__}[{plan_key(op.input)}]"
    return type(op).__name__
