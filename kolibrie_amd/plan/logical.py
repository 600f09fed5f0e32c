"""Logical plan IR (ref: streamertail_optimizer/operators/logical.rs:18-69).

Operators: Unit, Scan{QuadPattern+graph scope}, Union, Graph, Selection,
Projection, Join, Buffer, Subquery, Bind, Values, MLPredict.  Graph scope is
carried on each scan (reference lowers GRAPH onto scans, utils.rs:402-577).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from ..storage.terms import Term, TriplePattern

# graph scope: None = dataset default view; ("const", gid) = fixed named
# graph; ("var", name) = GRAPH ?g variable scan over named graphs.
GraphScope = Optional[Tuple[str, object]]


@dataclass
class LogicalOp:
    pass


@dataclass
class LUnit(LogicalOp):
    pass


@dataclass
class LScan(LogicalOp):
    pattern: TriplePattern
    graph: GraphScope = None


@dataclass
class LJoin(LogicalOp):
    left: LogicalOp
    right: LogicalOp


@dataclass
class LUnion(LogicalOp):
    left: LogicalOp
    right: LogicalOp


@dataclass
class LSelection(LogicalOp):
    condition: object          # compiled filter expr (engine/filters.py)
    input: LogicalOp = field(default_factory=LUnit)


@dataclass
class LProjection(LogicalOp):
    variables: List[str] = field(default_factory=list)
    input: LogicalOp = field(default_factory=LUnit)


@dataclass
class LBind(LogicalOp):
    expr: object               # compiled bind expr
    var: str = ""
    input: LogicalOp = field(default_factory=LUnit)


@dataclass
class LValues(LogicalOp):
    variables: List[str] = field(default_factory=list)
    rows: List[List[Optional[int]]] = field(default_factory=list)  # i32 ids / None=UNDEF
    input: LogicalOp = field(default_factory=LUnit)


@dataclass
class LSubquery(LogicalOp):
    select: object             # CompiledSubquery (plan/lower.py)
    input: LogicalOp = field(default_factory=LUnit)


@dataclass
class LMLPredict(LogicalOp):
    info: dict = field(default_factory=dict)
    input: LogicalOp = field(default_factory=LUnit)


@dataclass
class LMinus(LogicalOp):
    """NAF anti-join (NOT { ... } in rule bodies / MINUS)."""
    left: LogicalOp = field(default_factory=LUnit)
    right: LogicalOp = field(default_factory=LUnit)


@dataclass
class LLeftJoin(LogicalOp):
    """OPTIONAL left outer join."""
    left: LogicalOp = field(default_factory=LUnit)
    right: LogicalOp = field(default_factory=LUnit)


def scans_of(op: LogicalOp) -> List[LScan]:
    out: List[LScan] = []

    def rec(x: LogicalOp):
        if isinstance(x, LScan):
            out.append(x)
        elif isinstance(x, LJoin) or isinstance(x, LUnion):
            rec(x.left)
            rec(x.right)
        elif hasattr(x, "input"):
            rec(x.input)

    rec(op)
    return out
