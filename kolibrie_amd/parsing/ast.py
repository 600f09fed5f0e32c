"""Query AST (parity: shared/src/query.rs — FilterExpression :15,
ArithmeticExpression :25, GroupGraphPattern :105-121, SelectQuery :364,
UpdateOperation :379-402, CombinedQuery :411-424, WindowClause :206-253,
CombinedRule :303-322, ModelDecl/NeuralRelationDecl :169-202).

Terms inside the AST stay *surface strings* (`?x`, `<iri>`, `"lit"`,
`pfx:name`, `<< .. >>`); the planner compiles them to dictionary IDs
(plan/lower.py, mirroring utils.rs:192 compile_term).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional


# ------------------------------------------------------------ expressions ---
@dataclass
class Expr:
    pass


@dataclass
class EVar(Expr):
    name: str  # without '?'


@dataclass
class ELit(Expr):
    value: str          # lexical value as stored in the dictionary
    is_number: bool = False


@dataclass
class EArith(Expr):
    op: str  # + - * /
    left: Expr
    right: Expr


@dataclass
class ECmp(Expr):
    op: str  # = != < > <= >=
    left: Expr
    right: Expr


@dataclass
class EAnd(Expr):
    left: Expr
    right: Expr


@dataclass
class EOr(Expr):
    left: Expr
    right: Expr


@dataclass
class ENot(Expr):
    inner: Expr


@dataclass
class EFunc(Expr):
    name: str            # upper-cased: CONCAT, TRIPLE, SUBJECT, ... or UDF name
    args: List[Expr]


# --------------------------------------------------------------- patterns ---
@dataclass
class TriplePatternAst:
    s: str
    p: str
    o: str
    path_mod: Optional[str] = None   # '+' | '*' transitive-closure steps


@dataclass
class GGP:
    """Group graph pattern node (ref query.rs:105-121)."""
    pass


@dataclass
class GUnit(GGP):
    pass


@dataclass
class GBgp(GGP):
    patterns: List[TriplePatternAst] = field(default_factory=list)


@dataclass
class GJoin(GGP):
    left: GGP
    right: GGP


@dataclass
class GUnion(GGP):
    left: GGP
    right: GGP


@dataclass
class GGraph(GGP):
    graph: str          # term or variable surface string
    inner: GGP


@dataclass
class GFilter(GGP):
    expr: Expr
    inner: GGP


@dataclass
class GBind(GGP):
    expr: Expr
    var: str            # target variable name (no '?')
    inner: GGP


@dataclass
class GValues(GGP):
    variables: List[str]
    rows: List[List[Optional[str]]]   # None == UNDEF
    inner: GGP


@dataclass
class GSubQuery(GGP):
    select: "SelectQuery"
    inner: GGP


@dataclass
class GWindowBlock(GGP):
    """RSP-QL `WINDOW :w { patterns }` block (parser.rs:249)."""
    window_iri: str
    inner: GGP


@dataclass
class GMinus(GGP):
    """NOT { ... } atoms in rule bodies (NAF)."""
    left: GGP
    right: GGP


@dataclass
class GOptional(GGP):
    """OPTIONAL { ... } — left outer join (engine extension beyond the
    reference's SPARQL subset)."""
    left: GGP
    right: GGP


# ------------------------------------------------------------------ select ---
@dataclass
class Projection:
    var: Optional[str] = None           # plain ?var
    aggregate: Optional[str] = None     # COUNT/SUM/AVG/MIN/MAX
    agg_arg: Optional[str] = None       # variable name or None for COUNT(*)
    alias: Optional[str] = None         # AS ?alias
    distinct: bool = False              # COUNT(DISTINCT ?x)

    def output_name(self) -> str:
        if self.alias:
            return self.alias
        if self.var:
            return self.var
        arg = self.agg_arg or "*"
        return f"{self.aggregate}({arg})"


@dataclass
class OrderCondition:
    var: str
    descending: bool = False


@dataclass
class SelectQuery:
    variables: List[Projection] = field(default_factory=list)  # empty => '*'
    distinct: bool = False
    from_graphs: List[str] = field(default_factory=list)
    from_named: List[str] = field(default_factory=list)
    where: GGP = field(default_factory=GUnit)
    group_by: List[str] = field(default_factory=list)
    order_by: List[OrderCondition] = field(default_factory=list)
    limit: Optional[int] = None
    offset: Optional[int] = None
    select_star: bool = False
    ask: bool = False                  # ASK query (engine extension)
    having: Optional["Expr"] = None    # HAVING expr (engine extension)


# ------------------------------------------------------------------ update ---
@dataclass
class QuadData:
    g: Optional[str]   # None = default graph
    s: str
    p: str
    o: str


@dataclass
class UpdateOperation:
    kind: str  # insert_data | delete_data | delete_where | modify | clear | create | drop
    quads: List[QuadData] = field(default_factory=list)       # data forms
    delete_templates: List[QuadData] = field(default_factory=list)
    insert_templates: List[QuadData] = field(default_factory=list)
    where: Optional[GGP] = None
    graph: Optional[str] = None       # clear/create/drop target
    silent: bool = False


# ---------------------------------------------------------------- streaming --
@dataclass
class WindowSpec:
    window_type: str = "RANGE"            # RANGE | TUMBLING | SLIDING
    width: int = 0                        # seconds
    slide: Optional[int] = None
    report: Optional[str] = None          # ON_WINDOW_CLOSE | ...
    tick: Optional[str] = None            # TIME_DRIVEN | ...


@dataclass
class SyncPolicy:
    kind: str = "Wait"                    # Steal | Wait | Timeout
    timeout_ms: Optional[int] = None
    fallback: str = "Steal"               # Timeout expiry action: Steal | Drop


@dataclass
class WindowClause:
    window_iri: str
    stream_iri: str
    spec: WindowSpec
    policy: Optional[SyncPolicy] = None


@dataclass
class RegisterClause:
    stream_type: str                      # RSTREAM | ISTREAM | DSTREAM
    output_iri: str
    select: SelectQuery
    windows: List[WindowClause] = field(default_factory=list)


# --------------------------------------------------------------------- rules --
@dataclass
class ProbAnnotation:
    provenance: str = "independent"       # independent|min|max|topk|wmc|minmax|hybrid
    threshold: Optional[float] = None
    confidence: Optional[float] = None
    extra: Dict[str, str] = field(default_factory=dict)


@dataclass
class CombinedRule:
    name: str
    head_vars: List[str] = field(default_factory=list)
    stream_type: Optional[str] = None
    windows: List[WindowClause] = field(default_factory=list)
    conclusions: List[TriplePatternAst] = field(default_factory=list)
    body: GGP = field(default_factory=GUnit)
    negated: List[TriplePatternAst] = field(default_factory=list)
    prob: Optional[ProbAnnotation] = None
    ml_predict: Optional[dict] = None


# ----------------------------------------------------------------- ML decls --
@dataclass
class ModelDecl:
    name: str
    path: str
    options: Dict[str, str] = field(default_factory=dict)


@dataclass
class NeuralRelationDecl:
    name: str
    model: str
    inputs: List[str] = field(default_factory=list)
    output_predicate: Optional[str] = None
    options: Dict[str, str] = field(default_factory=dict)


@dataclass
class TrainNeuralRelationDecl:
    name: str
    target: str                              # target rule/relation name
    data_patterns: List[TriplePatternAst] = field(default_factory=list)
    data_query: Optional[SelectQuery] = None
    options: Dict[str, str] = field(default_factory=dict)


@dataclass
class RetrieveClause:
    mode: str        # SOME | EVERY
    state: str       # LATENT | ACTIVE
    streams: List[str] = field(default_factory=list)


# ------------------------------------------------------------ combined query --
@dataclass
class CombinedQuery:
    prefixes: Dict[str, str] = field(default_factory=dict)
    models: List[ModelDecl] = field(default_factory=list)
    neural_relations: List[NeuralRelationDecl] = field(default_factory=list)
    train_decls: List[TrainNeuralRelationDecl] = field(default_factory=list)
    rules: List[CombinedRule] = field(default_factory=list)
    register: Optional[RegisterClause] = None
    retrieve: Optional[RetrieveClause] = None
    select: Optional[SelectQuery] = None
    construct: Optional[List[TriplePatternAst]] = None   # CONSTRUCT templates
    describe: Optional[List[str]] = None                 # DESCRIBE terms
    updates: List[UpdateOperation] = field(default_factory=list)
