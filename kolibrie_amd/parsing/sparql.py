"""SPARQL + extension grammar (recursive descent).

Parity surface: kolibrie/src/parser.rs (4 115 LoC nom grammar) — standards
SPARQL SELECT/UPDATE core plus the extension grammar: RSP-QL REGISTER with
FROM NAMED WINDOW and window specs (:2594-2860), WINDOW blocks (:249),
RULE ... :- CONSTRUCT {..} WHERE {..} with PROB annotations (:3101, :2931),
MODEL (:2219), NEURAL RELATION (:2291), TRAIN NEURAL RELATION (:2391),
ML.PREDICT (:2504), RETRIEVE (:3201).

Entry points: parse_combined_query(text) -> CombinedQuery;
parse_sparql_query(text) -> SelectQuery.
"""
from __future__ import annotations

import re
from typing import Dict, List, Optional

from .ast import (
    CombinedQuery, CombinedRule, EAnd, EArith, ECmp, EFunc, ELit, ENot, EOr,
    EVar, Expr, GBgp, GBind, GFilter, GGP, GGraph, GJoin, GMinus, GOptional,
    GSubQuery,
    GUnion, GUnit, GValues, GWindowBlock, ModelDecl, NeuralRelationDecl,
    OrderCondition, ProbAnnotation, Projection, QuadData, RegisterClause,
    RetrieveClause, SelectQuery, SyncPolicy, TrainNeuralRelationDecl,
    TriplePatternAst, UpdateOperation, WindowClause, WindowSpec,
)


class ParseError(ValueError):
    """Parse failure with positional context (ref: error_handler.rs:14)."""

    def __init__(self, msg: str, text: str = "", pos: int = 0):
        self.msg = msg
        self.text = text
        self.pos = pos
        super().__init__(self.pretty())

    def pretty(self) -> str:
        if not self.text:
            return self.msg
        line_no = self.text.count("\n", 0, self.pos) + 1
        line_start = self.text.rfind("\n", 0, self.pos) + 1
        line_end = self.text.find("\n", self.pos)
        if line_end < 0:
            line_end = len(self.text)
        col = self.pos - line_start
        line = self.text[line_start:line_end]
        caret = " " * col + "^"
        return f"{self.msg}\n  --> line {line_no}:{col + 1}\n   | {line}\n   | {caret}"


_TOK = re.compile(
    r"""
      (?P<ws>\s+|\#[^\n]*)
    | (?P<qopen><<)
    | (?P<qclose>>>)
    | (?P<aopen>\{\|)
    | (?P<aclose>\|\})
    | (?P<iri><[^<>"{}|^`\\\s]*>)
    | (?P<string>
        ("(?:[^"\\]|\\.)*"|'(?:[^'\\]|\\.)*')
        (?:\^\^<[^<>\s]*>|\^\^[A-Za-z_][\w.-]*:[\w.-]*|@[A-Za-z][A-Za-z0-9-]*)?
      )
    | (?P<var>[?$][A-Za-z_][\w]*)
    | (?P<number>[+-]?(\d+\.\d*|\.\d+|\d+)([eE][+-]?\d+)?)
    | (?P<ruleop>:-)
    | (?P<op><=|>=|!=|&&|\|\||[=<>!+\-*/^|])
    | (?P<punct>[{}()\[\],;.])
    | (?P<bnode>_:[A-Za-z0-9_.-]+)
    | (?P<pname>[A-Za-z_][\w.-]*:[\w.-]*|:[\w.-]+)
    | (?P<name>[A-Za-z_][\w.-]*)
    | (?P<at>@[A-Za-z][A-Za-z0-9-]*)
    """,
    re.VERBOSE | re.DOTALL,
)

KEYWORDS = {
    "SELECT", "WHERE", "DISTINCT", "FROM", "NAMED", "GROUP", "ORDER", "BY",
    "ASC", "DESC", "LIMIT", "OFFSET", "PREFIX", "FILTER", "BIND", "VALUES",
    "UNION", "GRAPH", "AS", "UNDEF", "OPTIONAL", "INSERT", "DELETE", "DATA", "CLEAR",
    "CREATE", "DROP", "SILENT", "ALL", "DEFAULT", "WINDOW", "REGISTER",
    "RSTREAM", "ISTREAM", "DSTREAM", "RANGE", "TUMBLING", "SLIDING", "STEP",
    "REPORT", "TICK", "ON", "STREAM", "WITH", "POLICY", "RULE", "CONSTRUCT",
    "NOT", "HAVING", "MODEL", "NEURAL", "RELATION", "TRAIN", "USING", "RETRIEVE",
    "SOME", "EVERY", "LATENT", "ACTIVE", "PROB", "COUNT", "SUM", "AVG",
    "MIN", "MAX", "MINUS", "ASK",
}

AGGREGATES = {"COUNT", "SUM", "AVG", "MIN", "MAX"}


class _Tok:
    __slots__ = ("kind", "text", "pos")

    def __init__(self, kind, text, pos):
        self.kind = kind
        self.text = text
        self.pos = pos

    def upper(self):
        return self.text.upper()

    def __repr__(self):
        return f"{self.kind}:{self.text!r}"


def _tokenize(text: str) -> List[_Tok]:
    toks = []
    pos = 0
    n = len(text)
    while pos < n:
        m = _TOK.match(text, pos)
        if m is None:
            raise ParseError(f"unexpected character {text[pos]!r}", text, pos)
        if m.lastgroup != "ws":
            toks.append(_Tok(m.lastgroup, m.group(), pos))
        pos = m.end()
    return toks


class Parser:
    def __init__(self, text: str):
        self.text = text
        self.toks = _tokenize(text)
        self.i = 0

    # ------------------------------------------------------------- helpers --
    def peek(self, ahead: int = 0) -> Optional[_Tok]:
        j = self.i + ahead
        return self.toks[j] if j < len(self.toks) else None

    def at_kw(self, *kws: str, ahead: int = 0) -> bool:
        t = self.peek(ahead)
        return t is not None and t.kind in ("name", "pname") and t.upper() in kws

    def at(self, text: str, ahead: int = 0) -> bool:
        t = self.peek(ahead)
        return t is not None and t.text == text

    def next(self) -> _Tok:
        t = self.peek()
        if t is None:
            raise ParseError("unexpected end of query", self.text, len(self.text))
        self.i += 1
        return t

    def expect(self, text: str) -> _Tok:
        t = self.next()
        if t.text != text:
            raise ParseError(f"expected {text!r}, found {t.text!r}", self.text, t.pos)
        return t

    def expect_kw(self, kw: str) -> _Tok:
        t = self.next()
        if t.upper() != kw:
            raise ParseError(f"expected {kw}, found {t.text!r}", self.text, t.pos)
        return t

    def eof(self) -> bool:
        return self.i >= len(self.toks)

    def err(self, msg: str) -> ParseError:
        t = self.peek()
        return ParseError(msg, self.text, t.pos if t else len(self.text))

    # ---------------------------------------------------------------- terms --
    def parse_term(self) -> str:
        """One term in surface syntax (quoted triples re-assembled)."""
        t = self.peek()
        if t is None:
            raise self.err("expected term")
        if t.kind == "qopen":
            self.next()
            s = self.parse_term()
            p = self.parse_term()
            o = self.parse_term()
            tt = self.next()
            if tt.kind != "qclose":
                raise ParseError("expected '>>'", self.text, tt.pos)
            return f"<< {s} {p} {o} >>"
        if t.kind in ("iri", "string", "var", "number", "bnode", "pname"):
            return self.next().text
        if t.kind == "name":
            return self.next().text
        raise ParseError(f"expected term, found {t.text!r}", self.text, t.pos)

    # ------------------------------------------------------------- prologue --
    def parse_prefixes(self, prefixes: Dict[str, str]):
        while self.at_kw("PREFIX") or self.at("@prefix"):
            self.next()
            name_tok = self.next()
            name = name_tok.text
            if name.endswith(":"):
                name = name[:-1]
            elif name_tok.kind == "pname" and name.startswith(":"):
                name = ""
            elif name_tok.kind == "pname":
                name = name.split(":", 1)[0]
            iri_tok = self.next()
            iri = iri_tok.text
            if iri.startswith("<") and iri.endswith(">"):
                iri = iri[1:-1]
            prefixes[name] = iri
            if self.at("."):
                self.next()

    # ---------------------------------------------------------------- entry --
    def parse_combined(self) -> CombinedQuery:
        cq = CombinedQuery()
        while not self.eof():
            self.parse_prefixes(cq.prefixes)
            if self.eof():
                break
            if self.at_kw("MODEL"):
                cq.models.append(self.parse_model_decl())
            elif self.at_kw("NEURAL") and self.at_kw("RELATION", ahead=1):
                cq.neural_relations.append(self.parse_neural_relation())
            elif self.at_kw("TRAIN"):
                cq.train_decls.append(self.parse_train_decl())
            elif self.at_kw("RULE"):
                cq.rules.append(self.parse_rule())
            elif self.at_kw("REGISTER"):
                cq.register = self.parse_register()
            elif self.at_kw("RETRIEVE"):
                cq.retrieve = self.parse_retrieve()
            elif self.at_kw("SELECT"):
                cq.select = self.parse_select_core()
            elif self.at_kw("CONSTRUCT"):
                # standalone CONSTRUCT query (engine extension)
                self.next()
                self.expect("{")
                pats = []
                while not self.at("}"):
                    if self.at("."):
                        self.next()
                        continue
                    pats.extend(self.parse_triples_block().patterns)
                self.expect("}")
                self.expect_kw("WHERE")
                q = SelectQuery(select_star=True)
                q.where = self.parse_group()
                while True:
                    if self.at_kw("LIMIT"):
                        self.next()
                        q.limit = int(self.next().text)
                    elif self.at_kw("OFFSET"):
                        self.next()
                        q.offset = int(self.next().text)
                    else:
                        break
                cq.select = q
                cq.construct = pats
            elif self.at_kw("DESCRIBE"):
                # DESCRIBE <iri>... [WHERE {...}] (engine extension)
                self.next()
                terms = []
                while not self.eof() and not self.at_kw("WHERE") \
                        and (self.peek().kind in ("iri", "pname", "var")):
                    terms.append(self.next().text)
                q = SelectQuery(select_star=True)
                if self.at_kw("WHERE"):
                    self.next()
                    q.where = self.parse_group()
                cq.select = q
                cq.describe = terms
            elif self.at_kw("ASK"):
                # ASK { pattern } — boolean query (engine extension)
                self.next()
                q = SelectQuery(select_star=True, ask=True, limit=1)
                q.where = self.parse_group()
                cq.select = q
            elif self.at_kw("INSERT", "DELETE", "CLEAR", "CREATE", "DROP"):
                cq.updates.extend(self.parse_update_ops())
            elif self.at(";"):
                self.next()
            else:
                raise self.err(f"unexpected token {self.peek().text!r}")
        return cq

    # --------------------------------------------------------------- select --
    def parse_select_core(self) -> SelectQuery:
        self.expect_kw("SELECT")
        q = SelectQuery()
        if self.at_kw("DISTINCT"):
            self.next()
            q.distinct = True
        # projections
        while True:
            t = self.peek()
            if t is None:
                raise self.err("unterminated SELECT")
            if t.text == "*":
                self.next()
                q.select_star = True
                continue
            if t.kind == "var":
                self.next()
                q.variables.append(Projection(var=t.text[1:]))
                continue
            if t.text == "(":
                self.next()
                proj = self.parse_aggregate_projection(require_alias=True)
                self.expect(")")
                q.variables.append(proj)
                continue
            if t.kind in ("name",) and t.upper() in AGGREGATES:
                q.variables.append(self.parse_aggregate_projection(require_alias=False))
                continue
            break
        # datasets
        while self.at_kw("FROM"):
            self.next()
            if self.at_kw("NAMED"):
                self.next()
                if self.at_kw("WINDOW"):
                    # caller (REGISTER) handles FROM NAMED WINDOW; rewind
                    self.i -= 2
                    break
                q.from_named.append(self.parse_term())
            else:
                q.from_graphs.append(self.parse_term())
        if self.at_kw("WHERE"):
            self.next()
            q.where = self.parse_group()
        # solution modifiers
        while True:
            if self.at_kw("GROUP"):
                self.next()
                self.expect_kw("BY")
                while self.peek() is not None and self.peek().kind == "var":
                    q.group_by.append(self.next().text[1:])
            elif self.at_kw("HAVING"):
                self.next()
                q.having = self.parse_filter_expr()
            elif self.at_kw("ORDER"):
                self.next()
                self.expect_kw("BY")
                while True:
                    if self.at_kw("ASC") or self.at_kw("DESC"):
                        d = self.next().upper() == "DESC"
                        self.expect("(")
                        v = self.next()
                        self.expect(")")
                        q.order_by.append(OrderCondition(v.text[1:], d))
                    elif self.peek() is not None and self.peek().kind == "var":
                        q.order_by.append(OrderCondition(self.next().text[1:], False))
                    else:
                        break
            elif self.at_kw("LIMIT"):
                self.next()
                q.limit = int(self.next().text)
            elif self.at_kw("OFFSET"):
                self.next()
                q.offset = int(self.next().text)
            else:
                break
        return q

    def parse_aggregate_projection(self, require_alias: bool) -> Projection:
        func = self.next().upper()
        if func not in AGGREGATES:
            raise self.err(f"unknown aggregate {func}")
        self.expect("(")
        distinct = False
        if self.at_kw("DISTINCT"):
            self.next()
            distinct = True
        arg: Optional[str] = None
        t = self.next()
        if t.text == "*":
            arg = None
        elif t.kind == "var":
            arg = t.text[1:]
        else:
            raise ParseError("expected ?var or * in aggregate", self.text, t.pos)
        self.expect(")")
        alias = None
        if self.at_kw("AS"):
            self.next()
            alias = self.next().text[1:]
        elif require_alias:
            raise self.err("aggregate projection requires AS ?alias")
        return Projection(aggregate=func, agg_arg=arg, alias=alias, distinct=distinct)

    # ---------------------------------------------------------------- group --
    def parse_group(self) -> GGP:
        """`{ ... }` — returns the group pattern; FILTERs are deferred to the
        end of the group (SPARQL filter-scope, ref utils.rs:443-452)."""
        self.expect("{")
        current: GGP = GUnit()
        filters: List[Expr] = []
        while not self.at("}"):
            t = self.peek()
            if t is None:
                raise self.err("unterminated group")
            if t.text == "{":
                # nested group or subquery, possibly UNION chain
                node = self.parse_group_or_subquery()
                while self.at_kw("UNION"):
                    self.next()
                    rhs = self.parse_group_or_subquery()
                    node = GUnion(node, rhs)
                current = self._join(current, node)
            elif self.at_kw("GRAPH"):
                self.next()
                g = self.parse_term()
                inner = self.parse_group()
                current = self._join(current, GGraph(g, inner))
            elif self.at_kw("WINDOW"):
                self.next()
                w = self.parse_term()
                inner = self.parse_group()
                current = self._join(current, GWindowBlock(w, inner))
            elif self.at_kw("FILTER"):
                self.next()
                filters.append(self.parse_filter_expr())
            elif self.at_kw("BIND"):
                self.next()
                self.expect("(")
                expr = self.parse_expr()
                self.expect_kw("AS")
                v = self.next()
                if v.kind != "var":
                    raise ParseError("BIND requires ?var", self.text, v.pos)
                self.expect(")")
                current = GBind(expr, v.text[1:], current)
            elif self.at_kw("VALUES"):
                self.next()
                vars_, rows = self.parse_values_body()
                current = GValues(vars_, rows, current)
            elif self.at_kw("NOT", "MINUS"):
                self.next()
                inner = self.parse_group()
                current = GMinus(current, inner)
            elif self.at_kw("OPTIONAL"):
                self.next()
                inner = self.parse_group()
                current = GOptional(current, inner)
            elif t.text == ".":
                self.next()
            else:
                bgp = self.parse_triples_block(allow_alts=True)
                node: GGP = bgp
                for var, alts in getattr(self, "_alt_values", []):
                    node = GValues([var], [[a] for a in alts], node)
                if getattr(self, "_alt_values", None):
                    self._alt_values = []
                current = self._join(current, node)
        self.expect("}")
        for f in filters:
            current = GFilter(f, current)
        return current

    def parse_group_or_subquery(self) -> GGP:
        # lookahead for subquery
        if self.at("{") and self.at_kw("SELECT", ahead=1):
            self.expect("{")
            sub = self.parse_select_core()
            self.expect("}")
            return GSubQuery(sub, GUnit())
        return self.parse_group()

    @staticmethod
    def _join(left: GGP, right: GGP) -> GGP:
        if isinstance(left, GUnit):
            return right
        if isinstance(right, GUnit):
            return left
        if isinstance(left, GBgp) and isinstance(right, GBgp):
            return GBgp(left.patterns + right.patterns)
        return GJoin(left, right)

    _path_var_counter = [0]

    def _parse_path_steps(self):
        """Property-path subset (engine extension): sequences `p1/p2` and
        inverse steps `^p`, desugared to fresh-variable join chains."""
        steps = []
        while True:
            inv = False
            if self.at("^"):
                self.next()
                inv = True
            if self.at("("):
                # alternatives (p1|p2|...): desugar to a variable predicate
                # constrained by VALUES (engine extension)
                self.next()
                alts = [self.parse_term()]
                while self.at("|"):
                    self.next()
                    alts.append(self.parse_term())
                self.expect(")")
                pred = self._fresh_path_var()
                if not hasattr(self, "_alt_values"):
                    self._alt_values = []
                self._alt_values.append((pred[1:], alts))
            else:
                pred = self.parse_term()
            mod = None
            if self.at("+") or self.at("*"):
                mod = self.next().text
            steps.append((inv, pred, mod))
            if self.at("/"):
                self.next()
                continue
            break
        return steps

    def _fresh_path_var(self) -> str:
        Parser._path_var_counter[0] += 1
        return f"?__pp{Parser._path_var_counter[0]}"

    def parse_triples_block(self, allow_alts: bool = False) -> GBgp:
        """allow_alts: only parse_group consumes the (p1|p2) VALUES
        desugaring; every other caller (rule conclusions, ML inputs,
        CONSTRUCT/DESCRIBE templates) must reject alternatives instead of
        leaking the pending constraint into a later group."""
        pats: List[TriplePatternAst] = []
        s = self.parse_term()
        while True:
            steps = self._parse_path_steps()
            while True:
                o = self.parse_term()
                cur = s
                for j, (inv, pred, mod) in enumerate(steps):
                    nxt = o if j == len(steps) - 1 else self._fresh_path_var()
                    if inv:
                        pats.append(TriplePatternAst(nxt, pred, cur, mod))
                    else:
                        pats.append(TriplePatternAst(cur, pred, nxt, mod))
                    cur = nxt
                if self.at(","):
                    self.next()
                    continue
                break
            if self.at(";"):
                self.next()
                nxt = self.peek()
                if nxt is None or nxt.text in (".", "}", ";"):
                    continue
                continue
            break
        if self.at("."):
            self.next()
        if not allow_alts and getattr(self, "_alt_values", None):
            self._alt_values = []
            raise self.err("path alternatives (p|q) are only supported in "
                           "WHERE groups")
        return GBgp(pats)

    def parse_values_body(self):
        vars_: List[str] = []
        rows: List[List[Optional[str]]] = []
        t = self.peek()
        if t is not None and t.text == "(":
            self.next()
            while self.peek() is not None and self.peek().kind == "var":
                vars_.append(self.next().text[1:])
            self.expect(")")
            self.expect("{")
            while self.at("("):
                self.next()
                row: List[Optional[str]] = []
                while not self.at(")"):
                    if self.at_kw("UNDEF"):
                        self.next()
                        row.append(None)
                    else:
                        row.append(self.parse_term())
                self.expect(")")
                rows.append(row)
            self.expect("}")
        else:
            v = self.next()
            if v.kind != "var":
                raise ParseError("VALUES requires ?var", self.text, v.pos)
            vars_.append(v.text[1:])
            self.expect("{")
            while not self.at("}"):
                if self.at_kw("UNDEF"):
                    self.next()
                    rows.append([None])
                else:
                    rows.append([self.parse_term()])
            self.expect("}")
        return vars_, rows

    # ---------------------------------------------------------- expressions --
    def parse_filter_expr(self) -> Expr:
        if self.at("("):
            self.next()
            e = self.parse_expr()
            self.expect(")")
            return e
        return self.parse_expr()

    def parse_expr(self) -> Expr:
        return self.parse_or()

    def parse_or(self) -> Expr:
        left = self.parse_and()
        while self.at("||"):
            self.next()
            left = EOr(left, self.parse_and())
        return left

    def parse_and(self) -> Expr:
        left = self.parse_cmp()
        while self.at("&&"):
            self.next()
            left = EAnd(left, self.parse_cmp())
        return left

    def parse_cmp(self) -> Expr:
        left = self.parse_additive()
        t = self.peek()
        if t is not None and t.text in ("=", "!=", "<", ">", "<=", ">="):
            op = self.next().text
            right = self.parse_additive()
            return ECmp(op, left, right)
        return left

    def parse_additive(self) -> Expr:
        left = self.parse_multiplicative()
        while self.at("+") or self.at("-"):
            op = self.next().text
            left = EArith(op, left, self.parse_multiplicative())
        return left

    def parse_multiplicative(self) -> Expr:
        left = self.parse_unary()
        while self.at("*") or self.at("/"):
            op = self.next().text
            left = EArith(op, left, self.parse_unary())
        return left

    def parse_unary(self) -> Expr:
        if self.at("!"):
            self.next()
            return ENot(self.parse_unary())
        if self.at("("):
            self.next()
            e = self.parse_expr()
            self.expect(")")
            return e
        t = self.peek()
        if t is None:
            raise self.err("expected expression")
        if t.kind == "var":
            self.next()
            return EVar(t.text[1:])
        if t.kind == "number":
            self.next()
            return ELit(t.text, is_number=True)
        if t.kind == "string":
            self.next()
            return ELit(t.text)  # raw surface; resolved at compile time
        if t.kind == "iri":
            self.next()
            return ELit(t.text)
        if t.kind in ("name", "pname"):
            # function call or bare prefixed-name literal
            if self.at("(", ahead=1):
                name = self.next().text
                self.expect("(")
                args: List[Expr] = []
                while not self.at(")"):
                    args.append(self.parse_expr())
                    if self.at(","):
                        self.next()
                self.expect(")")
                return EFunc(name.upper() if name.upper() in (
                    "CONCAT", "TRIPLE", "SUBJECT", "PREDICATE", "OBJECT",
                    "ISTRIPLE", "STR", "UCASE", "LCASE",
                ) else name, args)
            self.next()
            return ELit(t.text)
        raise ParseError(f"unexpected token in expression: {t.text!r}", self.text, t.pos)

    # ---------------------------------------------------------------- update --
    def parse_update_ops(self) -> List[UpdateOperation]:
        ops: List[UpdateOperation] = []
        while True:
            if self.at_kw("INSERT"):
                self.next()
                if self.at_kw("DATA"):
                    self.next()
                    ops.append(UpdateOperation("insert_data", quads=self.parse_quad_block()))
                else:
                    tmpl = self.parse_quad_block()
                    if self.at_kw("WHERE"):
                        self.next()
                        where = self.parse_group()
                        ops.append(UpdateOperation("modify", insert_templates=tmpl, where=where))
                    else:
                        # data alias: INSERT { ... } (parser.rs data aliases)
                        ops.append(UpdateOperation("insert_data", quads=tmpl))
            elif self.at_kw("DELETE"):
                self.next()
                if self.at_kw("DATA"):
                    self.next()
                    ops.append(UpdateOperation("delete_data", quads=self.parse_quad_block()))
                elif self.at_kw("WHERE"):
                    self.next()
                    tmpl = self.parse_quad_block()
                    ops.append(UpdateOperation("delete_where", delete_templates=tmpl))
                else:
                    dt = self.parse_quad_block()
                    it: List[QuadData] = []
                    if self.at_kw("INSERT"):
                        self.next()
                        it = self.parse_quad_block()
                    if self.at_kw("WHERE"):
                        self.next()
                        where = self.parse_group()
                        ops.append(UpdateOperation(
                            "modify", delete_templates=dt, insert_templates=it, where=where))
                    else:
                        ops.append(UpdateOperation("delete_data", quads=dt))
            elif self.at_kw("CLEAR", "CREATE", "DROP"):
                kind = self.next().upper().lower()
                silent = False
                if self.at_kw("SILENT"):
                    self.next()
                    silent = True
                graph = None
                if self.at_kw("GRAPH"):
                    self.next()
                    graph = self.parse_term()
                elif self.at_kw("DEFAULT"):
                    self.next()
                    graph = None
                elif self.at_kw("NAMED"):
                    self.next()
                    graph = "NAMED"
                elif self.at_kw("ALL"):
                    self.next()
                    graph = "ALL"
                elif not self.eof() and self.peek().kind in ("iri", "pname"):
                    graph = self.parse_term()
                ops.append(UpdateOperation(kind, graph=graph, silent=silent))
            else:
                break
            if self.at(";"):
                self.next()
                continue
            break
        return ops

    def parse_quad_block(self) -> List[QuadData]:
        """`{ triples... GRAPH <g> { triples... } ... }`"""
        self.expect("{")
        quads: List[QuadData] = []
        graph: Optional[str] = None
        while not self.at("}"):
            if self.at_kw("GRAPH"):
                self.next()
                g = self.parse_term()
                self.expect("{")
                while not self.at("}"):
                    if self.at("."):
                        self.next()
                        continue
                    quads.extend(self._parse_quad_triples(g))
                self.expect("}")
            elif self.at("."):
                self.next()
            else:
                quads.extend(self._parse_quad_triples(graph))
        self.expect("}")
        return quads

    def _parse_quad_triples(self, graph: Optional[str]) -> List[QuadData]:
        out = []
        s = self.parse_term()
        while True:
            p = self.parse_term()
            while True:
                o = self.parse_term()
                out.append(QuadData(graph, s, p, o))
                if self.at(","):
                    self.next()
                    continue
                break
            if self.at(";"):
                self.next()
                continue
            break
        if self.at("."):
            self.next()
        return out

    # ------------------------------------------------------------- streaming --
    def parse_register(self) -> RegisterClause:
        self.expect_kw("REGISTER")
        stream_type = "RSTREAM"
        if self.at_kw("RSTREAM", "ISTREAM", "DSTREAM"):
            stream_type = self.next().upper()
        out_iri = self.parse_term()
        self.expect_kw("AS")
        sel = SelectQuery()
        self.expect_kw("SELECT")
        if self.at_kw("DISTINCT"):
            self.next()
            sel.distinct = True
        while True:
            t = self.peek()
            if t is None:
                break
            if t.text == "*":
                self.next()
                sel.select_star = True
            elif t.kind == "var":
                self.next()
                sel.variables.append(Projection(var=t.text[1:]))
            elif t.text == "(":
                self.next()
                sel.variables.append(self.parse_aggregate_projection(require_alias=True))
                self.expect(")")
            elif t.kind == "name" and t.upper() in AGGREGATES:
                sel.variables.append(self.parse_aggregate_projection(require_alias=False))
            else:
                break
        windows: List[WindowClause] = []
        while self.at_kw("FROM"):
            windows.append(self.parse_from_named_window())
        if self.at_kw("WHERE"):
            self.next()
            sel.where = self.parse_group()
        # trailing modifiers
        while True:
            if self.at_kw("GROUP"):
                self.next()
                self.expect_kw("BY")
                while self.peek() is not None and self.peek().kind == "var":
                    sel.group_by.append(self.next().text[1:])
            elif self.at_kw("HAVING"):
                self.next()
                sel.having = self.parse_filter_expr()
            elif self.at_kw("LIMIT"):
                self.next()
                sel.limit = int(self.next().text)
            else:
                break
        return RegisterClause(stream_type, out_iri, sel, windows)

    def parse_from_named_window(self) -> WindowClause:
        self.expect_kw("FROM")
        self.expect_kw("NAMED")
        self.expect_kw("WINDOW")
        window_iri = self.parse_term()
        self.expect_kw("ON")
        if self.at_kw("STREAM"):
            self.next()
        stream_iri = self.parse_term()
        spec = self.parse_window_spec()
        policy = None
        if self.at_kw("WITH"):
            self.next()
            self.expect_kw("POLICY")
            if self.at("("):
                # reference form (parser.rs:2737 parse_sync_policy_timeout):
                # WITH POLICY (timeout = 5s, fallback = steal|drop)
                self.next()
                self.expect_kw("TIMEOUT")
                self.expect("=")
                dur = self.parse_policy_duration()
                self.expect(",")
                self.expect_kw("FALLBACK")
                self.expect("=")
                fb = self.next().upper().capitalize()
                if fb not in ("Steal", "Drop"):
                    raise ParseError(f"bad timeout fallback {fb!r}",
                                     self.text, self.peek().pos
                                     if self.peek() else 0)
                self.expect(")")
                policy = SyncPolicy("Timeout", dur, fb)
            else:
                kind = self.next().upper()
                if kind == "TIMEOUT":
                    dur = self.parse_policy_duration()
                    policy = SyncPolicy("Timeout", dur)
                elif kind in ("STEAL", "WAIT"):
                    policy = SyncPolicy(kind.capitalize())
                else:
                    policy = SyncPolicy("Wait")
        return WindowClause(window_iri, stream_iri, spec, policy)

    def parse_window_spec(self) -> WindowSpec:
        self.expect("[")
        wt = self.next().upper()
        if wt not in ("RANGE", "TUMBLING", "SLIDING"):
            raise self.err(f"unknown window type {wt}")
        width = self.parse_duration_seconds()
        spec = WindowSpec(window_type=wt, width=width)
        while not self.at("]"):
            t = self.next()
            u = t.upper()
            if u in ("STEP", "SLIDE"):
                spec.slide = self.parse_duration_seconds()
            elif u == "REPORT":
                spec.report = self.next().upper()
            elif u == "TICK":
                spec.tick = self.next().upper()
            else:
                raise ParseError(f"unexpected token in window spec: {t.text!r}",
                                 self.text, t.pos)
        self.expect("]")
        if spec.slide is None:
            spec.slide = spec.width if spec.window_type == "TUMBLING" else spec.width
        return spec

    def parse_duration_seconds(self) -> int:
        t = self.next()
        txt = t.text
        if t.kind == "number":
            return int(float(txt))
        m = re.fullmatch(r"PT(\d+)([SMH])", txt, re.IGNORECASE)
        if m:
            v = int(m.group(1))
            unit = m.group(2).upper()
            return v * {"S": 1, "M": 60, "H": 3600}[unit]
        raise ParseError(f"bad duration {txt!r}", self.text, t.pos)

    def parse_policy_duration(self) -> int:
        """Timeout duration -> milliseconds (`5s`, `5000ms`, `PT5S`, int)."""
        t = self.next()
        txt = t.text
        m = re.fullmatch(r"PT(\d+)([SMH])", txt, re.IGNORECASE)
        if m:
            return int(m.group(1)) * {"S": 1, "M": 60, "H": 3600}[m.group(2).upper()] * 1000
        if t.kind == "number":
            num = int(float(txt))
            nxt = self.peek()
            if nxt is not None and nxt.kind == "name" and nxt.text in ("ms", "s"):
                unit = self.next().text
                return num if unit == "ms" else num * 1000
            return num * 1000
        m = re.fullmatch(r"(\d+)(ms|s)", txt)
        if m:
            return int(m.group(1)) * (1 if m.group(2) == "ms" else 1000)
        raise ParseError(f"bad policy duration {txt!r}", self.text, t.pos)

    # ----------------------------------------------------------------- rules --
    def parse_rule(self) -> CombinedRule:
        self.expect_kw("RULE")
        name_tok = self.next()
        name = name_tok.text
        head_vars: List[str] = []
        if self.at("("):
            self.next()
            while not self.at(")"):
                t = self.next()
                if t.kind == "var":
                    head_vars.append(t.text[1:])
                if self.at(","):
                    self.next()
            self.expect(")")
        prob = None
        if self.at_kw("PROB"):
            prob = self.parse_prob_annotation()
        t = self.next()
        if t.text != ":-":
            raise ParseError("expected ':-' in RULE", self.text, t.pos)
        stream_type = None
        if self.at_kw("RSTREAM", "ISTREAM", "DSTREAM"):
            stream_type = self.next().upper()
        windows: List[WindowClause] = []
        while self.at_kw("FROM"):
            windows.append(self.parse_from_named_window())
        self.expect_kw("CONSTRUCT")
        self.expect("{")
        conclusions: List[TriplePatternAst] = []
        while not self.at("}"):
            if self.at("."):
                self.next()
                continue
            conclusions.extend(self.parse_triples_block().patterns)
        self.expect("}")
        self.expect_kw("WHERE")
        body = self.parse_group()
        if self.at("."):
            self.next()
        ml_predict = None
        if self.at_kw("ML.PREDICT") or (self.at_kw("ML") and self.at(".", ahead=1)):
            ml_predict = self.parse_ml_predict()
        # pull NOT atoms out of the body into negated list
        negated: List[TriplePatternAst] = []
        body = _extract_minus(body, negated)
        return CombinedRule(
            name=name, head_vars=head_vars, stream_type=stream_type,
            windows=windows, conclusions=conclusions, body=body,
            negated=negated, prob=prob, ml_predict=ml_predict,
        )

    _HYBRID_PROB_KEYS = {"band_epsilon", "marginal_floor", "k_initial",
                         "k_max", "k_growth", "topk_budget_ms",
                         "sdd_budget_ms", "node_budget"}

    def _parse_threshold_value(self, pos) -> tuple:
        """float -> (v, "Explicit"); `auto:cost(fp=a,fn=b)` -> (fp/(fp+fn),
        "CostRatio"); everything else (incl. auto:quantile) rejected
        (ref parser.rs:2896 parse_hybrid_threshold)."""
        t = self.next().text
        if not t.lower().startswith("auto"):
            import math
            try:
                v = float(t)
            except ValueError:
                raise ParseError(f"bad threshold {t!r}", self.text, pos)
            if not math.isfinite(v):
                raise ParseError(f"bad threshold {t!r}", self.text, pos)
            return v, "Explicit"
        spec = t
        while self.at(":"):
            spec += self.next().text + self.next().text
        if not self.at("("):
            raise ParseError(f"bad hybrid threshold {spec!r}", self.text, pos)
        self.next()
        parts = []
        while not self.at(")"):
            parts.append(self.next().text)
        self.expect(")")
        body = "".join(parts)
        if not spec.lower().endswith("cost"):
            raise ParseError(f"bad hybrid threshold policy {spec!r}",
                             self.text, pos)
        costs = {}
        import math
        for pair in body.split(","):
            if "=" not in pair:
                raise ParseError("bad auto:cost spec", self.text, pos)
            k, v = pair.split("=", 1)
            k = k.strip().lower()
            if k not in ("fp", "fn") or k in costs:
                raise ParseError(f"bad auto:cost key {k!r}", self.text, pos)
            try:
                fv = float(v)
            except ValueError:
                raise ParseError(f"bad auto:cost value {v!r}", self.text, pos)
            if not math.isfinite(fv) or fv < 0.0:
                raise ParseError(f"bad auto:cost value {v!r}", self.text, pos)
            costs[k] = fv
        if set(costs) != {"fp", "fn"} or costs["fp"] + costs["fn"] <= 0.0:
            raise ParseError("auto:cost needs fp and fn with fp+fn > 0",
                             self.text, pos)
        return costs["fp"] / (costs["fp"] + costs["fn"]), "CostRatio"

    def parse_prob_annotation(self) -> ProbAnnotation:
        """PROB(key=value, ...).  Hybrid provenance is strictly validated
        (ref parser.rs:2931-3090: allowed keys only, threshold required,
        no duplicates, no confidence, auto:cost thresholds)."""
        tok = self.expect_kw("PROB")
        self.expect("(")
        ann = ProbAnnotation()
        seen = set()
        duplicate = False
        while not self.at(")"):
            key = self.next().text.lower()
            self.expect("=")
            if key in seen:
                duplicate = True
            seen.add(key)
            if key == "threshold":
                ann.threshold, policy = self._parse_threshold_value(tok.pos)
                ann.extra["threshold_policy"] = policy
                if self.at(","):
                    self.next()
                continue
            val = self.next().text
            if key in ("provenance", "combination"):
                ann.provenance = val.lower()
            elif key == "confidence":
                ann.confidence = float(val)
            else:
                ann.extra[key] = val
            if self.at(","):
                self.next()
        self.expect(")")
        if ann.provenance == "hybrid":
            unknown = (set(ann.extra) - self._HYBRID_PROB_KEYS
                       - {"threshold_policy"})
            if (duplicate or unknown or ann.confidence is not None
                    or ann.threshold is None):
                raise ParseError(
                    "invalid hybrid PROB annotation (threshold required; "
                    f"unknown/duplicate keys: {sorted(unknown)})",
                    self.text, tok.pos)
            if (ann.extra.get("threshold_policy") == "Explicit"
                    and not (0.0 <= ann.threshold <= 1.0)):
                raise ParseError("hybrid threshold must be in [0,1]",
                                 self.text, tok.pos)
        return ann

    def parse_ml_predict(self) -> dict:
        # ML.PREDICT(MODEL "name", INPUT { patterns }) — lowering alias
        name_parts = [self.next().text]
        if name_parts[0].upper() == "ML":
            self.expect(".")
            name_parts.append(self.expect_kw("PREDICT").text)
        self.expect("(")
        out: dict = {"model": None, "input": [], "output_var": None}
        while not self.at(")"):
            if self.at_kw("MODEL"):
                self.next()
                out["model"] = self.next().text.strip('"')
            elif self.at_kw("INPUT"):
                self.next()
                self.expect("{")
                while not self.at("}"):
                    if self.at("."):
                        self.next()
                        continue
                    out["input"].extend(self.parse_triples_block().patterns)
                self.expect("}")
            elif self.peek().kind == "var":
                out["output_var"] = self.next().text[1:]
            else:
                self.next()
            if self.at(","):
                self.next()
        self.expect(")")
        return out

    # ---------------------------------------------------------------- decls --
    def parse_model_decl(self) -> ModelDecl:
        self.expect_kw("MODEL")
        name = self.next().text.strip('"')
        self.expect("{")
        opts: Dict[str, str] = {}
        # ARCH MLP { HIDDEN [sizes] } OUTPUT EXCLUSIVE {labels} | BINARY {label}
        while not self.at("}"):
            if self.at_kw("ARCH"):
                self.next()
                opts["arch"] = self.next().text.upper()
                self.expect("{")
                while not self.at("}"):
                    if self.at_kw("HIDDEN"):
                        self.next()
                        self.expect("[")
                        sizes = []
                        while not self.at("]"):
                            t = self.next()
                            if t.kind == "number":
                                sizes.append(t.text)
                        self.expect("]")
                        opts["hidden"] = ",".join(sizes)
                    else:
                        self.next()
                self.expect("}")
            elif self.at_kw("OUTPUT"):
                self.next()
                kind = self.next().upper()
                opts["output"] = kind
                if self.at("{"):
                    self.next()
                    labels = []
                    while not self.at("}"):
                        t = self.next()
                        if t.kind == "string":
                            from ..storage.database import literal_lexical_value
                            labels.append(literal_lexical_value(t.text))
                        elif t.text != ",":
                            labels.append(t.text)
                    self.expect("}")
                    opts["labels"] = "\x1f".join(labels)
            else:
                self.next()
        self.expect("}")
        return ModelDecl(name=name, path="", options=opts)

    def parse_neural_relation(self) -> NeuralRelationDecl:
        self.expect_kw("NEURAL")
        self.expect_kw("RELATION")
        pred = self.parse_term()
        self.expect_kw("USING")
        self.expect_kw("MODEL")
        model = self.next().text.strip('"')
        decl = NeuralRelationDecl(name=pred, model=model)
        self.expect("{")
        while not self.at("}"):
            if self.at_kw("INPUT"):
                self.next()
                self.expect("{")
                pats = []
                while not self.at("}"):
                    if self.at("."):
                        self.next()
                        continue
                    pats.extend(self.parse_triples_block().patterns)
                self.expect("}")
                decl.options["input_patterns"] = repr(
                    [(p.s, p.p, p.o) for p in pats])
                decl.inputs = [(p.s, p.p, p.o) for p in pats]  # type: ignore
            elif self.at_kw("FEATURES"):
                self.next()
                self.expect("{")
                feats = []
                while not self.at("}"):
                    t = self.next()
                    if t.kind == "var":
                        feats.append(t.text[1:])
                self.expect("}")
                decl.options["features"] = ",".join(feats)
            else:
                self.next()
        self.expect("}")
        return decl

    def parse_train_decl(self) -> TrainNeuralRelationDecl:
        self.expect_kw("TRAIN")
        self.expect_kw("NEURAL")
        self.expect_kw("RELATION")
        name = self.parse_term()
        decl = TrainNeuralRelationDecl(name=name, target=name)
        if self.at_kw("USING"):
            self.next()
            self.expect_kw("MODEL")
            decl.options["model"] = self.next().text.strip('"')
        self.expect("{")
        while not self.at("}"):
            if self.at_kw("DATA"):
                self.next()
                if self.at("{"):
                    self.next()
                    while not self.at("}"):
                        if self.at("."):
                            self.next()
                            continue
                        decl.data_patterns.extend(self.parse_triples_block().patterns)
                    self.expect("}")
            elif self.at_kw("QUERY"):
                self.next()
                if self.at("{"):
                    self.next()
                    decl.data_query = self.parse_select_core()
                    self.expect("}")
            else:
                t = self.next()
                if t.kind in ("name", "pname") and self.at("="):
                    self.next()
                    decl.options[t.text.lower()] = self.next().text
        self.expect("}")
        return decl

    def parse_retrieve(self) -> RetrieveClause:
        self.expect_kw("RETRIEVE")
        mode = self.next().upper()
        state = self.next().upper()
        streams: List[str] = []
        while not self.eof() and self.peek().kind in ("iri", "pname"):
            streams.append(self.parse_term())
        return RetrieveClause(mode=mode, state=state, streams=streams)


def _extract_minus(g: GGP, out: List[TriplePatternAst]) -> GGP:
    if isinstance(g, GMinus):
        left = _extract_minus(g.left, out)
        if isinstance(g.right, GBgp):
            out.extend(g.right.patterns)
            return left
        return GMinus(left, g.right)
    if isinstance(g, GFilter):
        return GFilter(g.expr, _extract_minus(g.inner, out))
    if isinstance(g, GJoin):
        return GJoin(_extract_minus(g.left, out), _extract_minus(g.right, out))
    return g


# ------------------------------------------------------------------ entries --
def parse_combined_query(text: str) -> CombinedQuery:
    """Full entry: prologue + extensions + SELECT/UPDATE
    (ref parser.rs:3264 parse_combined_query)."""
    return Parser(text).parse_combined()


def parse_sparql_query(text: str) -> SelectQuery:
    """Standards-track SELECT only (ref parser.rs:1951)."""
    p = Parser(text)
    cq = p.parse_combined()
    if cq.select is None:
        raise ParseError("not a SELECT query")
    return cq.select
