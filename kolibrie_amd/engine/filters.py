"""Compiled FILTER / BIND expressions.

Semantics preserved exactly from the reference
(streamertail_optimizer/execution/types.rs:373-460):
  - var-var `=` / `!=` compare raw dictionary IDs (types.rs:396-398)
  - ordering ops (< > <= >=) compare float64 values, where a term's value is
    `parse::<f64>().unwrap_or(0.0)` — non-numeric => 0.0 (types.rs:349-359);
    we pre-parse once into the device value column so kernels never see
    strings
  - var-const `=` is lexical string equality == ID equality (dictionary is
    injective)
  - unbound variable => false
  - AND / OR / NOT; arithmetic with division-by-zero => false
  - FILTER function calls support isTRIPLE (types.rs:444-455)

The vectorized evaluator below is the CPU oracle and the torch fallback; the
HIP K5 kernel consumes the same tree serialized to postfix bytecode
(ops/filter_bytecode.py).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ..parsing.ast import (
    EAnd, EArith, ECmp, EFunc, ELit, ENot, EOr, EVar, Expr,
)
from ..storage.terms import UNBOUND
from .bindings import Bindings


class CompiledExpr:
    """Filter predicate over a Bindings table."""

    def __init__(self, ast: Expr, db, prefixes: Optional[Dict[str, str]] = None):
        self.ast = ast
        self.prefixes = prefixes or {}
        # encode literal terms now (dictionary writes happen at compile time,
        # mirroring utils.rs compile_term)
        self._prepare(ast, db)

    def _prepare(self, e: Expr, db):
        if isinstance(e, ELit):
            if e.is_number:
                e.num_value = float(e.value)  # type: ignore[attr-defined]
                # lexical id if interned (for = comparisons)
                e.term_id = db.dictionary.lookup(e.value)  # type: ignore[attr-defined]
            else:
                lex = db.resolve_lexical(e.value, self.prefixes)
                e.term_id = db.dictionary.encode(lex)  # type: ignore[attr-defined]
                try:
                    e.num_value = float(lex)  # type: ignore[attr-defined]
                except ValueError:
                    e.num_value = 0.0  # type: ignore[attr-defined]
        for child in _children(e):
            self._prepare(child, db)

    # -------------------------------------------------------------- evaluate
    def eval_mask(self, b: Bindings, db) -> torch.Tensor:
        if b.n > 0 and b.device.type == "cuda":
            from ..ops import native_for
            from ..ops.filter_bytecode import compile_filter
            probe_col = next(iter(b.cols.values())) if b.cols else None
            native = native_for(probe_col) if probe_col is not None else None
            if native is not None:
                prog = compile_filter(self.ast, b, cache_holder=self)
                if prog is not None:
                    ops_t, args_t, consts_t, col_list = prog
                    return native.filter_bytecode(
                        ops_t, args_t, consts_t, col_list,
                        db.value_column(), b.n)
        return _eval_bool(self.ast, b, db)

    def variables(self) -> List[str]:
        out: List[str] = []
        _collect_vars(self.ast, out)
        return out


def _children(e: Expr):
    if isinstance(e, (EAnd, EOr)):
        return (e.left, e.right)
    if isinstance(e, ECmp):
        return (e.left, e.right)
    if isinstance(e, EArith):
        return (e.left, e.right)
    if isinstance(e, ENot):
        return (e.inner,)
    if isinstance(e, EFunc):
        return tuple(e.args)
    return ()


def _collect_vars(e: Expr, out: List[str]):
    if isinstance(e, EVar):
        if e.name not in out:
            out.append(e.name)
    for c in _children(e):
        _collect_vars(c, out)


def _ids_of(e: Expr, b: Bindings) -> Optional[torch.Tensor]:
    """Raw ID column for an expression if it denotes a term (var/literal)."""
    if isinstance(e, EVar):
        if b.has(e.name):
            return b.col(e.name)
        return torch.full((b.n,), UNBOUND, dtype=torch.int32, device=b.device)
    if isinstance(e, ELit):
        tid = getattr(e, "term_id", None)
        if tid is None:
            return None
        t = tid - 0x1_0000_0000 if tid >= 0x8000_0000 else tid
        return torch.full((b.n,), t, dtype=torch.int32, device=b.device)
    return None


def _values_of(e: Expr, b: Bindings, db) -> torch.Tensor:
    """float64 value column of an expression (non-numeric => 0.0)."""
    if isinstance(e, EVar):
        if not b.has(e.name):
            return torch.zeros(b.n, dtype=torch.float64, device=b.device)
        ids = b.col(e.name).to(torch.int64) & 0xFFFFFFFF
        vc = db.value_column()
        if vc.numel() == 0:
            return torch.zeros(b.n, dtype=torch.float64, device=b.device)
        # ids outside the interned vocabulary have no lexical form:
        # value 0.0, exactly like the K5 kernel's `u < value_n` guard
        # (clamping to the last entry borrowed ITS value — wrong)
        in_range = ids < vc.numel()
        vals = vc[torch.clamp(ids, max=vc.numel() - 1)]
        return torch.where(in_range, vals,
                           torch.zeros((), dtype=torch.float64,
                                       device=b.device))
    if isinstance(e, ELit):
        return torch.full((b.n,), getattr(e, "num_value", 0.0),
                          dtype=torch.float64, device=b.device)
    if isinstance(e, EArith):
        lv = _values_of(e.left, b, db)
        rv = _values_of(e.right, b, db)
        if e.op == "+":
            return lv + rv
        if e.op == "-":
            return lv - rv
        if e.op == "*":
            return lv * rv
        if e.op == "/":
            out = torch.where(rv != 0, lv / torch.where(rv == 0, torch.ones_like(rv), rv),
                              torch.full_like(lv, float("nan")))
            return out
        raise ValueError(f"unknown arithmetic op {e.op}")
    raise ValueError(f"expression has no numeric value: {e}")


def _bound_mask(e: Expr, b: Bindings) -> torch.Tensor:
    """True where every variable referenced by e is bound."""
    mask = torch.ones(b.n, dtype=torch.bool, device=b.device)
    vars_: List[str] = []
    _collect_vars(e, vars_)
    for v in vars_:
        if b.has(v):
            mask &= b.col(v) != UNBOUND
        else:
            mask &= False
    return mask


def _eval_bool(e: Expr, b: Bindings, db) -> torch.Tensor:
    dev = b.device
    if isinstance(e, EAnd):
        return _eval_bool(e.left, b, db) & _eval_bool(e.right, b, db)
    if isinstance(e, EOr):
        return _eval_bool(e.left, b, db) | _eval_bool(e.right, b, db)
    if isinstance(e, ENot):
        return ~_eval_bool(e.inner, b, db)
    if isinstance(e, ECmp):
        bound = _bound_mask(e.left, b) & _bound_mask(e.right, b)
        if e.op in ("=", "!="):
            li = _ids_of(e.left, b)
            ri = _ids_of(e.right, b)
            if li is not None and ri is not None:
                eq = li == ri
                res = eq if e.op == "=" else ~eq
                return res & bound
            # arithmetic equality falls back to values
            lv = _values_of(e.left, b, db)
            rv = _values_of(e.right, b, db)
            eq = (lv == rv) & ~torch.isnan(lv) & ~torch.isnan(rv)
            return (eq if e.op == "=" else ~eq) & bound
        lv = _values_of(e.left, b, db)
        rv = _values_of(e.right, b, db)
        ok = ~torch.isnan(lv) & ~torch.isnan(rv)
        if e.op == "<":
            return (lv < rv) & bound & ok
        if e.op == ">":
            return (lv > rv) & bound & ok
        if e.op == "<=":
            return (lv <= rv) & bound & ok
        if e.op == ">=":
            return (lv >= rv) & bound & ok
        raise ValueError(f"unknown comparison {e.op}")
    if isinstance(e, EFunc):
        if e.name == "ISTRIPLE":
            arg = e.args[0]
            ids = _ids_of(arg, b)
            if ids is None:
                return torch.zeros(b.n, dtype=torch.bool, device=dev)
            # quoted IDs have bit 31 set => negative int32 (and != UNBOUND)
            return (ids < 0) & (ids != UNBOUND)
        if e.name == "BOUND" and e.args and isinstance(e.args[0], EVar):
            # BOUND(?v) — matches the K5 bytecode OP_BOUND semantics
            # (engine extension used with OPTIONAL; UNDEF-aware)
            v = e.args[0].name
            if not b.has(v):
                return torch.zeros(b.n, dtype=torch.bool, device=dev)
            return b.col(v) != UNBOUND
        # other functions are false in FILTER context (ref types.rs:444-455)
        return torch.zeros(b.n, dtype=torch.bool, device=dev)
    if isinstance(e, EVar):
        # bare variable: effective boolean value — bound and != "false"
        if not b.has(e.name):
            return torch.zeros(b.n, dtype=torch.bool, device=dev)
        return b.col(e.name) != UNBOUND
    raise ValueError(f"cannot evaluate {e} as boolean")


# --------------------------------------------------------------------- BIND --
class CompiledBind:
    """BIND(expr AS ?v) evaluator (ref engine.rs:527-677: CONCAT / UDF /
    TRIPLE / SUBJECT / PREDICATE / OBJECT / isTRIPLE; we add arithmetic).

    String-producing functions run on host (strings never touch the GPU);
    TRIPLE/SUBJECT/... use the quoted-triple store.
    """

    def __init__(self, ast: Expr, db, prefixes: Optional[Dict[str, str]] = None):
        self.ast = ast
        self.prefixes = prefixes or {}
        CompiledExpr(ast, db)  # prepares literal ids/values in-place
        # SUBJECT/PREDICATE/OBJECT of a non-triple yield UNBOUND cells
        self.may_produce_unbound = (
            isinstance(ast, EFunc)
            and ast.name in ("SUBJECT", "PREDICATE", "OBJECT"))

    def eval_ids(self, b: Bindings, db) -> torch.Tensor:
        e = self.ast
        return self._eval_term_ids(e, b, db)

    def _eval_term_ids(self, e: Expr, b: Bindings, db) -> torch.Tensor:
        dev = b.device
        if isinstance(e, (EVar, ELit)):
            ids = _ids_of(e, b)
            if ids is None:  # pure number literal not interned
                tid = db.dictionary.encode(_fmt_num(getattr(e, "num_value", 0.0)))
                t = tid - 0x1_0000_0000 if tid >= 0x8000_0000 else tid
                return torch.full((b.n,), t, dtype=torch.int32, device=dev)
            return ids
        if isinstance(e, EArith):
            vals = _values_of(e, b, db).cpu().numpy()
            out = [db.dictionary.encode(_fmt_num(v)) for v in vals]
            return _ids_list_to_tensor(out, dev)
        if isinstance(e, EFunc):
            name = e.name
            if name == "CONCAT":
                parts = [self._decode_col(a, b, db) for a in e.args]
                out = [db.dictionary.encode("".join(p[i] for p in parts))
                       for i in range(b.n)]
                return _ids_list_to_tensor(out, dev)
            if name == "TRIPLE":
                s_ids = self._eval_term_ids(e.args[0], b, db).cpu().tolist()
                p_ids = self._eval_term_ids(e.args[1], b, db).cpu().tolist()
                o_ids = self._eval_term_ids(e.args[2], b, db).cpu().tolist()
                out = [db.quoted_triples.encode(s & 0xFFFFFFFF, p & 0xFFFFFFFF, o & 0xFFFFFFFF)
                       for s, p, o in zip(s_ids, p_ids, o_ids)]
                return _ids_list_to_tensor(out, dev)
            if name in ("SUBJECT", "PREDICATE", "OBJECT"):
                pos = {"SUBJECT": 0, "PREDICATE": 1, "OBJECT": 2}[name]
                ids = self._eval_term_ids(e.args[0], b, db).cpu().tolist()
                out = []
                for x in ids:
                    t = db.quoted_triples.decode(x & 0xFFFFFFFF)
                    out.append(t[pos] if t is not None else 0xFFFFFFFF)
                return _ids_list_to_tensor(out, dev)
            if name == "ISTRIPLE":
                ids = self._eval_term_ids(e.args[0], b, db)
                is_t = (ids < 0) & (ids != UNBOUND)
                t_id = db.dictionary.encode("true")
                f_id = db.dictionary.encode("false")
                return torch.where(
                    is_t,
                    torch.full((b.n,), t_id, dtype=torch.int32, device=dev),
                    torch.full((b.n,), f_id, dtype=torch.int32, device=dev),
                )
            if name in ("UCASE", "LCASE", "STR"):
                vals = self._decode_col(e.args[0], b, db)
                fn = {"UCASE": str.upper, "LCASE": str.lower, "STR": lambda s: s}[name]
                out = [db.dictionary.encode(fn(v)) for v in vals]
                return _ids_list_to_tensor(out, dev)
            # UDF (ref sparql_database.rs:2130 register_udf)
            udf = db.udfs.get(name.upper())
            if udf is not None:
                arg_cols = [self._decode_col(a, b, db) for a in e.args]
                out = [db.dictionary.encode(str(udf(*(c[i] for c in arg_cols))))
                       for i in range(b.n)]
                return _ids_list_to_tensor(out, dev)
            raise ValueError(f"unknown BIND function {name}")
        raise ValueError(f"cannot BIND expression {e}")

    def _decode_col(self, e: Expr, b: Bindings, db) -> List[str]:
        ids = self._eval_term_ids(e, b, db).cpu().tolist()
        return [db.decode_term(x & 0xFFFFFFFF) or "" for x in ids]


def _fmt_num(v: float) -> str:
    if v == int(v) and abs(v) < 1e15:
        return str(int(v))
    return repr(v)


def _ids_list_to_tensor(ids: List[int], device) -> torch.Tensor:
    conv = [(x & 0xFFFFFFFF) for x in ids]
    conv = [x - 0x1_0000_0000 if x >= 0x8000_0000 else x for x in conv]
    return torch.tensor(conv, dtype=torch.int32, device=device)
