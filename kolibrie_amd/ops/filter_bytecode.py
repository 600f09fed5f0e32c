"""Compile a CompiledExpr AST to the K5 postfix bytecode (must mirror the
opcode enum in csrc/kernels.hip and the torch semantics in
engine/filters.py)."""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from ..parsing.ast import (
    EAnd, EArith, ECmp, EFunc, ELit, ENot, EOr, EVar, Expr,
)

# opcodes — keep in sync with FilterOp in kernels.hip
(OP_PUSH_ID, OP_PUSH_CONST_ID, OP_PUSH_VAL, OP_PUSH_CONST_VAL,
 OP_EQ_ID, OP_NE_ID, OP_LT, OP_GT, OP_LE, OP_GE, OP_EQ_VAL, OP_NE_VAL,
 OP_ADD, OP_SUB, OP_MUL, OP_DIV, OP_AND, OP_OR, OP_NOT, OP_BOUND,
 OP_IS_TRIPLE, OP_PUSH_TRUE, OP_PUSH_FALSE) = range(23)

_ARITH = {"+": OP_ADD, "-": OP_SUB, "*": OP_MUL, "/": OP_DIV}
_CMP_VAL = {"<": OP_LT, ">": OP_GT, "<=": OP_LE, ">=": OP_GE,
            "=": OP_EQ_VAL, "!=": OP_NE_VAL}


class BytecodeError(ValueError):
    pass


class _Compiler:
    def __init__(self, var_cols: Dict[str, int]):
        self.var_cols = var_cols  # var name -> column slot
        self.ops: List[int] = []
        self.args: List[int] = []
        self.consts: List[float] = []

    def emit(self, op: int, arg: int = 0):
        self.ops.append(op)
        self.args.append(arg)

    def const_val(self, v: float) -> int:
        self.consts.append(float(v))
        return len(self.consts) - 1

    # ---- id-expression: var or literal with a term id ----------------------
    def _id_slot(self, e: Expr) -> Optional[int]:
        if isinstance(e, EVar):
            return self.var_cols.get(e.name, -1)
        return None

    def _has_id(self, e: Expr) -> bool:
        if isinstance(e, EVar):
            return True
        if isinstance(e, ELit):
            return getattr(e, "term_id", None) is not None
        return False

    def push_id(self, e: Expr):
        if isinstance(e, EVar):
            slot = self.var_cols.get(e.name)
            if slot is None:
                raise BytecodeError(f"unbound column for ?{e.name}")
            self.emit(OP_PUSH_ID, slot)
            return
        if isinstance(e, ELit):
            tid = getattr(e, "term_id", None)
            if tid is None:
                raise BytecodeError("literal without term id")
            t = tid & 0xFFFFFFFF
            t = t - 0x1_0000_0000 if t >= 0x8000_0000 else t
            self.emit(OP_PUSH_CONST_ID, t)
            return
        raise BytecodeError(f"not an id expression: {e}")

    def push_val(self, e: Expr):
        if isinstance(e, EVar):
            slot = self.var_cols.get(e.name)
            if slot is None:
                raise BytecodeError(f"unbound column for ?{e.name}")
            self.emit(OP_PUSH_VAL, slot)
            return
        if isinstance(e, ELit):
            self.emit(OP_PUSH_CONST_VAL,
                      self.const_val(getattr(e, "num_value", 0.0)))
            return
        if isinstance(e, EArith):
            self.push_val(e.left)
            self.push_val(e.right)
            self.emit(_ARITH[e.op])
            return
        raise BytecodeError(f"not a value expression: {e}")

    def push_bool(self, e: Expr):
        if isinstance(e, EAnd):
            self.push_bool(e.left)
            self.push_bool(e.right)
            self.emit(OP_AND)
            return
        if isinstance(e, EOr):
            self.push_bool(e.left)
            self.push_bool(e.right)
            self.emit(OP_OR)
            return
        if isinstance(e, ENot):
            self.push_bool(e.inner)
            self.emit(OP_NOT)
            return
        if isinstance(e, ECmp):
            if e.op in ("=", "!=") and self._has_id(e.left) and self._has_id(e.right):
                # missing columns => always-false comparison
                for side in (e.left, e.right):
                    if isinstance(side, EVar) and side.name not in self.var_cols:
                        self.emit(OP_PUSH_FALSE)
                        return
                self.push_id(e.left)
                self.push_id(e.right)
                self.emit(OP_EQ_ID if e.op == "=" else OP_NE_ID)
                return
            for side in (e.left, e.right):
                for v in _vars_of(side):
                    if v not in self.var_cols:
                        self.emit(OP_PUSH_FALSE)
                        return
            self.push_val(e.left)
            self.push_val(e.right)
            self.emit(_CMP_VAL[e.op])
            return
        if isinstance(e, EFunc):
            if e.name == "ISTRIPLE" and len(e.args) == 1 and self._has_id(e.args[0]):
                if isinstance(e.args[0], EVar) and e.args[0].name not in self.var_cols:
                    self.emit(OP_PUSH_FALSE)
                    return
                self.push_id(e.args[0])
                self.emit(OP_IS_TRIPLE)
                return
            if (e.name == "BOUND" and len(e.args) == 1
                    and isinstance(e.args[0], EVar)):
                # must match engine/filters.py _eval_bool BOUND semantics
                if e.args[0].name not in self.var_cols:
                    self.emit(OP_PUSH_FALSE)
                    return
                self.emit(OP_BOUND, self.var_cols[e.args[0].name])
                return
            # other functions are false in FILTER context
            self.emit(OP_PUSH_FALSE)
            return
        if isinstance(e, EVar):
            if e.name not in self.var_cols:
                self.emit(OP_PUSH_FALSE)
                return
            self.emit(OP_BOUND, self.var_cols[e.name])
            return
        raise BytecodeError(f"cannot compile {e} to bytecode")


def _vars_of(e: Expr) -> List[str]:
    from ..engine.filters import _collect_vars
    out: List[str] = []
    _collect_vars(e, out)
    return out


def compile_filter(ast: Expr, bindings, cache_holder=None) -> Optional[Tuple]:
    """Compile to (ops, args, consts, col_list) device tensors.

    Returns None when the expression can't be expressed in bytecode (caller
    falls back to the vectorized torch path).

    The program depends only on the AST and WHICH vars are bound, so when
    `cache_holder` is given (the CompiledExpr) the three device tensors are
    cached on it — recompiling per evaluation cost ~3 small H2D uploads
    (~12 µs) per query in the serving loop."""
    used = _vars_of(ast)
    present = tuple(v for v in used if bindings.has(v))
    dev = bindings.device
    cache = None
    if cache_holder is not None:
        cache = getattr(cache_holder, "_fb_cache", None)
        if cache is None:
            cache = cache_holder._fb_cache = {}
        hit = cache.get((present, dev))
        if hit is False:
            return None
        if hit is not None:
            ops_t, args_t, consts_t = hit
            return ops_t, args_t, consts_t, \
                [bindings.col(v).contiguous() for v in present]
    if len(present) > 16:
        if cache is not None:
            cache[(present, dev)] = False
        return None
    var_cols = {v: i for i, v in enumerate(present)}
    c = _Compiler(var_cols)
    try:
        c.push_bool(ast)
    except BytecodeError:
        if cache is not None:
            cache[(present, dev)] = False
        return None
    if len(c.ops) > 128:
        if cache is not None:
            cache[(present, dev)] = False
        return None
    ops_t = torch.tensor(c.ops, dtype=torch.int32, device=dev)
    args_t = torch.tensor(c.args, dtype=torch.int32, device=dev)
    consts_t = torch.tensor(c.consts or [0.0], dtype=torch.float64, device=dev)
    if cache is not None:
        cache[(present, dev)] = (ops_t, args_t, consts_t)
    return ops_t, args_t, consts_t, \
        [bindings.col(v).contiguous() for v in present]
