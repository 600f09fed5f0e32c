"""Core term model: dictionary IDs, variables, patterns.

MI355X-first design: on device, every RDF term is a 32-bit dictionary ID and
solution sequences are *columnar* int32 row tables (struct-of-arrays), never
per-row hash maps.  Strings live only on the host.

Reference parity: shared/src/terms.rs:14-42 (Term, TriplePattern, Bindings),
shared/src/quoted_triple_store.rs:17 (QUOTED_TRIPLE_ID_BIT).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple, Union

# ID space -------------------------------------------------------------------
# Plain terms get IDs in [1, 2^31).  RDF-star quoted triples get IDs with bit
# 31 set (ref: quoted_triple_store.rs:17 QUOTED_TRIPLE_ID_BIT = 0x8000_0000).
# On device we store IDs as int32 (two's complement), so quoted IDs are the
# negative range; UNBOUND is the reserved all-ones pattern 0xFFFF_FFFF == -1,
# which the quoted-ID allocator never reaches.
QUOTED_TRIPLE_ID_BIT = 0x8000_0000
UNBOUND_U32 = 0xFFFF_FFFF
UNBOUND = -1  # int32 view of UNBOUND_U32
NULL_ID = 0   # dictionary slot 0 is reserved (empty string / default graph)


def u32_to_i32(x: int) -> int:
    """Canonical u32 dictionary ID -> int32 device representation."""
    x &= 0xFFFFFFFF
    return x - 0x1_0000_0000 if x >= 0x8000_0000 else x


def i32_to_u32(x: int) -> int:
    """int32 device value -> canonical u32 dictionary ID."""
    return x & 0xFFFFFFFF


def is_quoted_id(term_id: int) -> bool:
    """True if this u32 ID denotes an RDF-star quoted triple."""
    return bool(i32_to_u32(term_id) & QUOTED_TRIPLE_ID_BIT) and i32_to_u32(term_id) != UNBOUND_U32


@dataclass(frozen=True)
class Variable:
    """A SPARQL variable, e.g. ?x (name stored without the '?')."""
    name: str

    def __repr__(self):
        return f"?{self.name}"


@dataclass(frozen=True)
class Constant:
    """A dictionary-encoded constant term (u32 ID)."""
    id: int

    def __repr__(self):
        return f"#{self.id}"


@dataclass(frozen=True)
class QuotedTriplePattern:
    """An RDF-star quoted triple appearing in a pattern position: << s p o >>."""
    s: "Term"
    p: "Term"
    o: "Term"


Term = Union[Variable, Constant, QuotedTriplePattern]


@dataclass(frozen=True)
class TriplePattern:
    """One triple pattern of a BGP (ref: shared/src/terms.rs TriplePattern)."""
    s: Term
    p: Term
    o: Term

    def terms(self) -> Tuple[Term, Term, Term]:
        return (self.s, self.p, self.o)

    def variables(self):
        out = []
        for t in self.terms():
            _collect_vars(t, out)
        return out


def _collect_vars(t: Term, out: list):
    if isinstance(t, Variable):
        if t.name not in out:
            out.append(t.name)
    elif isinstance(t, QuotedTriplePattern):
        for sub in (t.s, t.p, t.o):
            _collect_vars(sub, out)


@dataclass(frozen=True)
class Triple:
    """A concrete dictionary-encoded triple (u32 IDs).

    Ref: shared/src/triple.rs:14-18 — Triple{subject,predicate,object: u32}.
    """
    s: int
    p: int
    o: int

    def as_tuple(self) -> Tuple[int, int, int]:
        return (self.s, self.p, self.o)


@dataclass(frozen=True)
class Quad:
    """A triple in a named graph (graph=NULL_ID means the default graph)."""
    g: int
    s: int
    p: int
    o: int
