"""Provenance semirings (ref: shared/src/provenance.rs, 690 LoC).

Trait surface (provenance.rs:18-59): zero / one / plus (⊕) / times (⊗) /
negate / saturate / tag_from_probability / recover.

Implementations:
  MinMaxProbability  (:69)   ⊕=max, ⊗=min over [0,1]
  AddMultProbability (:111)  ⊕=saturating add, ⊗=multiply
  BooleanProvenance  (:153)
  TopKProofs         (:191)  k-best proof sets (proof = frozenset of seed
                             vars, k ∈ [1,63]); ⊕ = merge-truncate by proof
                             probability, ⊗ = pairwise proof union;
                             probability via inclusion-exclusion (:300-317)
  DnfWmcProvenance   (:323)  exact DNF WMC with signed literals, De Morgan
                             negation, contradiction pruning
  ExpirationProvenance (:460) ⊕=max, ⊗=min over expiry timestamps

Tag values for MinMax/AddMult are plain floats so the K6 device fixpoint
can carry them as an f32 column; the structured semirings (TopK/DNF) stay
host-side.
"""
from __future__ import annotations

import itertools
from abc import ABC, abstractmethod
from typing import Dict, FrozenSet, List, Optional, Tuple


class Provenance(ABC):
    """Semiring operations over tag values."""

    name = "provenance"

    @abstractmethod
    def zero(self): ...

    @abstractmethod
    def one(self): ...

    @abstractmethod
    def plus(self, a, b): ...

    @abstractmethod
    def times(self, a, b): ...

    def negate(self, a):
        raise NotImplementedError(f"{self.name} does not support negation")

    def saturate(self, a):
        return a

    @abstractmethod
    def tag_from_probability(self, p: float, seed_id: Optional[int] = None): ...

    @abstractmethod
    def recover(self, tag) -> float:
        """Tag -> probability-like scalar."""

    def better(self, a, b) -> bool:
        """True if a strictly improves on b (drives delta re-entry)."""
        return self.recover(a) > self.recover(b)


class MinMaxProbability(Provenance):
    name = "minmax"

    def zero(self):
        return 0.0

    def one(self):
        return 1.0

    def plus(self, a, b):
        return max(a, b)

    def times(self, a, b):
        return min(a, b)

    def negate(self, a):
        return 1.0 - a

    def tag_from_probability(self, p, seed_id=None):
        return float(p)

    def recover(self, tag):
        return float(tag)


class AddMultProbability(Provenance):
    name = "addmult"

    def zero(self):
        return 0.0

    def one(self):
        return 1.0

    def plus(self, a, b):
        # noisy-or disjunction (ref provenance.rs:119 AddMultProbability)
        return a + b - a * b

    def times(self, a, b):
        return a * b

    def tag_from_probability(self, p, seed_id=None):
        return float(min(1.0, max(0.0, p)))

    def recover(self, tag):
        return float(tag)


class BooleanProvenance(Provenance):
    name = "boolean"

    def zero(self):
        return False

    def one(self):
        return True

    def plus(self, a, b):
        return a or b

    def times(self, a, b):
        return a and b

    def negate(self, a):
        return not a

    def tag_from_probability(self, p, seed_id=None):
        return p > 0.0

    def recover(self, tag):
        return 1.0 if tag else 0.0


class ExpirationProvenance(Provenance):
    """Per-fact expiry propagation (ref provenance.rs:460): a derivation
    lives while all premises live (⊗=min); the best derivation survives
    (⊕=max)."""
    name = "expiration"

    def zero(self):
        return float("-inf")

    def one(self):
        return float("inf")

    def plus(self, a, b):
        return max(a, b)

    def times(self, a, b):
        return min(a, b)

    def tag_from_probability(self, p, seed_id=None):
        return float(p)

    def recover(self, tag):
        return float(tag)


Proof = FrozenSet[int]  # set of seed variable ids


class TopKProofs(Provenance):
    """k-best proof sets (ref provenance.rs:191-321).

    Tag = tuple of proofs sorted by descending probability, length <= k.
    Each seed var has a probability in `weights`.
    """
    name = "topk"

    def __init__(self, k: int = 8, weights: Optional[Dict[int, float]] = None):
        if not (1 <= k <= 63):
            raise ValueError("k must be in [1, 63]")
        self.k = k
        self.weights: Dict[int, float] = weights or {}

    def proof_probability(self, proof: Proof) -> float:
        p = 1.0
        for v in proof:
            p *= self.weights.get(v, 1.0)
        return p

    def zero(self):
        return ()

    def one(self):
        return (frozenset(),)

    def _truncate(self, proofs) -> Tuple[Proof, ...]:
        uniq: List[Proof] = []
        seen = set()
        for pr in sorted(proofs, key=lambda q: (-self.proof_probability(q), sorted(q))):
            if pr not in seen:
                # subsumption: a proof superset of a retained one adds nothing
                if any(kept <= pr for kept in uniq):
                    continue
                seen.add(pr)
                uniq.append(pr)
            if len(uniq) >= self.k:
                break
        return tuple(uniq)

    def plus(self, a, b):
        return self._truncate(list(a) + list(b))

    def times(self, a, b):
        return self._truncate(p1 | p2 for p1 in a for p2 in b)

    def negate(self, a):
        # approximate synthetic-seed negation (ref provenance.rs:259-275)
        p = self.recover(a)
        sid = -(abs(hash(a)) % (1 << 30)) - 1
        self.weights[sid] = max(0.0, 1.0 - p)
        return (frozenset([sid]),)

    def tag_from_probability(self, p, seed_id=None):
        if seed_id is None:
            seed_id = len(self.weights) + 1
        self.weights[seed_id] = float(p)
        return (frozenset([seed_id]),)

    def recover(self, tag) -> float:
        """Inclusion-exclusion over the retained proofs (ref :300-317).
        Beyond 16 proofs exact inclusion-exclusion is infeasible (2^n
        terms); fall back to the independence approximation
        1 - prod(1 - p_i) (a lower bound for monotone DNFs)."""
        proofs = list(tag)
        if not proofs:
            return 0.0
        n = len(proofs)
        if n > 16:
            q = 1.0
            for pr in proofs:
                q *= 1.0 - self.proof_probability(pr)
            return min(1.0, max(0.0, 1.0 - q))
        total = 0.0
        for r in range(1, n + 1):
            for combo in itertools.combinations(proofs, r):
                union: FrozenSet[int] = frozenset().union(*combo)
                total += ((-1) ** (r + 1)) * self.proof_probability(union)
        return min(1.0, max(0.0, total))


Literal = int  # +var / -var signed literal
Clause = FrozenSet[Literal]


class DnfWmcProvenance(Provenance):
    """Exact DNF weighted model counting (ref provenance.rs:323-458).

    Tag = frozenset of clauses (each a frozenset of signed literals).
    negate via De Morgan (product of negated clauses expanded back to DNF,
    contradiction-pruned); recover = exact WMC via inclusion-exclusion on
    clauses (independent variables).
    """
    name = "wmc"

    def __init__(self, weights: Optional[Dict[int, float]] = None,
                 max_clauses: int = 4096):
        self.weights: Dict[int, float] = weights or {}
        self.max_clauses = max_clauses

    def zero(self):
        return frozenset()

    def one(self):
        return frozenset([frozenset()])

    @staticmethod
    def _consistent(clause: Clause) -> bool:
        return not any(-lit in clause for lit in clause)

    def plus(self, a, b):
        out = set(a) | set(b)
        if len(out) > self.max_clauses:
            raise OverflowError("DNF clause budget exceeded")
        return frozenset(out)

    def times(self, a, b):
        out = set()
        for c1 in a:
            for c2 in b:
                merged = c1 | c2
                if self._consistent(merged):
                    out.add(merged)
        if len(out) > self.max_clauses:
            raise OverflowError("DNF clause budget exceeded")
        return frozenset(out)

    def negate(self, a):
        # De Morgan: ¬(C1 ∨ C2 ∨ ...) = ¬C1 ∧ ¬C2 ∧ ... ; each ¬Ci is a
        # disjunction of negated literals; expand the product back to DNF
        # with contradiction pruning.
        result = self.one()
        for clause in a:
            disj = frozenset(frozenset([-lit]) for lit in clause)
            result = self.times(result, disj)
        return result

    def tag_from_probability(self, p, seed_id=None):
        if seed_id is None:
            seed_id = len(self.weights) + 1
        self.weights[seed_id] = float(p)
        return frozenset([frozenset([seed_id])])

    def _lit_prob(self, lit: Literal) -> float:
        w = self.weights.get(abs(lit), 1.0)
        return w if lit > 0 else 1.0 - w

    def recover(self, tag) -> float:
        clauses = [c for c in tag if self._consistent(c)]
        if not clauses:
            return 0.0
        if frozenset() in clauses:
            return 1.0
        n = len(clauses)
        if n > 16:
            # exact WMC beyond 16 clauses goes through the SDD engine
            from .sdd import SddManager
            m = SddManager()
            for c in clauses:
                for lit in c:
                    if abs(lit) not in m.var_order:
                        m.declare_var(abs(lit),
                                      pos_weight=self.weights.get(abs(lit), 1.0))
            node = m.false_node()
            for c in clauses:
                term = m.true_node()
                for lit in c:
                    term = m.conjoin(term, m.literal(abs(lit), lit > 0))
                node = m.disjoin(node, term)
            return m.wmc(node)
        total = 0.0
        for r in range(1, n + 1):
            for combo in itertools.combinations(clauses, r):
                merged = frozenset().union(*combo)
                if not self._consistent(merged):
                    continue
                prod = 1.0
                for lit in merged:
                    prod *= self._lit_prob(lit)
                total += ((-1) ** (r + 1)) * prod
        return min(1.0, max(0.0, total))


def semiring_by_name(name: str, **kw) -> Provenance:
    name = name.lower()
    if name in ("minmax", "min"):
        return MinMaxProbability()
    if name in ("addmult", "independent", "add"):
        return AddMultProbability()
    if name in ("boolean", "bool"):
        return BooleanProvenance()
    if name == "topk":
        return TopKProofs(**kw)
    if name == "wmc":
        return DnfWmcProvenance(**kw)
    if name == "expiration":
        return ExpirationProvenance()
    raise ValueError(f"unknown provenance semiring {name!r}")
