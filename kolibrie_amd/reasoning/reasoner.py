"""Reasoner facade (ref: datalog/src/reasoning.rs:33-186 — owns dictionary,
rules, fact store, RuleIndex, constraints, probability seeds;
add_abox_triple :96, query_abox :116, violates_constraints :137,
compute_repairs :148-186)."""
from __future__ import annotations

from itertools import combinations
from typing import Dict, List, Optional, Set, Tuple

import torch

from ..storage.dictionary import Dictionary, QuotedTripleStore
from ..storage.terms import Constant, TriplePattern, Variable
from .rule import Rule, RuleIndex
from .seminaive import FactStore, infer_fixpoint


class _MiniDb:
    """Just enough SparqlDatabase surface for filter eval inside rules."""

    def __init__(self, dictionary: Dictionary, device):
        self.dictionary = dictionary
        self.quoted_triples = QuotedTripleStore()
        self.device = torch.device(device)
        self._value_col_cache = None
        self._value_col_len = 0
        self.prefixes: Dict[str, str] = {}
        self.udfs: Dict[str, object] = {}

    def resolve_lexical(self, term: str, prefixes=None) -> str:
        t = term.strip()
        if t.startswith("<") and t.endswith(">"):
            return t[1:-1]
        if t[:1] in "\"'":
            from ..storage.database import literal_lexical_value
            return literal_lexical_value(t)
        if prefixes and ":" in t:
            pre, local = t.split(":", 1)
            if pre in prefixes:
                return prefixes[pre] + local
        return t

    def value_column(self):
        n = len(self.dictionary)
        if self._value_col_cache is None or self._value_col_len < n:
            import numpy as np
            arr = self.dictionary.values_array()
            self._value_col_cache = torch.from_numpy(arr).to(self.device)
            self._value_col_len = n
        return self._value_col_cache

    def quoted_columns(self):
        e = torch.empty(0, dtype=torch.int32, device=self.device)
        return e, e.clone(), e.clone()


class Reasoner:
    """Datalog engine over device-resident fact columns."""

    def __init__(self, device: str = "cpu", dictionary: Optional[Dictionary] = None):
        self.device = torch.device(device)
        self.dictionary = dictionary if dictionary is not None else Dictionary()
        self.db = _MiniDb(self.dictionary, device)
        self.facts = FactStore(device)
        self.rules: List[Rule] = []
        self.rule_index = RuleIndex()
        self.constraints: List[Rule] = []
        self.probability_seeds: Dict[Tuple[int, int, int], float] = {}
        self._pending: List[Tuple[int, int, int]] = []
        self._base_count = 0  # facts present before inference

    # ------------------------------------------------------------- facts --
    def _i32(self, x: int) -> int:
        x &= 0xFFFFFFFF
        return x - 0x1_0000_0000 if x >= 0x8000_0000 else x

    def add_tbox_triple(self, s: str, p: str, o: str):
        """Schema-level assertion — stored in the same fact store (the
        reference keeps one store too; ref reasoning.rs add_tbox_triple)."""
        self.add_abox_triple(s, p, o)

    def add_abox_triple(self, s: str, p: str, o: str):
        self._pending.append((
            self._i32(self.dictionary.encode(s)),
            self._i32(self.dictionary.encode(p)),
            self._i32(self.dictionary.encode(o)),
        ))

    def add_fact_ids(self, s: int, p: int, o: int):
        self._pending.append((self._i32(s), self._i32(p), self._i32(o)))

    def add_fact_columns(self, s, p, o):
        self._flush()
        self.facts.add_columns(s, p, o)

    def _flush(self):
        if not self._pending:
            return
        import numpy as np
        arr = np.asarray(self._pending, dtype=np.int32).reshape(-1, 3)
        t = torch.from_numpy(arr).to(self.device)
        from ..engine.tensor_utils import membership_mask, unique_rows
        s, p, o = unique_rows([t[:, 0], t[:, 1], t[:, 2]])
        if self.facts.n:
            known = self.facts.sorted_unique_rows()
            hit = membership_mask([s, p, o], known)
            s, p, o = s[~hit], p[~hit], o[~hit]
        self.facts.add_columns(s, p, o)
        self._pending.clear()

    # ------------------------------------------------------------- rules --
    def encode_term(self, term: str) -> int:
        """Intern a raw term string, returning its u32 id (reference
        PyKnowledgeGraph.encode_term, py_knowledge_graph.rs:232) — for
        building Constant terms in hand-constructed rules."""
        return self.dictionary.encode(term)

    def add_rule(self, rule: Rule):
        self.rules.append(rule)
        self.rule_index.add_rule(rule)

    def add_rule_text(self, text: str, prefixes: Optional[Dict[str, str]] = None):
        """Parse `RULE ... :- CONSTRUCT {..} WHERE {..}` text."""
        from ..parsing.sparql import parse_combined_query
        from .rule import convert_combined_rule
        cq = parse_combined_query(text)
        pfx = dict(cq.prefixes)
        if prefixes:
            pfx.update(prefixes)
        rdb = _RuleDb(self)
        for cr in cq.rules:
            self.add_rule(convert_combined_rule(cr, rdb, pfx))

    def add_constraint(self, rule: Rule):
        self.constraints.append(rule)

    def infer_new_facts_semi_naive_with_repairs(self) -> int:
        """Semi-naive fixpoint, then drop facts participating in minimal
        constraint repairs (ref reasoning.rs
        infer_new_facts_semi_naive_with_repairs)."""
        n = self.infer_new_facts_semi_naive()
        repairs = self.compute_repairs()
        removed = 0
        for rep in repairs[:1]:   # apply one minimal repair, ref behavior
            for (fs, fp, fo) in rep:
                self.facts.remove(self._i32(fs), self._i32(fp), self._i32(fo))
                removed += 1
        return n - removed

    # --------------------------------------------------------- inference --
    def infer_new_facts(self) -> int:
        """Naive materialisation (ref my_naive.rs:79)."""
        self._flush()
        self._base_count = self.facts.n if self._base_count == 0 else self._base_count
        return infer_fixpoint(self.rules, self.facts, self.db, semi_naive=False)

    def infer_new_facts_semi_naive(self) -> int:
        self._flush()
        self._base_count = self.facts.n if self._base_count == 0 else self._base_count
        return infer_fixpoint(self.rules, self.facts, self.db, semi_naive=True)

    # alias (ref semi_naive_parallel) — on device, parallel IS the default
    infer_new_facts_semi_naive_parallel = infer_new_facts_semi_naive

    # ------------------------------------------------------------- query --
    def query_abox(self, s: Optional[str] = None, p: Optional[str] = None,
                   o: Optional[str] = None) -> List[Tuple[str, str, str]]:
        self._flush()
        from ..engine.scan import scan_unit
        consts = {}
        for i, t in enumerate((s, p, o)):
            if t is not None:
                tid = self.dictionary.lookup(t)
                if tid is None:
                    return []
                consts[i] = self._i32(tid)
        cs, cp, co = scan_unit(self.facts.index(), consts)
        out = []
        d = self.dictionary
        for a, b, c in zip(
            (cs.to(torch.int64) & 0xFFFFFFFF).cpu().tolist(),
            (cp.to(torch.int64) & 0xFFFFFFFF).cpu().tolist(),
            (co.to(torch.int64) & 0xFFFFFFFF).cpu().tolist(),
        ):
            out.append((d.decode(a) or "", d.decode(b) or "", d.decode(c) or ""))
        return sorted(out)

    def all_fact_tuples(self) -> Set[Tuple[int, int, int]]:
        self._flush()
        s = (self.facts.s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        p = (self.facts.p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        o = (self.facts.o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        return set(zip(s, p, o))

    def contains_fact(self, s: str, p: str, o: str) -> bool:
        ids = [self.dictionary.lookup(x) for x in (s, p, o)]
        if any(x is None for x in ids):
            return False
        self._flush()
        return self.facts.index().contains(*[i & 0xFFFFFFFF for i in ids])  # type: ignore

    # ------------------------------------------------ provenance fixpoints --
    def infer_new_facts_with_provenance(self, semiring=None,
                                        seeds: Optional[Dict] = None):
        """Tagged materialisation (ref provenance_semi_naive.rs:210).

        `seeds`: {(s,p,o) id triple: probability}; defaults to
        `probability_seeds`; facts without a seed get one().  Scalar
        semirings (minmax/addmult/expiration) run the device-tagged K6
        path; structured semirings (TopK, DNF-WMC, SDD) run the host
        oracle.  Returns {triple: tag}.
        """
        from .provenance import MinMaxProbability, Provenance
        self._flush()
        if semiring is None:
            semiring = MinMaxProbability()
        if isinstance(semiring, str):
            from .device_tags import ScalarSemiring
            try:
                sr = ScalarSemiring(semiring)
            except ValueError:
                from .provenance import semiring_by_name
                semiring = semiring_by_name(semiring)
            else:
                return self._infer_device_tags(sr, seeds)
        if getattr(semiring, "name", "") in ("minmax", "addmult", "expiration"):
            from .device_tags import ScalarSemiring
            return self._infer_device_tags(ScalarSemiring(semiring.name), seeds)
        seed_map = self._seed_tags(semiring, seeds)
        from .provenance_fixpoint import infer_with_provenance
        return infer_with_provenance(self.rules, seed_map, semiring, self.db)

    def _seed_tags(self, semiring, seeds):
        raw = seeds if seeds is not None else self.probability_seeds
        seed_map = {}
        # deterministic seed-id assignment: sorted by triple
        # (ref provenance_semi_naive.rs:220-228)
        for t in sorted(self.all_fact_tuples()):
            p = raw.get(t)
            if p is None:
                seed_map[t] = semiring.one()
            else:
                seed_map[t] = semiring.tag_from_probability(p)
        return seed_map

    def _infer_device_tags(self, sr, seeds):
        from .device_tags import infer_with_provenance_device
        raw = seeds if seeds is not None else self.probability_seeds
        seed_map = {t: float(raw.get(t, 1.0))
                    for t in sorted(self.all_fact_tuples())}
        return infer_with_provenance_device(
            self.rules, seed_map, sr, device=str(self.device), db=self.db)

    def infer_with_sdd_seeds(self, seeds: Optional[Dict] = None):
        """Semi-naive with SddProvenance initial tags (ref
        sdd_seed_materialise.rs): returns ({triple: sdd node}, manager)."""
        from .sdd import SddProvenance
        prov = SddProvenance()
        seed_map = self._seed_tags(prov, seeds)
        from .provenance_fixpoint import infer_with_provenance
        tags = infer_with_provenance(self.rules, seed_map, prov, self.db)
        return tags, prov

    # ---------------------------------------------------------- IAR queries --
    def query_with_repairs(self, s=None, p=None, o=None, max_size: int = 3):
        """Inconsistency-tolerant (IAR) answers: facts derivable under EVERY
        minimal repair (ref semi_naive_with_repairs.rs / repairs.rs)."""
        repairs = self.compute_repairs(max_size=max_size)
        results = None
        base = sorted(self.all_fact_tuples())
        for removed in repairs:
            trial = Reasoner(device=str(self.device), dictionary=self.dictionary)
            for f in base:
                if f not in removed:
                    trial.add_fact_ids(*f)
            trial.rules = self.rules
            trial.infer_new_facts_semi_naive()
            answers = set(trial.query_abox(s, p, o))
            results = answers if results is None else (results & answers)
        return sorted(results or [])

    # -------------------------------------------------- backward chaining --
    def backward_chaining(self, goal, max_depth: int = 24):
        """SLD goal resolution (ref backward_chaining.rs:150).  `goal` is a
        TriplePattern or an (s,p,o) tuple of strings/'?var's."""
        from ..storage.terms import Constant, TriplePattern, Variable
        if not isinstance(goal, TriplePattern):
            def term(x):
                if isinstance(x, str) and x.startswith("?"):
                    return Variable(x[1:])
                return Constant(self._i32(self.dictionary.encode(x)))
            goal = TriplePattern(*(term(x) for x in goal))
        from .backward import backward_chain
        return backward_chain(goal, self, max_depth=max_depth)

    # ------------------------------------------------------- constraints --
    def violates_constraints(self) -> bool:
        """A constraint is a rule whose body matching any facts = violation
        (ref reasoning.rs:137)."""
        return len(self.constraint_violations()) > 0

    def constraint_violations(self) -> List[Tuple]:
        self._flush()
        from .seminaive import (_apply_negative, _eval_filters,
                                _join_premise_all_facts,
                                _match_premise_against_delta)
        out = []
        idx = self.facts.index()
        for c in self.constraints:
            if not c.premise:
                continue
            b = _match_premise_against_delta(
                c.premise[0], self.facts.s, self.facts.p, self.facts.o,
                self.device)
            if b is None:
                continue
            for prem in c.premise[1:]:
                b = _join_premise_all_facts(b, prem, idx, self.device)
                if b.is_empty():
                    break
            if b.is_empty():
                continue
            b = _eval_filters(c, b, self.db)
            b = _apply_negative(b, c.negative_premise, idx, self.device)
            if not b.is_empty():
                out.append((c, b))
        return out

    # ----------------------------------------------------------- repairs --
    def compute_repairs(self, max_size: int = 3) -> List[Set[Tuple[int, int, int]]]:
        """Exhaustive subset-removal repairs (ref reasoning.rs:148-186):
        smallest sets of base facts whose removal restores consistency."""
        self._flush()
        if not self.violates_constraints():
            return [set()]
        base = sorted(self.all_fact_tuples())
        repairs: List[Set[Tuple[int, int, int]]] = []
        for size in range(1, min(max_size, len(base)) + 1):
            for combo in combinations(base, size):
                removed = set(combo)
                trial = Reasoner(device=str(self.device), dictionary=self.dictionary)
                for f in base:
                    if f not in removed:
                        trial.add_fact_ids(*f)
                trial.rules = self.rules
                trial.constraints = self.constraints
                trial.infer_new_facts_semi_naive()
                if not trial.violates_constraints():
                    repairs.append(removed)
            if repairs:
                break
        return repairs


class _RuleDb:
    """Adapter exposing SparqlDatabase-ish surface over a Reasoner for rule
    compilation (compile_term needs dictionary + resolve_lexical)."""

    def __init__(self, reasoner: Reasoner):
        self.dictionary = reasoner.dictionary
        self.quoted_triples = reasoner.db.quoted_triples
        self.device = reasoner.device
        self._r = reasoner

    def resolve_lexical(self, term, prefixes=None):
        return self._r.db.resolve_lexical(term, prefixes)

    def value_column(self):
        return self._r.db.value_column()
