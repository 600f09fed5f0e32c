"""Hybrid anytime probabilistic inference.

Ref parity: shared/src/hybrid.rs (2 315 LoC):
  SeedRegistry (stream/static/exclusive seeds, :108-276), SeedSnapshot
  (:285), LineageStore (AND/OR/NOT DAG with structural hashing, :405-577),
  LineageProvenance (:579), HybridConfig (defaults: threshold 0.5, policy
  Explicit|CostRatio, band_epsilon 0.02, k 8 -> 64 growth x2, topk budget
  25 ms, SDD budget 250 ms / 100K nodes; validation :737-771),
  evaluate_hybrid[_with_clock] (:1377-1386) — the escalation controller:
  top-K proof enumeration under deadline -> probability bounds interval ->
  if still inconclusive, compile the lineage to an SDD for exact WMC;
  HybridProbabilityResult::{Decided, NeedsExact, ...} (:839), per-stage
  metrics (:818), RDF-star result encoding (:1593), HybridClock (:28)
  allowing fake clocks in budget tests.

Plus hybrid materialisation (datalog/src/materialisation/
hybrid_materialisation.rs): build lineage circuits during inference
(materialize_lineage :136), validate monotone rules (:79), evaluate a
HybridProbabilityResult per derived triple (:35).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Set, Tuple

from .provenance import TopKProofs
from .rule import Rule
from .sdd import SddManager, SddOperationBudget

Triple = Tuple[int, int, int]


# ------------------------------------------------------------------- clock --
class HybridClock:
    """Injectable clock (ref hybrid.rs:28) for deterministic budget tests."""

    def now(self) -> float:
        return time.monotonic()


class FakeClock(HybridClock):
    def __init__(self):
        self.t = 0.0

    def now(self) -> float:
        return self.t

    def advance(self, dt: float):
        self.t += dt


# ------------------------------------------------------------------- seeds --
class SeedKind:
    STREAM = "stream"
    STATIC = "static"
    EXCLUSIVE = "exclusive"


@dataclass
class Seed:
    seed_id: int
    triple: Triple
    probability: float
    kind: str = SeedKind.STREAM
    group: Optional[str] = None      # exclusive group name


class SeedRegistry:
    """(ref hybrid.rs:108-276)"""

    def __init__(self):
        self.seeds: Dict[Triple, Seed] = {}
        self.by_id: Dict[int, Seed] = {}
        self.exclusive_groups: Dict[str, List[int]] = {}
        self._next_id = 1

    def register(self, triple: Triple, probability: float,
                 kind: str = SeedKind.STREAM,
                 group: Optional[str] = None) -> Seed:
        t = _norm(triple)
        if t in self.seeds:
            s = self.seeds[t]
            s.probability = probability
            return s
        s = Seed(self._next_id, t, probability, kind, group)
        self._next_id += 1
        self.seeds[t] = s
        self.by_id[s.seed_id] = s
        if group is not None:
            self.exclusive_groups.setdefault(group, []).append(s.seed_id)
        return s

    def snapshot(self) -> "SeedSnapshot":
        return SeedSnapshot(
            {t: (s.seed_id, s.probability) for t, s in self.seeds.items()},
            {g: list(ids) for g, ids in self.exclusive_groups.items()},
        )


@dataclass
class SeedSnapshot:
    """Immutable view taken per evaluation (ref hybrid.rs:285)."""
    seeds: Dict[Triple, Tuple[int, float]]
    exclusive_groups: Dict[str, List[int]] = field(default_factory=dict)

    def seed_for(self, triple: Triple) -> Optional[Tuple[int, float]]:
        return self.seeds.get(_norm(triple))


# ----------------------------------------------------------------- lineage --
AND, OR, NOT, LEAF, CONST_TRUE = "and", "or", "not", "leaf", "true"


class LineageStore:
    """Structural-hashed AND/OR/NOT DAG (ref hybrid.rs:405-577)."""

    def __init__(self):
        self.nodes: List[Tuple[str, Tuple[int, ...]]] = [(CONST_TRUE, ())]
        self.unique: Dict[Tuple[str, Tuple[int, ...]], int] = {
            (CONST_TRUE, ()): 0}

    def _mk(self, kind: str, children: Tuple[int, ...]) -> int:
        key = (kind, children)
        nid = self.unique.get(key)
        if nid is None:
            nid = len(self.nodes)
            self.nodes.append(key)
            self.unique[key] = nid
        return nid

    def true_node(self) -> int:
        return 0

    def leaf(self, seed_id: int) -> int:
        return self._mk(LEAF, (seed_id,))

    def and_node(self, children: Sequence[int]) -> int:
        ch = tuple(sorted(set(children)))
        ch = tuple(c for c in ch if c != 0)
        if not ch:
            return 0
        if len(ch) == 1:
            return ch[0]
        return self._mk(AND, ch)

    def or_node(self, children: Sequence[int]) -> int:
        ch = tuple(sorted(set(children)))
        if 0 in ch:
            return 0
        if len(ch) == 1:
            return ch[0]
        return self._mk(OR, ch)

    def not_node(self, child: int) -> int:
        return self._mk(NOT, (child,))

    def proofs(self, node: int, k_limit: int, deadline: Optional[float],
               clock: HybridClock) -> Optional[List[Set[int]]]:
        """Enumerate up to k_limit proofs (sets of seed ids); None on
        deadline (drives escalation)."""
        memo: Dict[int, List[Set[int]]] = {}

        def rec(nid: int) -> Optional[List[Set[int]]]:
            if deadline is not None and clock.now() > deadline:
                return None
            hit = memo.get(nid)
            if hit is not None:
                return hit
            kind, ch = self.nodes[nid]
            if kind == CONST_TRUE:
                out = [set()]
            elif kind == LEAF:
                out = [{ch[0]}]
            elif kind == OR:
                out = []
                for c in ch:
                    sub = rec(c)
                    if sub is None:
                        return None
                    out.extend(sub)
                    if len(out) > 4 * k_limit:
                        out = out[:4 * k_limit]
            elif kind == AND:
                out = [set()]
                for c in ch:
                    sub = rec(c)
                    if sub is None:
                        return None
                    out = [a | b for a in out for b in sub][:4 * k_limit]
            elif kind == NOT:
                return None   # proofs don't cover negation: escalate
            else:
                raise ValueError(kind)
            memo[nid] = out
            return out

        res = rec(node)
        return None if res is None else res[:4 * k_limit]

    def to_sdd(self, node: int, manager: SddManager,
               weights: Dict[int, float],
               budget: Optional[SddOperationBudget] = None) -> Optional[int]:
        """Compile the lineage to an SDD (ref compile_lineage_to_sdd:1248)."""
        for sid, p in weights.items():
            manager.declare_var(sid, pos_weight=p)
        memo: Dict[int, Optional[int]] = {}

        def rec(nid: int) -> Optional[int]:
            if nid in memo:
                return memo[nid]
            kind, ch = self.nodes[nid]
            if kind == CONST_TRUE:
                out = manager.true_node()
            elif kind == LEAF:
                out = manager.literal(ch[0], True)
            elif kind in (AND, OR):
                op = "and" if kind == AND else "or"
                out = manager.true_node() if kind == AND else manager.false_node()
                for c in ch:
                    sub = rec(c)
                    if sub is None:
                        return None
                    if budget is not None:
                        nxt = manager.try_apply(op, out, sub, budget)
                        if nxt is None:
                            memo[nid] = None
                            return None
                        out = nxt
                    else:
                        out = manager.apply(op, out, sub)
            elif kind == NOT:
                sub = rec(ch[0])
                if sub is None:
                    return None
                out = manager.negate(sub)
            else:
                raise ValueError(kind)
            memo[nid] = out
            return out

        return rec(node)


class LineageProvenance:
    """Semiring over LineageStore nodes (ref hybrid.rs:579)."""
    name = "lineage"

    def __init__(self, store: Optional[LineageStore] = None):
        self.store = store if store is not None else LineageStore()
        self._probs: Dict[int, float] = {}
        self._next = 1

    def zero(self):
        return -1   # sentinel: no derivation

    def one(self):
        return self.store.true_node()

    def plus(self, a, b):
        if a == -1:
            return b
        if b == -1:
            return a
        return self.store.or_node([a, b])

    def times(self, a, b):
        if a == -1 or b == -1:
            return -1
        return self.store.and_node([a, b])

    def negate(self, a):
        if a == -1:
            return self.store.true_node()
        return self.store.not_node(a)

    def saturate(self, a):
        return a

    def tag_from_probability(self, p, seed_id=None):
        sid = seed_id if seed_id is not None else self._next
        self._next = max(self._next, sid) + 1
        self._probs[sid] = float(p)
        return self.store.leaf(sid)

    def recover(self, tag) -> float:
        if tag == -1:
            return 0.0
        m = SddManager()
        node = self.store.to_sdd(tag, m, self._probs)
        return 0.0 if node is None else m.wmc(node)

    def better(self, a, b) -> bool:
        return a != b and a != -1 and b == -1


# ------------------------------------------------------------------ config --
@dataclass
class HybridConfig:
    """(ref hybrid.rs:661-705; defaults and validation :737-771)"""
    threshold: float = 0.5
    policy: str = "Explicit"            # Explicit | CostRatio
    band_epsilon: float = 0.02
    k_initial: int = 8
    k_max: int = 64
    k_growth: int = 2
    topk_budget_ms: float = 25.0
    sdd_budget_ms: float = 250.0
    sdd_node_cap: int = 100_000
    confidence: float = 0.95

    def validate(self):
        if not (0.0 <= self.threshold <= 1.0):
            raise ValueError("threshold must be in [0,1]")
        if not (0.0 <= self.band_epsilon <= 0.5):
            raise ValueError("band_epsilon must be in [0,0.5]")
        if self.k_initial < 1 or self.k_max < self.k_initial:
            raise ValueError("invalid k range")
        if self.k_growth < 2:
            raise ValueError("k growth factor must be >= 2")
        return self

    @staticmethod
    def from_prob_annotation(ann) -> "HybridConfig":
        cfg = HybridConfig()
        if ann.threshold is not None:
            cfg.threshold = ann.threshold
        if ann.confidence is not None:
            cfg.confidence = ann.confidence
        for k, v in ann.extra.items():
            if hasattr(cfg, k):
                setattr(cfg, k, type(getattr(cfg, k))(v))
        return cfg.validate()


@dataclass
class HybridMetrics:
    """Per-stage latencies (ref hybrid.rs:818-836)."""
    topk_ms: float = 0.0
    bounds_ms: float = 0.0
    sdd_ms: float = 0.0
    k_used: int = 0
    escalated: bool = False


@dataclass
class HybridProbabilityResult:
    """(ref hybrid.rs:839 Decided / NeedsExact / ...)"""
    status: str                 # "Decided" | "DecidedExact" | "NeedsExact" | "Inconclusive"
    probability: Optional[float]
    lower: float = 0.0
    upper: float = 1.0
    above_threshold: Optional[bool] = None
    metrics: HybridMetrics = field(default_factory=HybridMetrics)


# -------------------------------------------------------------- controller --
def evaluate_hybrid(lineage: LineageStore, node: int,
                    weights: Dict[int, float], config: HybridConfig,
                    clock: Optional[HybridClock] = None
                    ) -> HybridProbabilityResult:
    """Escalation controller (ref evaluate_hybrid_with_clock:1377):
    1. top-K proof enumeration with growing k under a deadline;
    2. probability bounds from the retained proofs: lower = P(k proofs),
       upper = min(1, lower + mass unaccounted) — decide when the
       threshold lies outside [lower-eps, upper+eps];
    3. exact SDD WMC under its own budget when still inconclusive.
    """
    clock = clock or HybridClock()
    metrics = HybridMetrics()
    cfg = config
    t0 = clock.now()
    k = cfg.k_initial
    prob_sr = TopKProofs(k=min(63, cfg.k_max), weights=dict(weights))
    deadline = t0 + cfg.topk_budget_ms / 1000.0
    lower = 0.0
    upper = 1.0
    while True:
        proofs = lineage.proofs(node, k, deadline, clock)
        metrics.k_used = k
        if proofs is None:
            break   # deadline or negation: escalate
        kept = prob_sr._truncate([frozenset(p) for p in proofs])[:k]
        lower = prob_sr.recover(tuple(kept))
        exhausted = len(proofs) <= k
        upper = lower if exhausted else min(
            1.0, lower + sum(prob_sr.proof_probability(frozenset(p))
                             for p in proofs[k:]))
        metrics.topk_ms = (clock.now() - t0) * 1000.0
        thr = cfg.threshold
        if lower - cfg.band_epsilon > thr:
            return HybridProbabilityResult(
                "Decided", lower, lower, upper, True, metrics)
        if upper + cfg.band_epsilon < thr:
            return HybridProbabilityResult(
                "Decided", upper, lower, upper, False, metrics)
        if exhausted:
            return HybridProbabilityResult(
                "DecidedExact", lower, lower, lower, lower >= thr, metrics)
        if k >= cfg.k_max:
            break
        k = min(cfg.k_max, k * cfg.k_growth)
        if clock.now() > deadline:
            break

    # stage 3: exact SDD WMC (ref compile_lineage_to_sdd:1248).  Larger
    # seed sets compile on a BALANCED vtree (sdd_vtree.VtreeSddManager —
    # the general-vtree engine); tiny ones keep the
    # right-linear OBDD manager whose layout diff_sdd also understands.
    metrics.escalated = True
    t_sdd = clock.now()
    if len(weights) >= 8:
        from .sdd_vtree import VtreeSddManager
        manager = VtreeSddManager("balanced", sorted(weights))
    else:
        manager = SddManager()
    budget = SddOperationBudget(max_nodes=cfg.sdd_node_cap,
                                deadline_s=cfg.sdd_budget_ms / 1000.0)
    sdd_node = lineage.to_sdd(node, manager, weights, budget)
    metrics.sdd_ms = (clock.now() - t_sdd) * 1000.0
    if sdd_node is None:
        return HybridProbabilityResult(
            "Inconclusive", None, lower, upper, None, metrics)
    p = manager.wmc(sdd_node)
    return HybridProbabilityResult(
        "DecidedExact", p, p, p, p >= cfg.threshold, metrics)


# -------------------------------------------- materialisation + evaluation --
def materialize_lineage(rules: List[Rule], seeds: Dict[Triple, float],
                        deterministic: Optional[Set[Triple]] = None
                        ) -> Tuple[LineageStore, Dict[Triple, int], Dict[int, float]]:
    """Run the tagged fixpoint building lineage circuits
    (ref hybrid_materialisation.rs:136 materialize_lineage).  Returns
    (store, triple -> lineage node, seed weights)."""
    validate_monotone(rules)
    prov = LineageProvenance()
    tags: Dict[Triple, object] = {}
    weights: Dict[int, float] = {}
    for i, (t, p) in enumerate(sorted(seeds.items())):
        tag = prov.tag_from_probability(p, seed_id=i + 1)
        weights[i + 1] = p
        tags[_norm(t)] = tag
    for t in (deterministic or set()):
        tags[_norm(t)] = prov.one()
    from .provenance_fixpoint import infer_with_provenance
    known = infer_with_provenance(rules, tags, prov)
    return prov.store, {t: tag for t, tag in known.items()}, weights


def validate_monotone(rules: List[Rule]):
    """Hybrid evaluation requires monotone (negation-free) rules
    (ref hybrid_materialisation.rs:79)."""
    for r in rules:
        if r.negative_premise:
            raise ValueError(
                f"hybrid evaluation requires monotone rules; rule "
                f"{r.name or '<anon>'} uses negation")


def evaluate_hybrid_rules(rules: List[Rule], facts, db, config: HybridConfig,
                          snapshot: SeedSnapshot,
                          clock: Optional[HybridClock] = None):
    """Window-level entry used by SimpleR2R: materialize lineage over the
    store facts + seeds, evaluate each derived triple, return the derived
    triples above threshold + all results."""
    import torch
    su = (facts.s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    pu = (facts.p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    ou = (facts.o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    base = set(zip(su, pu, ou))
    seeds = {}
    deterministic = set()
    for t in base:
        s = snapshot.seed_for(t)
        if s is not None:
            seeds[t] = s[1]
        else:
            deterministic.add(t)
    store, node_by_triple, weights = materialize_lineage(
        rules, seeds, deterministic)
    results: Dict[Triple, HybridProbabilityResult] = {}
    derived: List[Triple] = []
    for t, tag in node_by_triple.items():
        if t in base:
            continue
        if tag == -1:
            continue
        res = evaluate_hybrid(store, tag, weights, config, clock)
        results[t] = res
        if res.above_threshold:
            derived.append(t)
    return derived, results


def encode_results_rdf_star(results: Dict[Triple, HybridProbabilityResult],
                            db):
    """RDF-star result encoding `<< s p o >> hybrid:p "0.93"`
    (ref hybrid.rs:1593)."""
    pv = db.dictionary.encode("http://kolibrie.amd/hybrid#p")
    st = db.dictionary.encode("http://kolibrie.amd/hybrid#status")
    for t, res in results.items():
        qt = db.quoted_triples.encode(*t)
        if res.probability is not None:
            db.store.insert_quad(
                0, qt, pv, db.dictionary.encode(f"{res.probability:g}"))
        db.store.insert_quad(0, qt, st, db.dictionary.encode(res.status))


def _norm(t: Triple) -> Triple:
    return tuple(x & 0xFFFFFFFF for x in t)  # type: ignore
