"""Device-tagged semi-naive fixpoint: scalar semiring tags as f32 columns.

Ref parity: datalog/src/materialisation/provenance_semi_naive.rs — the
GPU realization SURVEY §2.5/§2.9 calls for: MinMax / AddMult / Expiration
tags ride the columnar K6 fixpoint as a float32 column per fact; ⊗ applies
during joins, ⊕ merges duplicate derivations with a segmented reduce, and
tag-IMPROVED facts re-enter the delta (delta_improved :185-197).
Stratified NAF runs as a single negative pass after the positive fixpoint
(:297-389).

Structured-tag semirings (TopK proofs, DNF-WMC, lineage) stay on the host
path (provenance_fixpoint.py).

Implementation note: the float tag travels inside Bindings as an int32
bit-pattern column (`__tag`), so every join/gather/select applies to it
uniformly; it is reinterpreted back to f32 only at ⊗/⊕ points.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from ..engine.bindings import Bindings
from ..engine.executor import join_bindings
from ..engine.tensor_utils import group_index
from ..storage.terms import Constant, TriplePattern, Variable
from .provenance import Provenance
from .rule import Rule
from .seminaive import FactStore, membership_in_index

Triple = Tuple[int, int, int]

TAG = "__tag"


def _f2i(t: torch.Tensor) -> torch.Tensor:
    return t.view(torch.int32)


def _i2f(t: torch.Tensor) -> torch.Tensor:
    return t.view(torch.float32)


class ScalarSemiring:
    """Torch-vectorized ⊕/⊗ for float-valued semirings."""

    def __init__(self, name: str):
        name = name.lower()
        if name in ("minmax", "min"):
            self.plus = torch.maximum
            self.times = torch.minimum
            self.reduce = "amax"
            self.negate = lambda t: 1.0 - t
        elif name in ("addmult", "independent", "add"):
            # noisy-or disjunction a+b-ab (ref provenance.rs:119).
            # NOTE: on RECURSIVE rule sets the delta-re-entry evaluation
            # (reference provenance_semi_naive.rs:185-197 behavior)
            # re-accumulates cyclic contributions and saturates toward 1;
            # the host oracle's Jacobi iteration computes the
            # omega-continuous limit instead.  They agree exactly on
            # acyclic programs (see test_device_matches_host_addmult).
            self.plus = lambda a, b: a + b - a * b
            self.times = torch.mul
            self.reduce = "noisy_or"
            self.negate = None
        elif name == "expiration":
            self.plus = torch.maximum
            self.times = torch.minimum
            self.reduce = "amax"
            self.negate = None
        else:
            raise ValueError(f"no device path for semiring {name!r}")
        self.name = name

    def segment_plus(self, gid: torch.Tensor, ng: int, tags: torch.Tensor
                     ) -> torch.Tensor:
        out = torch.zeros(ng, dtype=torch.float32, device=tags.device)
        if self.reduce == "amax":
            out.fill_(float("-inf"))
            out.scatter_reduce_(0, gid, tags, reduce="amax")
        else:  # noisy-or: 1 - prod(1 - tag) per group (addmult ⊕)
            acc = torch.ones(ng, dtype=torch.float32, device=tags.device)
            acc.scatter_reduce_(0, gid, 1.0 - tags, reduce="prod")
            out = 1.0 - acc
        return out


class TaggedFactStore(FactStore):
    def __init__(self, device):
        super().__init__(device)
        self.tags = torch.empty(0, dtype=torch.float32,
                                device=torch.device(device))

    def add_tagged(self, s, p, o, tags):
        self.add_columns(s, p, o)
        self.tags = torch.cat([self.tags, tags])


def _match_delta_tagged(prem: TriplePattern, ds, dp, do_, dtags, device
                        ) -> Optional[Bindings]:
    n = ds.numel()
    mask = torch.ones(n, dtype=torch.bool, device=device)
    cols = (ds, dp, do_)
    var_first: Dict[str, torch.Tensor] = {}
    for i, t in enumerate(prem.terms()):
        if isinstance(t, Constant):
            mask &= cols[i] == t.id
        elif isinstance(t, Variable):
            if t.name in var_first:
                mask &= var_first[t.name] == cols[i]
            else:
                var_first[t.name] = cols[i]
        else:
            return None
    if not bool(mask.any()):
        return None
    out_cols = {v: c[mask] for v, c in var_first.items()}
    out_cols[TAG] = _f2i(dtags[mask])
    return Bindings(out_cols, int(mask.sum().item()), device)


def _rename_tag(b: Optional[Bindings], key: str) -> Optional[Bindings]:
    if b is None:
        return None
    cols = dict(b.cols)
    cols[key] = cols.pop(TAG)
    return Bindings(cols, b.n, b.device, b.maybe_unbound)


def infer_with_provenance_device(
    rules: List[Rule],
    seeds: Dict[Triple, float],
    semiring: ScalarSemiring,
    device="cpu",
    db=None,
    max_rounds: int = 10_000,
) -> Dict[Triple, float]:
    """Returns {triple: tag} for every fact (seeds possibly improved)."""
    dev = torch.device(device)
    facts = TaggedFactStore(dev)
    if seeds:
        import numpy as np
        items = sorted(seeds.items())
        arr = np.asarray([t for t, _ in items], dtype=np.int64).astype(np.int32)
        tags = torch.tensor([v for _, v in items], dtype=torch.float32,
                            device=dev)
        t = torch.from_numpy(arr).to(dev)
        facts.add_tagged(t[:, 0].contiguous(), t[:, 1].contiguous(),
                         t[:, 2].contiguous(), tags)
    positive = [r for r in rules if not r.negative_premise]
    naf = [r for r in rules if r.negative_premise]

    def run(active: List[Rule], known_for_naf: bool):
        ds, dp, do_ = facts.s.clone(), facts.p.clone(), facts.o.clone()
        dtags = facts.tags.clone()
        for _ in range(max_rounds):
            if ds.numel() == 0:
                break
            cand_s, cand_p, cand_o, cand_t = [], [], [], []
            for rule in active:
                np_ = len(rule.premise)
                for i in range(np_):
                    b = _match_delta_tagged(rule.premise[i], ds, dp, do_,
                                            dtags, dev)
                    if b is None or b.is_empty():
                        continue
                    ok = True
                    for j in range(np_):
                        if j == i:
                            continue
                        cand = _rename_tag(
                            _match_delta_tagged(
                                rule.premise[j], facts.s, facts.p, facts.o,
                                facts.tags, dev), "__tag2")
                        if cand is None or cand.is_empty():
                            ok = False
                            break
                        b = join_bindings(b, cand)
                        if b.is_empty():
                            ok = False
                            break
                        merged = semiring.times(_i2f(b.col(TAG)),
                                                _i2f(b.col("__tag2")))
                        cols = dict(b.cols)
                        cols.pop("__tag2")
                        cols[TAG] = _f2i(merged)
                        b = Bindings(cols, b.n, dev, b.maybe_unbound)
                    if not ok or b.is_empty():
                        continue
                    if db is not None and rule.filters:
                        for f in rule.filters:
                            b = b.select(f.eval_mask(b, db))
                            if b.is_empty():
                                break
                        if b.is_empty():
                            continue
                    if known_for_naf and rule.negative_premise:
                        b = _naf_pass(rule, b, facts, semiring, dev)
                        if b is None or b.is_empty():
                            continue
                    for concl in rule.conclusion:
                        cs = _concl_col(concl.s, b, dev)
                        cp = _concl_col(concl.p, b, dev)
                        co = _concl_col(concl.o, b, dev)
                        if cs is None or cp is None or co is None:
                            continue
                        cand_s.append(cs)
                        cand_p.append(cp)
                        cand_o.append(co)
                        cand_t.append(_i2f(b.col(TAG)))
            if not cand_s:
                break
            s = torch.cat(cand_s)
            p = torch.cat(cand_p)
            o = torch.cat(cand_o)
            t = torch.cat(cand_t)
            # ⊕-merge duplicate derivations
            gid, ng = group_index([s, p, o])
            merged_t = semiring.segment_plus(gid, ng, t)
            rep = torch.full((ng,), -1, dtype=torch.long, device=dev)
            rep.scatter_(0, gid, torch.arange(s.numel(), dtype=torch.long,
                                              device=dev))
            s, p, o = s[rep], p[rep], o[rep]
            t = merged_t
            # split into brand-new vs improvements of known facts
            idx = facts.index()
            tail = facts.tail_columns()
            known_mask = membership_in_index(idx, s, p, o)
            if tail[0].numel():
                from ..engine.tensor_utils import membership_mask, unique_rows
                known_mask |= membership_mask([s, p, o],
                                              unique_rows(list(tail)))
            new_mask = ~known_mask
            nds, ndp, ndo, ndt = [], [], [], []
            if bool(new_mask.any()):
                facts.add_tagged(s[new_mask], p[new_mask], o[new_mask],
                                 t[new_mask])
                nds.append(s[new_mask])
                ndp.append(p[new_mask])
                ndo.append(o[new_mask])
                ndt.append(t[new_mask])
            if bool(known_mask.any()):
                # tag improvement check against stored tags (host-side map
                # for the known positions — sizes here are the delta, small)
                ks, kp, ko = s[known_mask], p[known_mask], o[known_mask]
                kt = t[known_mask]
                impr_s, impr_p, impr_o, impr_t = _apply_improvements(
                    facts, ks, kp, ko, kt, semiring)
                if impr_s is not None:
                    nds.append(impr_s)
                    ndp.append(impr_p)
                    ndo.append(impr_o)
                    ndt.append(impr_t)
            if not nds:
                break
            ds = torch.cat(nds)
            dp = torch.cat(ndp)
            do_ = torch.cat(ndo)
            dtags = torch.cat(ndt)

    run(positive, known_for_naf=False)
    if naf:
        run(naf, known_for_naf=True)

    out: Dict[Triple, float] = {}
    su = (facts.s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    pu = (facts.p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    ou = (facts.o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    tt = facts.tags.cpu().tolist()
    for a, b_, c, v in zip(su, pu, ou, tt):
        out[(a, b_, c)] = v
    return out


def _naf_pass(rule: Rule, b: Bindings, facts: TaggedFactStore,
              semiring: ScalarSemiring, dev) -> Optional[Bindings]:
    """⊗ in the negated premises (absent -> 1.0, present -> negate(tag));
    semirings without negation drop derivations whose negated premise is
    present (boolean NAF)."""
    for neg in rule.negative_premise:
        cols = []
        ok = True
        for term in neg.terms():
            if isinstance(term, Constant):
                cols.append(torch.full((b.n,), term.id, dtype=torch.int32,
                                       device=dev))
            elif isinstance(term, Variable) and b.has(term.name):
                cols.append(b.col(term.name))
            else:
                ok = False
                break
        if not ok:
            return None
        idx = facts.index()
        present = membership_in_index(idx, cols[0], cols[1], cols[2])
        tail = facts.tail_columns()
        if tail[0].numel():
            from ..engine.tensor_utils import membership_mask, unique_rows
            present |= membership_mask(cols, unique_rows(list(tail)))
        if semiring.negate is None:
            b = b.select(~present)
        else:
            # factor = 1 where absent; negate(stored tag) where present —
            # look up the present facts' tags (delta-sized, host map)
            factor = torch.ones(b.n, dtype=torch.float32, device=dev)
            if bool(present.any()):
                ps = cols[0][present]
                pp = cols[1][present]
                po = cols[2][present]
                su = (ps.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                pu = (pp.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                ou = (po.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                fs = (facts.s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                fp = (facts.p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                fo = (facts.o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
                pos_map = {k: i for i, k in enumerate(zip(fs, fp, fo))}
                positions = torch.tensor(
                    [pos_map[k] for k in zip(su, pu, ou)],
                    dtype=torch.long, device=dev)
                negated = semiring.negate(facts.tags[positions])
                factor[present] = negated
            merged = semiring.times(_i2f(b.col(TAG)), factor)
            c2 = dict(b.cols)
            c2[TAG] = _f2i(merged)
            b = Bindings(c2, b.n, dev, b.maybe_unbound)
        if b.is_empty():
            return b
    return b


def _concl_col(term, b: Bindings, dev):
    if isinstance(term, Constant):
        return torch.full((b.n,), term.id, dtype=torch.int32, device=dev)
    if isinstance(term, Variable) and b.has(term.name):
        return b.col(term.name)
    return None


def _apply_improvements(facts: TaggedFactStore, s, p, o, t,
                        semiring: ScalarSemiring):
    """⊕-merge new derivations of KNOWN facts into their stored tags;
    returns the improved subset (re-enters the delta)."""
    # locate each (s,p,o) in the store via a packed host map of positions
    su = (s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    pu = (p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    ou = (o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
    pos_map = getattr(facts, "_pos_map", None)
    if pos_map is None or len(pos_map) != facts.n:
        fs = (facts.s.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        fp = (facts.p.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        fo = (facts.o.to(torch.int64) & 0xFFFFFFFF).cpu().tolist()
        pos_map = {k: i for i, k in enumerate(zip(fs, fp, fo))}
        facts._pos_map = pos_map  # type: ignore[attr-defined]
    positions = torch.tensor(
        [pos_map[k] for k in zip(su, pu, ou)], dtype=torch.long,
        device=facts.device)
    old = facts.tags[positions]
    merged = semiring.plus(old, t)
    improved = merged > old
    facts.tags[positions] = merged
    if not bool(improved.any()):
        return None, None, None, None
    return (s[improved], p[improved], o[improved], merged[improved])
