"""TagStore: triple -> provenance tag map with RDF-star encoding.

Ref parity: shared/src/tag_store.rs (406 LoC) — update_disjunction (:58),
RDF-star tag encoding `<< s p o >> prob:value "0.7"` (:89-186).
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

from .provenance import Provenance

Triple = Tuple[int, int, int]

PROB_VALUE_IRI = "http://kolibrie.amd/prob#value"


class TagStore:
    def __init__(self, semiring: Provenance):
        self.semiring = semiring
        self.tags: Dict[Triple, object] = {}

    def get(self, t: Triple):
        return self.tags.get(_norm(t))

    def set(self, t: Triple, tag):
        self.tags[_norm(t)] = tag

    def update_disjunction(self, t: Triple, tag) -> bool:
        """⊕-merge a new derivation's tag (ref tag_store.rs:58).
        Returns True if the stored tag changed."""
        t = _norm(t)
        prev = self.tags.get(t)
        if prev is None:
            self.tags[t] = tag
            return True
        merged = self.semiring.plus(prev, tag)
        if merged != prev:
            self.tags[t] = merged
            return True
        return False

    def probability(self, t: Triple) -> Optional[float]:
        tag = self.get(t)
        return None if tag is None else self.semiring.recover(tag)

    # ------------------------------------------------------- RDF-star I/O
    def encode_into_db(self, db):
        """Assert `<< s p o >> prob:value "<p>"` annotation triples
        (ref tag_store.rs:89-186)."""
        pv = db.dictionary.encode(PROB_VALUE_IRI)
        for (s, p, o), tag in self.tags.items():
            qt = db.quoted_triples.encode(s, p, o)
            prob = self.semiring.recover(tag)
            val = db.dictionary.encode(f"{prob:g}")
            db.store.insert_quad(0, qt, pv, val)

    @staticmethod
    def decode_from_db(db, semiring: Provenance) -> "TagStore":
        ts = TagStore(semiring)
        pv = db.dictionary.lookup(PROB_VALUE_IRI)
        if pv is None:
            return ts
        s_c, p_c, o_c = db.store.graph_index(0).lookup(None, pv, None)
        import torch
        for qt_id, val_id in zip(
            (s_c.to(torch.int64) & 0xFFFFFFFF).tolist(),
            (o_c.to(torch.int64) & 0xFFFFFFFF).tolist(),
        ):
            t = db.quoted_triples.decode(qt_id)
            if t is None:
                continue
            try:
                prob = float(db.dictionary.decode(val_id) or "0")
            except ValueError:
                continue
            ts.set(t, semiring.tag_from_probability(prob))
        return ts


def _norm(t: Triple) -> Triple:
    return tuple(x & 0xFFFFFFFF for x in t)  # type: ignore
