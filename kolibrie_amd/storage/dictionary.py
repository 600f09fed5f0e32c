"""Host-side bidirectional string <-> u32 dictionary.

Strings never touch the GPU: kernels see only int32 IDs.  Alongside the
string table we maintain a *numeric value column* (float64 per ID, parsed at
encode time, non-numeric => 0.0) that is uploaded to the device so FILTER
ordering comparisons and aggregates never decode strings (the reference
decodes and `parse::<f64>().unwrap_or(0.0)` per row —
streamertail_optimizer/execution/types.rs:349-359; we pre-parse once).

RDF-star quoted triples are interned with bit 31 set
(ref: shared/src/quoted_triple_store.rs:17-80) and recurse on encode/decode
(ref: shared/src/dictionary.rs:62 decode_term).

Parity surface: shared/src/dictionary.rs:17-91 (encode/decode/merge).
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

from .terms import QUOTED_TRIPLE_ID_BIT, UNBOUND_U32, is_quoted_id


def _try_parse_float(s: str) -> float:
    try:
        v = float(s)
        if math.isnan(v) or math.isinf(v):
            return 0.0
        return v
    except (ValueError, TypeError):
        return 0.0


class Dictionary:
    """Bidirectional string<->id map.  ID 0 is reserved for "" (and doubles
    as the default-graph ID)."""

    __slots__ = ("str_to_id", "id_to_str", "values", "_values_dirty",
                 "_np_table", "_np_table_n", "annex")

    def __init__(self):
        self.str_to_id: Dict[str, int] = {"": 0}
        self.id_to_str: List[str] = [""]
        # values[i] = float64 numeric interpretation of id_to_str[i] (0.0 if
        # non-numeric) — the device value column source.
        self.values: List[float] = [0.0]
        self._values_dirty = True
        self._np_table = None     # numpy object array mirror of id_to_str
        self._np_table_n = 0
        # native bulk-vocabulary annex (C++-primary dictionary tail): once
        # attached by a bulk file load, the Python prefix [0, P0) freezes
        # and every later term id (bulk or incremental) lives natively;
        # lookup/encode/decode consult it on prefix misses.  (module, handle)
        self.annex = None

    def attach_annex(self):
        if self.annex is None:
            from ..ops import _native
            if _native is None:
                raise RuntimeError("native extension required for the "
                                   "bulk-vocabulary annex")
            # seed with the frozen Python prefix so the native bulk
            # parser dedups against EXISTING ids ("" at id 0, or any
            # term interned before the first bulk load)
            self.annex = (_native,
                          _native.vocab_create_seeded(self.id_to_str))
        return self.annex

    def np_table(self):
        """Numpy object-array mirror of id_to_str for BATCH decode
        (decode_rows' per-cell Python loop was the materializing-query
        bottleneck — VERDICT r1 item 6).  Grown incrementally: the
        dictionary is append-only."""
        import numpy as np
        n = len(self.id_to_str)
        if self._np_table is None:
            self._np_table = np.array(self.id_to_str, dtype=object)
            self._np_table_n = n
        elif self._np_table_n < n:
            tail = np.array(self.id_to_str[self._np_table_n:], dtype=object)
            self._np_table = np.concatenate([self._np_table, tail])
            self._np_table_n = n
        return self._np_table

    def __len__(self):
        n = len(self.id_to_str)
        if self.annex is not None:
            n += self.annex[0].vocab_len(self.annex[1])
        return n

    def encode(self, s: str) -> int:
        """Intern a string, returning its u32 ID (ref dictionary.rs:32)."""
        i = self.str_to_id.get(s)
        if i is not None:
            return i
        if self.annex is not None:
            mod, h = self.annex
            i = mod.vocab_insert(h, s)
            if i >= QUOTED_TRIPLE_ID_BIT:
                raise OverflowError(
                    "dictionary ID space exhausted (2^31 terms)")
            self._values_dirty = True
            return i
        i = len(self.id_to_str)
        if i >= QUOTED_TRIPLE_ID_BIT:
            raise OverflowError("dictionary ID space exhausted (2^31 terms)")
        self.str_to_id[s] = i
        self.id_to_str.append(s)
        self.values.append(_try_parse_float(s))
        self._values_dirty = True
        return i

    def encode_many(self, strs) -> "object":
        """Bulk intern: returns a uint32 numpy array of IDs.  The tuned
        local-variable loop is ~2x a per-call encode() — this is the host
        half of the parallel bulk-parse pipeline."""
        import numpy as np
        if self.annex is not None:
            out = np.empty(len(strs), dtype=np.uint32)
            enc = self.encode
            for i, s in enumerate(strs):
                out[i] = enc(s)
            return out
        sti = self.str_to_id
        its = self.id_to_str
        vals = self.values
        get = sti.get
        append = its.append
        vappend = vals.append
        n = len(its)
        out = np.empty(len(strs), dtype=np.uint32)
        for i, s in enumerate(strs):
            x = get(s)
            if x is None:
                if n >= QUOTED_TRIPLE_ID_BIT:
                    raise OverflowError(
                        "dictionary ID space exhausted (2^31 terms)")
                x = n
                sti[s] = x
                append(s)
                vappend(_try_parse_float(s))
                n += 1
            out[i] = x
        self._values_dirty = True
        return out

    def lookup(self, s: str) -> Optional[int]:
        i = self.str_to_id.get(s)
        if i is None and self.annex is not None:
            j = self.annex[0].vocab_lookup(self.annex[1], s)
            return None if j < 0 else j
        return i

    def decode(self, i: int) -> Optional[str]:
        """ID -> string (plain terms only; ref dictionary.rs:49)."""
        i &= 0xFFFFFFFF
        if i < len(self.id_to_str):
            return self.id_to_str[i]
        if self.annex is not None:
            return self.annex[0].vocab_get(self.annex[1], i)
        return None

    def contains(self, s: str) -> bool:
        if s in self.str_to_id:
            return True
        return (self.annex is not None
                and self.annex[0].vocab_lookup(self.annex[1], s) >= 0)

    def numeric_value(self, i: int) -> float:
        i &= 0xFFFFFFFF
        if i < len(self.values):
            return self.values[i]
        if self.annex is not None:
            return self.annex[0].vocab_value(self.annex[1], i)
        return 0.0

    def values_array(self):
        """float64 numpy array over the WHOLE id space (prefix + annex) —
        the device value-column source."""
        import numpy as np
        pre = np.asarray(self.values, dtype=np.float64)
        if self.annex is None:
            return pre
        tail = self.annex[0].vocab_values(self.annex[1]).numpy()
        return np.concatenate([pre, tail]) if tail.size else pre

    def iter_strings(self):
        """Iterate every interned string in id order (checkpoint/merge);
        annex strings materialize in batches."""
        yield from self.id_to_str
        if self.annex is not None:
            mod, h = self.annex
            n = mod.vocab_len(h)
            for start in range(0, n, 1 << 16):
                yield from mod.vocab_export_strings(h, start, 1 << 16)

    def merge(self, other: "Dictionary") -> Dict[int, int]:
        """Merge `other` into self, returning an old-id -> new-id remap
        (ref dictionary.rs:82 — used by parallel parse shards)."""
        remap: Dict[int, int] = {}
        for old_id, s in enumerate(other.iter_strings()):
            remap[old_id] = self.encode(s)
        return remap


class QuotedTripleStore:
    """Interns RDF-star quoted triples as IDs with bit 31 set.

    Ref: shared/src/quoted_triple_store.rs:17-80.  A quoted triple is a
    (s,p,o) of u32 IDs (any of which may itself be quoted).  IDs are
    QUOTED_TRIPLE_ID_BIT | index; the allocator stays clear of UNBOUND_U32.
    """

    __slots__ = ("triple_to_id", "id_to_triple")

    def __init__(self):
        self.triple_to_id: Dict[Tuple[int, int, int], int] = {}
        self.id_to_triple: List[Tuple[int, int, int]] = []

    def __len__(self):
        return len(self.id_to_triple)

    def encode(self, s: int, p: int, o: int) -> int:
        key = (s & 0xFFFFFFFF, p & 0xFFFFFFFF, o & 0xFFFFFFFF)
        i = self.triple_to_id.get(key)
        if i is not None:
            return i
        idx = len(self.id_to_triple)
        qid = QUOTED_TRIPLE_ID_BIT | idx
        if qid >= UNBOUND_U32:
            raise OverflowError("quoted-triple ID space exhausted")
        self.triple_to_id[key] = qid
        self.id_to_triple.append(key)
        return qid

    def decode(self, qid: int) -> Optional[Tuple[int, int, int]]:
        qid &= 0xFFFFFFFF
        if not (qid & QUOTED_TRIPLE_ID_BIT):
            return None
        idx = qid & ~QUOTED_TRIPLE_ID_BIT
        if idx < len(self.id_to_triple):
            return self.id_to_triple[idx]
        return None

    def lookup(self, s: int, p: int, o: int) -> Optional[int]:
        return self.triple_to_id.get((s & 0xFFFFFFFF, p & 0xFFFFFFFF, o & 0xFFFFFFFF))

    def merge(self, other: "QuotedTripleStore", id_remap: Dict[int, int]) -> Dict[int, int]:
        """Merge quoted triples from `other` given a plain-ID remap; returns
        quoted-id remap (ref quoted_triple_store.rs:73)."""
        qremap: Dict[int, int] = {}

        def remap_one(old_qid: int) -> int:
            if old_qid in qremap:
                return qremap[old_qid]
            t = other.decode(old_qid)
            assert t is not None
            parts = []
            for x in t:
                if is_quoted_id(x):
                    parts.append(remap_one(x))
                else:
                    parts.append(id_remap.get(x, x))
            new_qid = self.encode(*parts)
            qremap[old_qid] = new_qid
            return new_qid

        for idx in range(len(other.id_to_triple)):
            remap_one(QUOTED_TRIPLE_ID_BIT | idx)
        return qremap
