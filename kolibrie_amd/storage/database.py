"""SparqlDatabase — the storage facade (ref: kolibrie/src/sparql_database.rs:172-188).

Owns the host dictionary + quoted-triple store, the device-resident QuadStore,
prefixes, UDFs, registered rules, neural declarations and cached statistics.
Strings are encoded exactly like the reference: IRIs stored bracket-less,
literals as their lexical form, blank nodes as `_:name`
(ref: streamertail_optimizer/utils.rs:233-252 resolve_sparql_lexical_value).
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

import torch

from .dataset import DEFAULT_GRAPH, QuadStore
from .dictionary import Dictionary, QuotedTripleStore
from .terms import is_quoted_id

RDF_TYPE = "http://www.w3.org/1999/02/22-rdf-syntax-ns#type"

# Built-in prefixes every query sees (reference registers rdf: implicitly via
# common usage; we keep the table minimal and query-extendable).
_BASE_PREFIXES = {
    "rdf": "http://www.w3.org/1999/02/22-rdf-syntax-ns#",
    "rdfs": "http://www.w3.org/2000/01/rdf-schema#",
    "xsd": "http://www.w3.org/2001/XMLSchema#",
}


def unescape_string_literal(body: str) -> str:
    out = []
    i = 0
    while i < len(body):
        c = body[i]
        if c == "\\" and i + 1 < len(body):
            n = body[i + 1]
            mapping = {"n": "\n", "t": "\t", "r": "\r", '"': '"', "'": "'",
                       "\\": "\\", "b": "\b", "f": "\f"}
            if n in mapping:
                out.append(mapping[n])
                i += 2
                continue
            if n == "u" and i + 6 <= len(body):
                out.append(chr(int(body[i + 2:i + 6], 16)))
                i += 6
                continue
            if n == "U" and i + 10 <= len(body):
                out.append(chr(int(body[i + 2:i + 10], 16)))
                i += 10
                continue
        out.append(c)
        i += 1
    return "".join(out)


def literal_lexical_value(tok: str) -> str:
    """`"abc"@en` / `"5"^^<...int>` / `'x'` -> lexical form `abc` / `5` / `x`."""
    tok = tok.strip()
    if not tok:
        return tok
    q = tok[0]
    if q not in "\"'":
        return tok
    # long quotes
    for quote in ('"""', "'''", '"', "'"):
        if tok.startswith(quote):
            end = tok.rfind(quote)
            if end > 0:
                return unescape_string_literal(tok[len(quote):end])
    return tok


def split_quoted_triple_content(inner: str) -> Tuple[str, str, str]:
    """Split `s p o` inside << >> honoring nested << >>, <iri> and quotes."""
    parts: List[str] = []
    buf = []
    depth = 0
    in_iri = False
    in_str: Optional[str] = None
    i = 0
    while i < len(inner):
        c = inner[i]
        if in_str:
            buf.append(c)
            if c == in_str and inner[i - 1] != "\\":
                in_str = None
            i += 1
            continue
        if c in "\"'":
            in_str = c
            buf.append(c)
            i += 1
            continue
        if inner.startswith("<<", i):
            depth += 1
            buf.append("<<")
            i += 2
            continue
        if inner.startswith(">>", i):
            depth -= 1
            buf.append(">>")
            i += 2
            continue
        if c == "<" and depth == 0 and not in_iri:
            in_iri = True
            buf.append(c)
            i += 1
            continue
        if c == ">" and in_iri:
            in_iri = False
            buf.append(c)
            i += 1
            continue
        if c.isspace() and depth == 0 and not in_iri:
            if buf:
                parts.append("".join(buf))
                buf = []
            if len(parts) == 2:
                parts.append(inner[i:].strip())
                return tuple(parts)  # type: ignore
            i += 1
            continue
        buf.append(c)
        i += 1
    if buf:
        parts.append("".join(buf))
    while len(parts) < 3:
        parts.append("")
    return parts[0], parts[1], parts[2]


class SparqlDatabase:
    """The database facade.  `device` selects where columns live
    ("cpu" for host plumbing/tests, "cuda:N" for an MI355X)."""

    def __init__(self, device: str = "cpu"):
        self.device = torch.device(device)
        self.dictionary = Dictionary()
        self.quoted_triples = QuotedTripleStore()
        self.store = QuadStore(device=device)
        self.prefixes: Dict[str, str] = dict(_BASE_PREFIXES)
        self.udfs: Dict[str, Callable[..., str]] = {}
        self.rule_map: Dict[str, str] = {}  # rule name -> rule text
        self.rules: List = []  # parsed shared Rule objects registered via queries
        self.neural_models: Dict[str, dict] = {}
        self.neural_relations: Dict[str, dict] = {}
        # pred_id -> [(s,p,o)] asserted by materialize_neural_relation;
        # cleared+re-asserted on each re-materialization (ref
        # neural_relations.rs neural_materialized_triples)
        self.neural_materialized_triples: Dict[int, List] = {}
        self.train_decls: List[dict] = []
        self.probability_seeds: Dict[Tuple[int, int, int], float] = {}
        self._stats = None
        self._stats_version = -1
        self._value_col_cache: Optional[torch.Tensor] = None
        self._value_col_len = 0

    # ------------------------------------------------------------------ terms
    def resolve_prefixed(self, name: str, prefixes: Optional[Dict[str, str]] = None) -> str:
        pfx = prefixes if prefixes is not None else self.prefixes
        if name == "a":
            return RDF_TYPE
        if ":" in name:
            pre, local = name.split(":", 1)
            base = pfx.get(pre) or self.prefixes.get(pre)
            if base is not None:
                return base + local
        return name

    def resolve_lexical(self, term: str, prefixes: Optional[Dict[str, str]] = None) -> str:
        """Raw query/parser token -> canonical stored string."""
        t = term.strip()
        if t.startswith("<") and t.endswith(">") and not t.startswith("<<"):
            return unescape_string_literal(t[1:-1])
        if t[:1] in "\"'":
            return literal_lexical_value(t)
        if t.startswith("_:"):
            return t
        return self.resolve_prefixed(t, prefixes)

    def encode_term(self, term: str, prefixes: Optional[Dict[str, str]] = None) -> int:
        return self.dictionary.encode(self.resolve_lexical(term, prefixes))

    def encode_term_star(self, term: str, prefixes: Optional[Dict[str, str]] = None) -> int:
        """Encode a term that may be an RDF-star quoted triple `<< s p o >>`
        (ref sparql_database.rs:214 encode_term_star)."""
        t = term.strip()
        if t.startswith("<<") and t.endswith(">>"):
            s_s, p_s, o_s = split_quoted_triple_content(t[2:-2].strip())
            s_id = self.encode_term_star(s_s, prefixes)
            p_id = self.encode_term_star(p_s, prefixes)
            o_id = self.encode_term_star(o_s, prefixes)
            return self.quoted_triples.encode(s_id, p_id, o_id)
        return self.encode_term(t, prefixes)

    def decode_term(self, term_id: int) -> Optional[str]:
        """Recursive decode; quoted triples render as `<<s p o>>` with IRI-ish
        components bracketed (ref dictionary.rs:62 decode_term)."""
        if is_quoted_id(term_id):
            t = self.quoted_triples.decode(term_id)
            if t is None:
                return None
            parts = []
            for x in t:
                d = self.decode_term(x)
                if d is None:
                    return None
                parts.append(self._format_term_for_star(x, d))
            return "<<" + " ".join(parts) + ">>"
        return self.dictionary.decode(term_id)

    def _format_term_for_star(self, term_id: int, decoded: str) -> str:
        if is_quoted_id(term_id):
            return decoded
        if decoded.startswith("http://") or decoded.startswith("https://") or decoded.startswith("urn:"):
            return f"<{decoded}>"
        if decoded.startswith("_:"):
            return decoded
        return f'"{decoded}"'

    # ---------------------------------------------------------------- inserts
    def add_triple_parts(self, s: str, p: str, o: str):
        self.store.insert_quad(
            DEFAULT_GRAPH,
            self.encode_term_star(s),
            self.encode_term_star(p),
            self.encode_term_star(o),
        )

    # Back-compat alias matching the reference's most common test surface.
    add_triple = add_triple_parts

    def add_quad_parts(self, s: str, p: str, o: str, g: str):
        gid = DEFAULT_GRAPH if not g else self.dictionary.encode(self.resolve_lexical(g))
        self.store.insert_quad(
            gid, self.encode_term_star(s), self.encode_term_star(p), self.encode_term_star(o)
        )

    def delete_triple_parts(self, s: str, p: str, o: str) -> bool:
        """Returns True when the triple existed (ref sparql_database.rs
        delete_triple_parts -> bool)."""
        ids = [self.dictionary.lookup(self.resolve_lexical(x)) for x in (s, p, o)]
        if any(i is None for i in ids):
            return False
        existed = self.store.graph_index(DEFAULT_GRAPH).contains(*ids)
        self.store.delete_quad(DEFAULT_GRAPH, *ids)  # type: ignore
        return bool(existed)

    def triple_count(self) -> int:
        return self.store.triple_count()

    # -------------------------------------------------------------- bulk load
    def load_columns(self, s_ids, p_ids, o_ids, graph: int = DEFAULT_GRAPH):
        """Columnar bulk insert of already-encoded u32 arrays (generator /
        distributed-partition fast path — no per-triple Python)."""
        self.store.insert_bulk(graph, s_ids, p_ids, o_ids)

    # ---------------------------------------------------------------- parsing
    def parse_ntriples(self, text: str):
        from ..parsing.rdf_formats import parse_ntriples_into
        parse_ntriples_into(self, text)

    # reference naming: parse_ntriples_and_add (sparql_database.rs:1335)
    parse_ntriples_and_add = parse_ntriples

    def parse_ntriples_file(self, path: str):
        """File-to-store N-Triples ingest: the native reader + chunk-per-
        thread parser (GIL released) feed bulk dictionary interning and a
        columnar insert — no Python copy of the file text (the parallel
        bulk-parse pipeline, ref sparql_database.rs:630-804 crossbeam
        design)."""
        from ..ops import _native
        if _native is None:
            with open(path, "r", encoding="utf-8") as f:
                return self.parse_ntriples(f.read())
        from .terms import QUOTED_TRIPLE_ID_BIT
        d = self.dictionary
        mod, h = d.attach_annex()
        ids, fallback = mod.parse_ntriples_file_annex(
            path, 0, h, QUOTED_TRIPLE_ID_BIT)
        d._values_dirty = True
        if ids.numel():
            # one contiguous H2D of the whole [n,3] block, then device-side
            # column extraction (strided numpy slices would each force a
            # host-side contiguous copy before upload)
            import os as _os
            import time as _time
            dbg = _os.environ.get("KOLIBRIE_PARSE_DEBUG")
            if dbg:
                _t0 = _time.perf_counter()
            t = ids.to(self.store.device)
            if dbg:
                if str(self.store.device).startswith("cuda"):
                    torch.cuda.synchronize()
                print(f"[ingest] H2D: {_time.perf_counter()-_t0:.3f}s",
                      flush=True)
                _t0 = _time.perf_counter()
            self.store.insert_bulk(0, t[:, 0].contiguous(),
                                   t[:, 1].contiguous(),
                                   t[:, 2].contiguous())
            if dbg:
                if str(self.store.device).startswith("cuda"):
                    torch.cuda.synchronize()
                print(f"[ingest] insert_bulk: "
                      f"{_time.perf_counter()-_t0:.3f}s", flush=True)
        if fallback:
            from ..parsing.rdf_formats import _parse_ntriples_lines
            with open(path, "r", encoding="utf-8") as f:
                lines = f.read().split("\n")
            _parse_ntriples_lines(self, (lines[i] for i in fallback))

    def parse_nquads(self, text: str):
        from ..parsing.rdf_formats import parse_nquads_into
        parse_nquads_into(self, text)

    parse_nquads_and_add = parse_nquads

    def parse_turtle(self, text: str):
        from ..parsing.rdf_formats import parse_turtle_into
        parse_turtle_into(self, text)

    def parse_n3(self, text: str):
        from ..parsing.rdf_formats import parse_turtle_into
        parse_turtle_into(self, text)  # N3 triples subset shares the tokenizer

    def parse_rdf(self, xml_text: str):
        from ..parsing.rdf_formats import parse_rdf_xml_into
        parse_rdf_xml_into(self, xml_text)

    def parse_rdf_from_file(self, path: str):
        with open(path, "r", encoding="utf-8") as f:
            self.parse_rdf(f.read())

    def load_file(self, path: str):
        """Format-sniffing loader (ref python/src/py_query_builder.rs load_file)."""
        with open(path, "r", encoding="utf-8") as f:
            text = f.read()
        lower = path.lower()
        if lower.endswith((".rdf", ".xml", ".owl")):
            self.parse_rdf(text)
        elif lower.endswith(".nq"):
            self.parse_nquads(text)
        elif lower.endswith(".nt"):
            self.parse_ntriples(text)
        else:
            self.parse_turtle(text)

    # ------------------------------------------------------------- serializers
    def generate_ntriples(self) -> str:
        from .serialize import generate_ntriples
        return generate_ntriples(self)

    def generate_nquads(self) -> str:
        from .serialize import generate_nquads
        return generate_nquads(self)

    def generate_turtle(self) -> str:
        from .serialize import generate_turtle
        return generate_turtle(self)

    def generate_rdf_xml(self) -> str:
        from .serialize import generate_rdf_xml
        return generate_rdf_xml(self)

    # ------------------------------------------------------------------- UDFs
    def register_udf(self, name: str, fn: Callable[..., str]):
        """Register a user-defined function callable from BIND/FILTER
        (ref sparql_database.rs:2130)."""
        self.udfs[name.upper()] = fn

    # ------------------------------------------------------------------ stats
    def get_or_build_stats(self):
        from ..plan.stats import DatabaseStats
        if self._stats is None or self._stats_version != self.store.version:
            self._stats = DatabaseStats.gather(self)
            self._stats_version = self.store.version
        return self._stats

    def invalidate_stats_cache(self):
        """Force stats re-gather on next use (ref sparql_database.rs
        invalidate_stats_cache)."""
        self._stats = None
        self._stats_version = -1

    def build_all_indexes(self):
        """Reference-API parity (sparql_database.rs build_all_indexes):
        here the four sorted permutations are maintained on every commit,
        so this just flushes pending mutations."""
        for g in list(self.store.graphs):
            self.store.graph_index(g)

    def query_builder(self):
        """Fluent builder entry (the reference's `db.query()`; this class's
        `query(sparql)` executes SPARQL directly instead)."""
        from ..engine.query_builder import QueryBuilder
        return QueryBuilder(self)

    def decode_triple(self, triple) -> Optional[tuple]:
        """Decode an (s,p,o) id triple to strings; None if any id unknown
        (ref sparql_database.rs decode_triple)."""
        out = []
        for x in triple:
            s = self.decode_term(int(x))
            if s is None:
                return None
            out.append(s)
        return tuple(out)

    # ----------------------------------------------------------- value column
    def value_column(self) -> torch.Tensor:
        """float32 numeric value per dictionary ID, device-resident; grows
        append-only with the dictionary.  Kernels use it for ordering
        comparisons and aggregates so they never see strings."""
        n = len(self.dictionary)
        if self._value_col_cache is None or self._value_col_len < n:
            arr = self.dictionary.values_array()
            self._value_col_cache = torch.from_numpy(arr).to(self.device)
            self._value_col_len = n
        return self._value_col_cache

    # --------------------------------------------------------- quoted columns
    def quoted_columns(self):
        """Device-resident (s,p,o) int32 columns of the quoted-triple store,
        indexed by (qid & 0x7FFFFFFF) — lets kernels match RDF-star patterns
        without host round-trips."""
        n = len(self.quoted_triples)
        cache = getattr(self, "_qt_cols_cache", None)
        if cache is None or cache[0] != n:
            import numpy as np
            if n == 0:
                e = torch.empty(0, dtype=torch.int32, device=self.device)
                cols = (e, e.clone(), e.clone())
            else:
                arr = np.asarray(self.quoted_triples.id_to_triple,
                                 dtype=np.uint32).reshape(-1, 3).view(np.int32)
                t = torch.from_numpy(arr.copy()).to(self.device)
                cols = (t[:, 0].contiguous(), t[:, 1].contiguous(), t[:, 2].contiguous())
            self._qt_cols_cache = (n, cols)
            cache = self._qt_cols_cache
        return cache[1]

    # ------------------------------------------------------------------ query
    def query(self, sparql: Optional[str] = None):
        """With a SPARQL string: execute it (rows of decoded strings).
        With no argument: return a fluent QueryBuilder — the reference's
        PySparqlDatabase.query() surface (py_query_builder.rs:136)."""
        if sparql is None:
            return self.query_builder()
        from ..engine.query import execute_query
        return execute_query(sparql, self)

    def query_columns(self, sparql: str) -> Dict[str, List[str]]:
        """SELECT returning columnar results {var: [values...]} — skips
        the per-row Python list construction (engine extension; the
        row-shaped `query` keeps reference-API parity)."""
        from ..engine.query import execute_query_columns
        return execute_query_columns(sparql, self)

    # Reference public names (execute_query.rs).
    def exec_query(self, sparql: str) -> List[List[str]]:
        return self.query(sparql)

    def update(self, sparql: str):
        return self.query(sparql)

    # ------------------------------------------------------------------ union
    def union(self, other: "SparqlDatabase") -> "SparqlDatabase":
        """Re-encoding merge of two databases (ref sparql_database.rs:1819)."""
        out = SparqlDatabase(device=str(self.device))
        for db in (self, other):
            for (g, s, p, o) in db.store.all_quads():
                out.store.insert_quad(
                    out.dictionary.encode(db.dictionary.decode(g) or ""),
                    out._reencode(db, s),
                    out._reencode(db, p),
                    out._reencode(db, o),
                )
        out.store.commit_all()
        return out

    def _reencode(self, src: "SparqlDatabase", term_id: int) -> int:
        if is_quoted_id(term_id):
            t = src.quoted_triples.decode(term_id)
            assert t is not None
            return self.quoted_triples.encode(*(self._reencode(src, x) for x in t))
        return self.dictionary.encode(src.dictionary.decode(term_id) or "")

    # ------------------------------------------------------------- debug view
    def triples_as_strings(self) -> List[Tuple[str, str, str]]:
        out = []
        for (g, s, p, o) in self.store.all_quads():
            if g == DEFAULT_GRAPH:
                out.append((self.decode_term(s), self.decode_term(p), self.decode_term(o)))
        return out
