"""RDF data-format readers: N-Triples(-star), N-Quads, Turtle(-star), RDF/XML.

Parity surface: kolibrie/src/sparql_database.rs:630-1463 (RDF/XML pull
parser, Turtle-star tokenizer with `<< >>` depth and `{| p o |}` annotation
syntax, N3/N-Triples chunked parsers, N-Quads with optional graph term).

Host-side only: parsing produces u32 ID columns which bulk-insert into the
device store.  The tokenizer is shared across Turtle/N-Triples/N-Quads.
"""
from __future__ import annotations

import re
import xml.etree.ElementTree as ET
from typing import List, Optional, Tuple

RDF_NS = "http://www.w3.org/1999/02/22-rdf-syntax-ns#"
RDF_TYPE = RDF_NS + "type"


# --------------------------------------------------------------------- tokens
_TOKEN_RE = re.compile(
    r"""
      (?P<ws>\s+|\#[^\n]*)
    | (?P<qopen><<)
    | (?P<qclose>>>)
    | (?P<aopen>\{\|)
    | (?P<aclose>\|\})
    | (?P<iri><[^<>\s]*>)
    | (?P<literal>
        ("(?:[^"\\]|\\.)*"|'(?:[^'\\]|\\.)*')
        (?:\^\^<[^<>\s]*>|\^\^[A-Za-z_][\w.-]*:[\w.-]*|@[A-Za-z][A-Za-z0-9-]*)?
      )
    | (?P<punct>[;,.\[\]\(\)])
    | (?P<bnode>_:[A-Za-z0-9_.-]+)
    | (?P<name>[^\s;,.\[\]\(\)]+)
    """,
    re.VERBOSE | re.DOTALL,
)


def tokenize(text: str) -> List[str]:
    out = []
    pos = 0
    n = len(text)
    while pos < n:
        m = _TOKEN_RE.match(text, pos)
        if m is None:
            raise ValueError(f"tokenizer stuck at: {text[pos:pos+40]!r}")
        pos = m.end()
        if m.lastgroup == "ws":
            continue
        out.append(m.group())
    return out


class _TokenStream:
    def __init__(self, tokens: List[str]):
        self.toks = tokens
        self.i = 0

    def peek(self) -> Optional[str]:
        return self.toks[self.i] if self.i < len(self.toks) else None

    def next(self) -> str:
        if self.i >= len(self.toks):
            raise ValueError("unexpected end of RDF input")
        t = self.toks[self.i]
        self.i += 1
        return t

    def expect(self, tok: str):
        t = self.next()
        if t != tok:
            raise ValueError(f"expected {tok!r}, got {t!r}")

    def eof(self) -> bool:
        return self.i >= len(self.toks)


def _read_term(ts: _TokenStream) -> str:
    """Read one term (possibly a quoted triple) back into its surface string."""
    t = ts.peek()
    if t == "<<":
        ts.next()
        s = _read_term(ts)
        p = _read_term(ts)
        o = _read_term(ts)
        ts.expect(">>")
        return f"<< {s} {p} {o} >>"
    return ts.next()


# ------------------------------------------------------------------ N-Triples
def parse_ntriples_into(db, text: str):
    """Line-oriented N-Triples-star (ref sparql_database.rs:1345-1463).

    Bulk path: the native C++ tokenizer (ops._native.parse_ntriples_host,
    the MI355X replacement for the reference's crossbeam parse pipeline)
    interns terms locally; the unique-string table merges into the main
    dictionary and the id columns remap vectorized.  RDF-star lines fall
    back to the Python tokenizer."""
    from ..ops import _native
    if _native is not None and len(text) > 4096:
        if len(text) > (1 << 21):
            # chunk-per-thread native parse, GIL released (the MI355X
            # replacement for the reference's crossbeam pipeline)
            local_ids, strings, fallback = _native.parse_ntriples_host_mt(
                text, 0)
        else:
            local_ids, strings, fallback = _native.parse_ntriples_host(text)
        if local_ids.numel():
            decoded = [raw.decode("utf-8", "replace") for raw in strings]
            remap = db.dictionary.encode_many(decoded)
            arr = local_ids.numpy()
            mapped = remap[arr]
            db.store.insert_bulk(0, mapped[:, 0], mapped[:, 1], mapped[:, 2])
        if fallback:
            lines = text.split("\n")
            _parse_ntriples_lines(db, (lines[i] for i in fallback))
        return
    _parse_ntriples_lines(db, text.split("\n"))


def _parse_ntriples_lines(db, lines):
    for line in lines:
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        ts = _TokenStream(tokenize(line))
        s = _read_term(ts)
        p = _read_term(ts)
        o = _read_term(ts)
        if not ts.eof() and ts.peek() == ".":
            ts.next()
        db.store.insert_quad(
            0,
            db.encode_term_star(s),
            db.encode_term_star(p),
            db.encode_term_star(o),
        )


def parse_nquads_into(db, text: str):
    """N-Quads: optional 4th graph term (ref sparql_database.rs:1411-1463).

    Bulk path mirrors parse_ntriples_into: the native tokenizer interns
    terms locally (graph absent -> -1), the string table merges into the
    dictionary and rows bulk-insert per graph."""
    from ..ops import _native
    if _native is not None and len(text) > 4096:
        import numpy as np
        if len(text) > (1 << 21):
            local_ids, strings, fallback = _native.parse_nquads_host_mt(
                text, 0)
        else:
            local_ids, strings, fallback = _native.parse_nquads_host(text)
        if local_ids.numel():
            decoded = [raw.decode("utf-8", "replace") for raw in strings]
            remap = db.dictionary.encode_many(decoded).astype(np.int64)
            arr = local_ids.numpy()
            spo = remap[arr[:, :3]]
            g = arr[:, 3]
            gids = np.where(g < 0, 0, remap[np.maximum(g, 0)])
            for gid in np.unique(gids):
                m = gids == gid
                db.store.insert_bulk(int(gid) & 0xFFFFFFFF,
                                     spo[m, 0].astype(np.uint32),
                                     spo[m, 1].astype(np.uint32),
                                     spo[m, 2].astype(np.uint32))
        if fallback:
            lines = text.split("\n")
            _parse_nquads_lines(db, (lines[i] for i in fallback))
        return
    _parse_nquads_lines(db, text.split("\n"))


def _parse_nquads_lines(db, lines):
    for line in lines:
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        ts = _TokenStream(tokenize(line))
        s = _read_term(ts)
        p = _read_term(ts)
        o = _read_term(ts)
        g = None
        if not ts.eof() and ts.peek() != ".":
            g = _read_term(ts)
        gid = 0 if g is None else db.dictionary.encode(db.resolve_lexical(g))
        db.store.insert_quad(
            gid,
            db.encode_term_star(s),
            db.encode_term_star(p),
            db.encode_term_star(o),
        )


# --------------------------------------------------------------------- Turtle
def parse_turtle_into(db, text: str):
    """Turtle-star subset: @prefix/PREFIX, `a`, `;` and `,` groups, quoted
    triples, `{| p o |}` annotations (ref sparql_database.rs:965-1247)."""
    ts = _TokenStream(tokenize(text))
    local_prefixes = dict(db.prefixes)

    def resolve(term: str) -> str:
        return term

    while not ts.eof():
        t = ts.peek()
        if t in ("@prefix", "@PREFIX", "PREFIX", "prefix"):
            ts.next()
            name = ts.next()
            if name.endswith(":"):
                name = name[:-1]
            iri = ts.next()
            if iri.startswith("<") and iri.endswith(">"):
                iri = iri[1:-1]
            local_prefixes[name] = iri
            db.prefixes.setdefault(name, iri)
            if ts.peek() == ".":
                ts.next()
            continue
        if t in ("@base", "BASE", "base"):
            ts.next()
            ts.next()
            if ts.peek() == ".":
                ts.next()
            continue
        subj = _read_term(ts)
        while True:
            pred = _read_term(ts)
            while True:
                obj = _read_term(ts)
                s_id = db.encode_term_star(subj, local_prefixes)
                p_id = db.encode_term_star(pred, local_prefixes)
                o_id = db.encode_term_star(obj, local_prefixes)
                db.store.insert_quad(0, s_id, p_id, o_id)
                # annotation syntax: <s> <p> <o> {| <ap> <ao> |} — asserts
                # << s p o >> ap ao (ref tokenize_turtle_star_line:1144)
                while ts.peek() == "{|":
                    ts.next()
                    qt_id = db.quoted_triples.encode(s_id, p_id, o_id)
                    while ts.peek() not in ("|}", None):
                        ap = _read_term(ts)
                        ao = _read_term(ts)
                        db.store.insert_quad(
                            0,
                            qt_id,
                            db.encode_term_star(ap, local_prefixes),
                            db.encode_term_star(ao, local_prefixes),
                        )
                        if ts.peek() == ";":
                            ts.next()
                    ts.expect("|}")
                if ts.peek() == ",":
                    ts.next()
                    continue
                break
            if ts.peek() == ";":
                ts.next()
                if ts.peek() in (".", ";", None):
                    continue
                continue
            break
        if ts.peek() == ".":
            ts.next()


# --------------------------------------------------------------------- RDF/XML
def _expand_qname(tag: str) -> str:
    """ElementTree gives tags as {namespace}local -> namespace+local."""
    if tag.startswith("{"):
        ns, local = tag[1:].split("}", 1)
        return ns + local
    return tag


def parse_rdf_xml_into(db, xml_text: str):
    """RDF/XML reader (ref sparql_database.rs:630-963).  Supports
    rdf:Description/typed nodes, rdf:about/ID/nodeID, property elements with
    rdf:resource, nested nodes and text content.  Bulk-encodes into columns.
    """
    root = ET.fromstring(xml_text)
    triples: List[Tuple[str, str, str]] = []
    bnode_counter = [0]

    def fresh_bnode() -> str:
        bnode_counter[0] += 1
        return f"_:genid{bnode_counter[0]}"

    def node_id(el) -> str:
        about = el.get(f"{{{RDF_NS}}}about")
        if about is not None:
            return about
        rid = el.get(f"{{{RDF_NS}}}ID")
        if rid is not None:
            return "#" + rid
        nid = el.get(f"{{{RDF_NS}}}nodeID")
        if nid is not None:
            return "_:" + nid
        return fresh_bnode()

    def walk_node(el) -> str:
        subj = node_id(el)
        tag = _expand_qname(el.tag)
        if tag != RDF_NS + "Description":
            triples.append((subj, RDF_TYPE, tag))
        for k, v in el.attrib.items():
            ek = _expand_qname(k)
            if ek.startswith(RDF_NS):
                continue
            triples.append((subj, ek, v))
        for prop in el:
            pred = _expand_qname(prop.tag)
            res = prop.get(f"{{{RDF_NS}}}resource")
            nid = prop.get(f"{{{RDF_NS}}}nodeID")
            children = list(prop)
            if res is not None:
                triples.append((subj, pred, res))
            elif nid is not None:
                triples.append((subj, pred, "_:" + nid))
            elif children:
                for child in children:
                    triples.append((subj, pred, walk_node(child)))
            else:
                text = (prop.text or "").strip()
                triples.append((subj, pred, text))
        return subj

    root_tag = _expand_qname(root.tag)
    nodes = list(root) if root_tag == RDF_NS + "RDF" else [root]
    for el in nodes:
        walk_node(el)

    # bulk encode -> columnar insert (GPU-side dedup/sort on commit);
    # encode_many's tuned intern loop replaces per-term encode() calls
    n = len(triples)
    if n:
        flat = [t for tr in triples for t in tr]
        ids = db.dictionary.encode_many(flat).reshape(-1, 3)
        db.store.insert_bulk(0, ids[:, 0].copy(), ids[:, 1].copy(),
                             ids[:, 2].copy())
