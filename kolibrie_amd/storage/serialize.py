"""RDF serializers (ref: kolibrie/src/sparql_database.rs:457-737 —
generate_rdf_xml / generate_ntriples / generate_nquads / generate_turtle).

Round-trip contract: generate_nquads + parse_nquads reproduces the store
(the reference's persistence mechanism; SURVEY.md §5 checkpoint/resume).
"""
from __future__ import annotations

from typing import List

from .terms import is_quoted_id


def _escape(s: str) -> str:
    return (
        s.replace("\\", "\\\\")
        .replace('"', '\\"')
        .replace("\n", "\\n")
        .replace("\r", "\\r")
        .replace("\t", "\\t")
    )


def _is_iri(s: str) -> bool:
    return (
        s.startswith("http://")
        or s.startswith("https://")
        or s.startswith("urn:")
        or s.startswith("file://")
        or s.startswith("#")
        or s.startswith("mailto:")
    )


def format_term(db, term_id: int) -> str:
    """Render a term id in N-Triples surface syntax."""
    if is_quoted_id(term_id):
        t = db.quoted_triples.decode(term_id)
        if t is None:
            return "<<>>"
        return "<< " + " ".join(format_term(db, x) for x in t) + " >>"
    s = db.dictionary.decode(term_id) or ""
    if s.startswith("_:"):
        return s
    if _is_iri(s):
        return f"<{s}>"
    return f'"{_escape(s)}"'


def generate_ntriples(db) -> str:
    lines: List[str] = []
    for (g, s, p, o) in db.store.all_quads():
        if g != 0:
            continue
        lines.append(f"{format_term(db, s)} {format_term(db, p)} {format_term(db, o)} .")
    return "\n".join(lines) + ("\n" if lines else "")


def generate_nquads(db) -> str:
    lines: List[str] = []
    for (g, s, p, o) in db.store.all_quads():
        gpart = "" if g == 0 else f" <{db.dictionary.decode(g)}>"
        lines.append(
            f"{format_term(db, s)} {format_term(db, p)} {format_term(db, o)}{gpart} ."
        )
    return "\n".join(lines) + ("\n" if lines else "")


def generate_turtle(db) -> str:
    """Turtle with prefix use and `;`/`,` grouping (ref :568)."""
    prefixes = {v: k for k, v in db.prefixes.items()}
    used = {}

    def term(t_id: int) -> str:
        raw = format_term(db, t_id)
        if raw.startswith("<") and not raw.startswith("<<"):
            iri = raw[1:-1]
            for base, name in prefixes.items():
                if iri.startswith(base) and iri != base:
                    local = iri[len(base):]
                    if local and all(c.isalnum() or c in "_-." for c in local):
                        used[name] = base
                        return f"{name}:{local}"
        return raw

    by_subject = {}
    for (g, s, p, o) in db.store.all_quads():
        if g != 0:
            continue
        by_subject.setdefault(s, []).append((p, o))
    body: List[str] = []
    for s, pos_list in by_subject.items():
        by_pred = {}
        for p, o in pos_list:
            by_pred.setdefault(p, []).append(o)
        pred_parts = []
        for p, objs in by_pred.items():
            obj_txt = ", ".join(term(o) for o in objs)
            pred_parts.append(f"{term(p)} {obj_txt}")
        body.append(f"{term(s)} " + " ;\n    ".join(pred_parts) + " .")
    header = [f"@prefix {name}: <{base}> ." for name, base in sorted(used.items())]
    return "\n".join(header + ([""] if header else []) + body) + ("\n" if body else "")


def generate_rdf_xml(db) -> str:
    """Minimal RDF/XML writer (ref :457)."""
    lines = [
        '<?xml version="1.0" encoding="UTF-8"?>',
        '<rdf:RDF xmlns:rdf="http://www.w3.org/1999/02/22-rdf-syntax-ns#">',
    ]
    by_subject = {}
    for (g, s, p, o) in db.store.all_quads():
        if g != 0:
            continue
        by_subject.setdefault(s, []).append((p, o))
    for s, pos_list in by_subject.items():
        subj = db.dictionary.decode(s) or ""
        lines.append(f'  <rdf:Description rdf:about="{subj}">')
        for p, o in pos_list:
            pred = db.dictionary.decode(p) or ""
            # split namespace/local at last # or /
            cut = max(pred.rfind("#"), pred.rfind("/")) + 1
            ns, local = pred[:cut], pred[cut:] or "p"
            obj = db.dictionary.decode(o) or ""
            if _is_iri(obj):
                lines.append(f'    <{local} xmlns="{ns}" rdf:resource="{obj}"/>')
            else:
                from xml.sax.saxutils import escape
                lines.append(f'    <{local} xmlns="{ns}">{escape(obj)}</{local}>')
        lines.append("  </rdf:Description>")
    lines.append("</rdf:RDF>")
    return "\n".join(lines) + "\n"
