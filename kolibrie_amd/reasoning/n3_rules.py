"""N3 logic rule parser: `{ premise } => { conclusion } .`

Ref parity: datalog/src/parser_n3_logic.rs (360 LoC) —
parse_n3_rules_for_sds with WindowContext (predicate -> window map, window
widths); plain `{ p } => { c }` rules for SimpleR2R.load_rules.
"""
from __future__ import annotations

import re
from typing import Dict, List, Optional

from ..parsing.rdf_formats import tokenize
from ..storage.terms import Constant, TriplePattern, Variable
from .rule import Rule

_PREFIX_RE = re.compile(r"@prefix\s+([\w-]*):\s*<([^>]*)>\s*\.")


def _parse_term(tok: str, prefixes: Dict[str, str], db):
    tok = tok.strip()
    if tok.startswith("?"):
        return Variable(tok[1:])
    lex = db.resolve_lexical(tok, prefixes)
    x = db.dictionary.encode(lex) & 0xFFFFFFFF
    return Constant(x - 0x1_0000_0000 if x >= 0x8000_0000 else x)


def _parse_graph(text: str, prefixes: Dict[str, str], db) -> List[TriplePattern]:
    toks = tokenize(text)
    out: List[TriplePattern] = []
    i = 0
    while i < len(toks):
        if toks[i] == ".":
            i += 1
            continue
        if i + 2 >= len(toks):
            break
        s, p, o = toks[i], toks[i + 1], toks[i + 2]
        out.append(TriplePattern(
            _parse_term(s, prefixes, db),
            _parse_term("http://www.w3.org/1999/02/22-rdf-syntax-ns#type"
                        if p == "a" else p, prefixes, db),
            _parse_term(o, prefixes, db),
        ))
        i += 3
    return out


def _split_rules(text: str):
    """Yield (premise_text, conclusion_text, rule_start, rule_end) for each
    `{..} => {..}` (rule_end is just past the conclusion's brace)."""
    i = 0
    n = len(text)
    while i < n:
        start = text.find("{", i)
        if start < 0:
            return
        depth = 0
        j = start
        while j < n:
            if text[j] == "{":
                depth += 1
            elif text[j] == "}":
                depth -= 1
                if depth == 0:
                    break
            j += 1
        premise = text[start + 1:j]
        k = text.find("=>", j)
        if k < 0:
            return
        start2 = text.find("{", k)
        depth = 0
        j2 = start2
        while j2 < n:
            if text[j2] == "{":
                depth += 1
            elif text[j2] == "}":
                depth -= 1
                if depth == 0:
                    break
            j2 += 1
        conclusion = text[start2 + 1:j2]
        yield premise, conclusion, start, j2 + 1
        i = j2 + 1


def parse_n3_rules(text: str, db) -> List[Rule]:
    """Plain N3 rules (ref parser_n3_logic.rs `{ p } => { c }`).  A missing
    final `.` is tolerated; leftover non-whitespace that is not a rule is
    rejected (ref cross_window_tests.rs parser tests)."""
    prefixes: Dict[str, str] = {}
    for m in _PREFIX_RE.finditer(text):
        prefixes[m.group(1)] = m.group(2)
    body = _PREFIX_RE.sub("", text)
    rules: List[Rule] = []
    consumed_to = 0
    for premise_t, conclusion_t, start, end in _split_rules(body):
        gap = body[consumed_to:start].strip()
        if gap and gap.strip(".").strip():
            raise ValueError(f"n3 rules: unexpected input {gap[:40]!r}")
        rules.append(Rule(
            premise=_parse_graph(premise_t, prefixes, db),
            conclusion=_parse_graph(conclusion_t, prefixes, db),
        ))
        consumed_to = end
    tail = body[consumed_to:].strip()
    if tail and tail.strip(".").strip():
        raise ValueError(f"n3 rules: leftover input {tail[:40]!r}")
    return rules


def parse_n3_rules_for_sds(text: str, db, window_widths: Dict[str, int],
                           window_context: Optional[Dict[int, str]] = None
                           ) -> List[Rule]:
    """SDS variant: `WINDOW <iri> WIDTH n` directives map predicates to
    windows; rules stay plain datalog and the SDS translation annotates
    facts with their window of origin (ref parser_n3_logic.rs:28-36
    WindowContext)."""
    win_re = re.compile(r"WINDOW\s+<([^>]*)>\s+WIDTH\s+(\d+)", re.IGNORECASE)
    for m in win_re.finditer(text):
        window_widths[m.group(1)] = int(m.group(2))
    body = win_re.sub("", text)
    return parse_n3_rules(body, db)
