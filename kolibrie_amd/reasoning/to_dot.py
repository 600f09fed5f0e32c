"""Graphviz export of proof/derivation graphs (ref: datalog/src/reasoning/
to_dot.rs, 115 LoC)."""
from __future__ import annotations

from typing import Dict, List, Tuple

Triple = Tuple[int, int, int]


def _esc(s: str) -> str:
    return s.replace("\\", "\\\\").replace('"', '\\"')


def proof_graph_to_dot(reasoner, derived_from: Dict[Triple, List[Triple]],
                       title: str = "proof") -> str:
    """derived_from: derived triple -> premises used.  Base facts render as
    boxes, derived facts as ellipses, edges premise -> conclusion."""
    d = reasoner.dictionary

    def label(t: Triple) -> str:
        return _esc(" ".join(d.decode(x & 0xFFFFFFFF) or str(x) for x in t))

    ids: Dict[Triple, str] = {}

    def nid(t: Triple) -> str:
        if t not in ids:
            ids[t] = f"n{len(ids)}"
        return ids[t]

    lines = [f'digraph "{_esc(title)}" {{', "  rankdir=BT;"]
    derived = set(derived_from.keys())
    all_nodes = set(derived)
    for prems in derived_from.values():
        all_nodes.update(prems)
    for t in sorted(all_nodes):
        shape = "ellipse" if t in derived else "box"
        lines.append(f'  {nid(t)} [label="{label(t)}", shape={shape}];')
    for concl, prems in sorted(derived_from.items()):
        for p in prems:
            lines.append(f"  {nid(p)} -> {nid(concl)};")
    lines.append("}")
    return "\n".join(lines) + "\n"


def facts_to_dot(reasoner, title: str = "facts") -> str:
    """Render the fact graph: subjects/objects as nodes, predicates as
    edge labels."""
    d = reasoner.dictionary
    lines = [f'digraph "{_esc(title)}" {{']
    for (s, p, o) in sorted(reasoner.all_fact_tuples()):
        ls = _esc(d.decode(s) or str(s))
        lp = _esc(d.decode(p) or str(p))
        lo = _esc(d.decode(o) or str(o))
        lines.append(f'  "{ls}" -> "{lo}" [label="{lp}"];')
    lines.append("}")
    return "\n".join(lines) + "\n"
