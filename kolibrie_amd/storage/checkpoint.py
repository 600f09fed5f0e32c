"""Checkpoint / resume: dump and load the partitioned store + dictionary.

Ref parity: the reference persists via serialization round
trips — generate_nquads/parse_nquads_and_add (sparql_database.rs:529,1411)
and serde on DatasetIndex with legacy-format tolerance.  Here:

  - `save_nquads` / `load_nquads`: the interchange round trip (text);
  - `save_binary` / `load_binary`: the fast path — per-graph int32 columns
    (numpy .npz) + the dictionary strings, one file per GPU partition, so
    an 8-GPU node dumps/loads its shards independently.
"""
from __future__ import annotations

import json
import os


import numpy as np
import torch


def save_nquads(db, path: str):
    with open(path, "w", encoding="utf-8") as f:
        f.write(db.generate_nquads())


def load_nquads(db, path: str):
    with open(path, "r", encoding="utf-8") as f:
        db.parse_nquads(f.read())


def save_binary(db, path: str, rank: int = 0):
    """Dump the store columns + dictionary for one partition."""
    db.store.commit_all()
    arrays = {}
    graphs = []
    for g, buf in db.store.graphs.items():
        if buf.index.n == 0 and g != 0:
            continue
        s, p, o = buf.index.columns()
        arrays[f"g{g}_s"] = s.cpu().numpy()
        arrays[f"g{g}_p"] = p.cpu().numpy()
        arrays[f"g{g}_o"] = o.cpu().numpy()
        graphs.append(g)
    meta = {
        "version": 2,
        "rank": rank,
        "graphs": graphs,
        "catalog": sorted(db.store.catalog),
        "n_quoted": len(db.quoted_triples),
    }
    np.savez_compressed(path, **arrays)
    with open(path + ".meta.json", "w", encoding="utf-8") as f:
        json.dump(meta, f)
    with open(path + ".dict", "w", encoding="utf-8") as f:
        # one JSON string per line: robust against \n, \r, backslashes and
        # any other control characters inside terms (v1's ad-hoc escaping
        # corrupted strings containing '\r' or literal "\n")
        for s in db.dictionary.iter_strings():
            f.write(json.dumps(s, ensure_ascii=False) + "\n")
    if len(db.quoted_triples):
        qt = np.asarray(db.quoted_triples.id_to_triple, dtype=np.uint32)
        np.save(path + ".qt.npy", qt)


def _unescape_v1(s: str) -> str:
    """Single left-to-right pass over v1's escaping ('\\\\' and '\\n') —
    v1's loader unescaped '\\n' before '\\\\', decoding literal
    backslash-n wrong; this processes each escape exactly once."""
    out = []
    i = 0
    while i < len(s):
        c = s[i]
        if c == "\\" and i + 1 < len(s):
            n = s[i + 1]
            if n == "n":
                out.append("\n")
                i += 2
                continue
            if n == "\\":
                out.append("\\")
                i += 2
                continue
        out.append(c)
        i += 1
    return "".join(out)


def load_binary(db, path: str):
    """Restore a partition dumped by save_binary (legacy-tolerant: missing
    catalog/quoted files default empty, ref dataset_index.rs:69-71)."""
    with open(path + ".meta.json", "r", encoding="utf-8") as f:
        meta = json.load(f)
    with open(path + ".dict", "r", encoding="utf-8") as f:
        if meta.get("version", 1) >= 2:
            strings = [json.loads(line) for line in f if line.strip()]
        else:
            strings = [_unescape_v1(line[:-1]) for line in f]
    d = db.dictionary
    if len(d) > 1:
        # re-encoding into a non-empty dictionary would remap IDs and
        # silently corrupt the restored columns; allow only an exact
        # prefix match (e.g. re-loading into the same process)
        for i, s in enumerate(strings[:len(d)]):
            if d.decode(i) != s:
                raise ValueError(
                    "load_binary into a non-empty dictionary whose entries "
                    f"differ from the checkpoint at ID {i}: restored "
                    "columns would be remapped. Load into a fresh database.")
    for s in strings:
        d.encode(s)
    qt_path = path + ".qt.npy"
    if os.path.exists(qt_path):
        qt = np.load(qt_path)
        for row in qt:
            db.quoted_triples.encode(int(row[0]), int(row[1]), int(row[2]))
    data = np.load(path if path.endswith(".npz") else path + ".npz")
    for g in meta.get("graphs", []):
        s = data[f"g{g}_s"]
        p = data[f"g{g}_p"]
        o = data[f"g{g}_o"]
        db.store.insert_bulk(int(g), s, p, o)
    for g in meta.get("catalog", []):
        db.store.create_graph(int(g))
    return meta
