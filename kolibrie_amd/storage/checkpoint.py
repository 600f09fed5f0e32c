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
        "version": 1,
        "rank": rank,
        "graphs": graphs,
        "catalog": sorted(db.store.catalog),
        "n_quoted": len(db.quoted_triples),
    }
    np.savez_compressed(path, **arrays)
    with open(path + ".meta.json", "w", encoding="utf-8") as f:
        json.dump(meta, f)
    with open(path + ".dict", "w", encoding="utf-8") as f:
        for s in db.dictionary.id_to_str:
            f.write(s.replace("\\", "\\\\").replace("\n", "\\n") + "\n")
    if len(db.quoted_triples):
        qt = np.asarray(db.quoted_triples.id_to_triple, dtype=np.uint32)
        np.save(path + ".qt.npy", qt)


def load_binary(db, path: str):
    """Restore a partition dumped by save_binary (legacy-tolerant: missing
    catalog/quoted files default empty, ref dataset_index.rs:69-71)."""
    with open(path + ".meta.json", "r", encoding="utf-8") as f:
        meta = json.load(f)
    with open(path + ".dict", "r", encoding="utf-8") as f:
        strings = [line[:-1].replace("\\n", "\n").replace("\\\\", "\\")
                   for line in f]
    d = db.dictionary
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
