"""Numeric feature rows from SPARQL results (ref:
kolibrie/src/ml_feature_loader.rs:21-120).

On device, feature extraction is a gather from the f64 value column by the
binding rows' ID columns — strings are never parsed per row.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch

from ..engine.bindings import Bindings


def features_from_bindings(rows: Bindings, feature_vars: Sequence[str], db
                           ) -> torch.Tensor:
    """[n, k] float32 feature matrix from binding columns."""
    vc = db.value_column()
    cols = []
    for v in feature_vars:
        if not rows.has(v):
            cols.append(torch.zeros(rows.n, dtype=torch.float64,
                                    device=rows.device))
            continue
        ids = rows.col(v).to(torch.int64) & 0xFFFFFFFF
        from ..engine.tensor_utils import values_for_ids
        cols.append(values_for_ids(vc, ids))
    if not cols:
        return torch.zeros((rows.n, 0), dtype=torch.float32)
    return torch.stack(cols, dim=-1).to(torch.float32)


def features_from_query(sparql: str, feature_vars: Sequence[str], db,
                        label_var: Optional[str] = None
                        ) -> Tuple[torch.Tensor, Optional[List[str]]]:
    """Run a SELECT and build the feature matrix (+ decoded labels)."""
    rows = db.query(sparql)
    from ..parsing.sparql import parse_combined_query
    cq = parse_combined_query(sparql)
    names = [p.output_name() for p in cq.select.variables] \
        if cq.select and cq.select.variables else []
    feats = []
    labels: Optional[List[str]] = [] if label_var else None
    for r in rows:
        d = dict(zip(names, r))
        vec = []
        for v in feature_vars:
            try:
                vec.append(float(d.get(v, "0") or 0.0))
            except ValueError:
                vec.append(0.0)
        feats.append(vec)
        if labels is not None:
            labels.append(d.get(label_var, ""))
    x = torch.tensor(feats, dtype=torch.float32) if feats else \
        torch.zeros((0, len(feature_vars)), dtype=torch.float32)
    return x, labels
