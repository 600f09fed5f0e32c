"""PyTorch-ROCm neural predicates (ref: ml/src/candle_model.rs — replaced by
torch.nn on MI355X, the one place the reference's CPU Candle MLP maps
natively onto the GPU).

MlpNeuralPredicate: MLP with binary / categorical output over f64 feature
rows pulled from SPARQL results (ml_feature_loader.rs).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn


class MlpNeuralPredicate(nn.Module):
    def __init__(self, in_dim: int, hidden: List[int], out_dim: int,
                 output_type: str = "binary", labels: Optional[List[str]] = None,
                 device: str = "cpu"):
        super().__init__()
        dims = [in_dim] + list(hidden)
        layers: List[nn.Module] = []
        for a, b in zip(dims[:-1], dims[1:]):
            layers += [nn.Linear(a, b), nn.ReLU()]
        layers.append(nn.Linear(dims[-1], out_dim))
        self.net = nn.Sequential(*layers)
        self.output_type = output_type  # binary | categorical
        self.labels = labels or []
        self.to(device)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)

    def predict_proba(self, x: torch.Tensor) -> torch.Tensor:
        logits = self.forward(x)
        if self.output_type == "binary":
            return torch.sigmoid(logits).squeeze(-1)
        return torch.softmax(logits, dim=-1)

    def save(self, path: str):
        torch.save({"state": self.state_dict(),
                    "output_type": self.output_type,
                    "labels": self.labels}, path)


def predict_rows(info: dict, rows, db):
    """Evaluate a registered model over binding rows; binds the output var."""
    from ..engine.bindings import Bindings
    model_entry = db.neural_models.get(info.get("model") or "")
    model = model_entry.get("model") if model_entry else None
    if model is None:
        raise ValueError(f"ML.PREDICT: model {info.get('model')!r} not trained")
    feat_vars = model_entry.get("features", [])
    vc = db.value_column()
    feats = []
    for v in feat_vars:
        ids = rows.col(v).to(torch.int64) & 0xFFFFFFFF
        from ..engine.tensor_utils import values_for_ids
        feats.append(values_for_ids(vc, ids).to(torch.float32))
    x = torch.stack(feats, dim=-1) if feats else torch.zeros(rows.n, 0)
    proba = model.predict_proba(x)
    out_var = info.get("output_var") or "prediction"
    ids = []
    for pv in proba.detach().cpu().tolist():
        ids.append(db.dictionary.encode(repr(pv) if isinstance(pv, float) else str(pv)))
    import torch as _t
    col = _t.tensor([i - 0x1_0000_0000 if i >= 0x8000_0000 else i for i in ids],
                    dtype=_t.int32, device=rows.device)
    return rows.with_col(out_var, col)
