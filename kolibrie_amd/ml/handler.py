"""MLHandler: external model registry (ref: ml/src/lib.rs:48-489).

The reference embeds Python via PyO3 to load sklearn .pkl models with
companion MLSchema .ttl metadata; here the host runtime IS Python, so the
handler loads pickles directly, parses the .ttl companions with the
in-house Turtle parser, ranks models by their recorded evaluation measure
and predicts.
"""
from __future__ import annotations

import os
import pickle
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

MLS = "http://www.w3.org/ns/mls#"


@dataclass
class ModelInfo:
    name: str
    path: str
    model: object = None
    features: List[str] = field(default_factory=list)
    measure: float = 0.0          # recorded evaluation score
    task: str = ""


class MLHandler:
    def __init__(self):
        self.models: Dict[str, ModelInfo] = {}

    # ---------------------------------------------------------------- load
    def load_model(self, path: str, name: Optional[str] = None) -> ModelInfo:
        """Load a pickled model; if `<path minus ext>.ttl` exists, parse its
        MLSchema metadata (ref lib.rs:63-138)."""
        with open(path, "rb") as f:
            model = pickle.load(f)
        name = name or os.path.splitext(os.path.basename(path))[0]
        info = ModelInfo(name=name, path=path, model=model)
        ttl = os.path.splitext(path)[0] + ".ttl"
        if os.path.exists(ttl):
            self._parse_mlschema(ttl, info)
        self.models[name] = info
        return info

    def _parse_mlschema(self, ttl_path: str, info: ModelInfo):
        from ..storage.database import SparqlDatabase
        db = SparqlDatabase()
        with open(ttl_path, "r", encoding="utf-8") as f:
            db.parse_turtle(f.read())
        rows = db.query(f"""
            SELECT ?v WHERE {{ ?m <{MLS}hasQuality> ?q .
                               ?q <{MLS}hasValue> ?v }}""")
        if rows:
            try:
                info.measure = max(float(r[0]) for r in rows)
            except ValueError:
                pass
        feats = db.query(f"SELECT ?f WHERE {{ ?m <{MLS}hasInput> ?f }}")
        info.features = [r[0] for r in feats]
        task = db.query(f"SELECT ?t WHERE {{ ?m <{MLS}achieves> ?t }}")
        if task:
            info.task = task[0][0]

    def best_model(self) -> Optional[ModelInfo]:
        """Highest recorded evaluation measure (ref lib.rs model pick)."""
        if not self.models:
            return None
        return max(self.models.values(), key=lambda m: m.measure)

    # ------------------------------------------------------------- predict
    def predict(self, name: Optional[str], rows: Sequence[Sequence[float]]
                ) -> List[float]:
        info = self.models.get(name) if name else self.best_model()
        if info is None:
            raise ValueError(f"no model registered (requested {name!r})")
        model = info.model
        import numpy as np
        x = np.asarray(rows, dtype=float)
        if hasattr(model, "predict_proba"):
            proba = model.predict_proba(x)
            return [float(p[-1]) for p in proba]
        if hasattr(model, "predict"):
            return [float(v) for v in model.predict(x)]
        if callable(model):
            return [float(model(r)) for r in rows]
        raise TypeError(f"model {info.name} is not callable")
