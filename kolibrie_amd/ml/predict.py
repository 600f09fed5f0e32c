"""ML.PREDICT physical operator (ref: engine.rs:703-770 MLPredict dispatch,
ml_predict_candle.rs / ml_predict_runtime.rs)."""
from __future__ import annotations

from ..engine.bindings import Bindings


def execute_ml_predict(info: dict, rows: Bindings, db) -> Bindings:
    from .neural import predict_rows
    return predict_rows(info, rows, db)
