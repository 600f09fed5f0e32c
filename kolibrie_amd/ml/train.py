"""TRAIN NEURAL RELATION execution (ref: kolibrie/src/execute_ml_train.rs).

Full neurosymbolic training (SDD-WMC gradient bridge) lands with the
provenance phase; this module wires the declaration path so queries with
TRAIN decls register and train the PyTorch MLP on feature rows.
"""
from __future__ import annotations

from typing import Dict


def execute_train_decl(decl, db, prefixes: Dict[str, str]):
    from .neural import MlpNeuralPredicate, train_from_patterns
    d = decl
    model_name = d.options.get("model", d.target)
    entry = db.neural_models.setdefault(model_name, {})
    try:
        train_from_patterns(entry, d, db, prefixes)
    except NotImplementedError:
        # declaration registered; training deferred until features exist
        pass
