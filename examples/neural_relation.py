#!/usr/bin/env python3
"""Neurosymbolic MODEL / NEURAL RELATION / TRAIN (ref:
examples/real_scenario/fraud_detection_system.rs shape)."""
import sys
sys.path.insert(0, ".")
import torch
from kolibrie_amd import SparqlDatabase

torch.manual_seed(7)
EX = "http://example.org/"
db = SparqlDatabase()
for i in range(60):
    amount = 100 + i * 50
    label = "1" if amount > 1600 else "0"
    db.add_triple(f"<{EX}tx{i}>", f"<{EX}amount>", f'"{amount}"')
    db.add_triple(f"<{EX}tx{i}>", f"<{EX}fraudLabel>", f'"{label}"')
rows = db.query(f"""
    MODEL "fraud" {{ ARCH MLP {{ HIDDEN [8] }} OUTPUT BINARY {{"suspect"}} }}
    NEURAL RELATION <{EX}suspicious> USING MODEL "fraud" {{
        INPUT {{ ?x <{EX}amount> ?a }}
        FEATURES {{ ?a }}
    }}
    TRAIN NEURAL RELATION <{EX}suspicious> USING MODEL "fraud" {{
        DATA {{ ?x <{EX}amount> ?a . ?x <{EX}fraudLabel> ?y }}
        label = ?y
        epochs = 150
    }}
    SELECT ?x WHERE {{ ?x <{EX}suspicious> "suspect" }}
""")
print(f"{len(rows)} transactions flagged; e.g. {rows[:3]}")
