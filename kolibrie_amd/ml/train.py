"""TRAIN NEURAL RELATION execution — end-to-end neurosymbolic training.

Ref parity: kolibrie/src/execute_ml_train.rs (592 LoC)
execute_ml_training_owned (:69-210): per epoch/batch/sample, seed the
network outputs as probabilistic facts, materialize the lineage circuit,
compile the target lineage (budgeted), exact WMC p_q, loss gradient
(CE/NLL/MSE/BCE) x wmc_gradient chain rule, surrogate backward into the
MLP, SGD/Adam step.  The reference clones the ground reasoner per sample
(:138); here the lineage is materialized once per step over the shared
fact columns — no clone.

MI355X note: the MLP itself runs on the GPU via torch; the circuit work
(lineage/SDD/WMC) is host-side, matching the reference split.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch

Triple = Tuple[int, int, int]


def execute_train_decl(decl, db, prefixes: Dict[str, str]):
    d = decl
    model_name = d.options.get("model", d.target)
    entry = db.neural_models.setdefault(model_name, {})
    feat_vars = []
    nr = db.neural_relations.get(d.target) or db.neural_relations.get(d.name)
    if nr is not None:
        feat_vars = [v for v in
                     nr["decl"].options.get("features", "").split(",") if v]
    label_var = d.options.get("label", "").lstrip("?")
    if d.data_patterns and label_var:
        _train_supervised(entry, d, db, prefixes, feat_vars or None, label_var)
    # without labeled data the decl only registers; neurosymbolic training
    # is driven explicitly via train_neurosymbolic()


def _train_supervised(entry: dict, decl, db, prefixes,
                      feat_vars: Optional[List[str]], label_var: str):
    from .feature_loader import features_from_bindings
    from .neural import MlpNeuralPredicate
    from .neural_relations import _rows_for_patterns
    patterns = [(p.s, p.p, p.o) for p in decl.data_patterns]
    rows = _rows_for_patterns(patterns, db, prefixes)
    if rows.is_empty():
        return
    if feat_vars is None:
        feat_vars = [v for v in rows.variables if v != label_var]
    x = features_from_bindings(rows, feat_vars, db)
    # standardize features: raw literal magnitudes (salaries etc.) saturate
    # the MLP otherwise; the normalization ships with the model entry
    mu = x.mean(dim=0, keepdim=True)
    sigma = x.std(dim=0, keepdim=True).clamp(min=1e-6)
    x = (x - mu) / sigma
    labels_ids = rows.col(label_var).to(torch.int64) & 0xFFFFFFFF
    vc = db.value_column()
    from ..engine.tensor_utils import values_for_ids
    y = values_for_ids(vc, labels_ids).to(torch.float32)
    hidden = [int(h) for h in
              entry.get("decl").options.get("hidden", "64,32").split(",")] \
        if entry.get("decl") is not None else [64, 32]
    model = MlpNeuralPredicate(x.shape[1], hidden, 1, "binary")
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    epochs = int(decl.options.get("epochs", 50))
    for _ in range(epochs):
        opt.zero_grad()
        p = model.predict_proba(x)
        loss = torch.nn.functional.binary_cross_entropy(p, y.clamp(0, 1))
        loss.backward()
        opt.step()
    entry["model"] = model
    entry["norm"] = (mu, sigma)
    entry["features"] = feat_vars
    entry["labels"] = (entry.get("decl").options.get("labels", "true").split("\x1f")
                       if entry.get("decl") is not None else ["true"])


def train_neurosymbolic(
    model: torch.nn.Module,
    samples: Sequence[dict],
    rules,
    optimizer: Optional[torch.optim.Optimizer] = None,
    epochs: int = 20,
    loss_kind: str = "bce",
) -> List[float]:
    """The SDD-WMC gradient bridge (ref execute_ml_train.rs:108-210).

    Each sample: {"x": [n_seeds, n_features] tensor, "seed_triples":
    [n_seeds] triples, "target": triple, "label": 0/1,
    "deterministic": set of ground triples}.

    Per step: p_i = model(x_i) become seed probabilities; the rules'
    lineage is materialized; p_q = exact WMC of the target's circuit;
    dL/dp_i = dL/dp_q * dWMC/dp_i (diff_sdd); the surrogate loss
    sum(dL/dp_i.detach() * p_i) backpropagates into the network.
    """
    from ..reasoning.diff_sdd import wmc_gradient
    from ..reasoning.hybrid import materialize_lineage
    from ..reasoning.sdd import SddManager
    if optimizer is None:
        optimizer = torch.optim.Adam(model.parameters(), lr=5e-2)
    losses: List[float] = []
    for _ in range(epochs):
        total = 0.0
        for sample in samples:
            optimizer.zero_grad()
            x = sample["x"]
            p = model.predict_proba(x)          # [n_seeds]
            probs = p.detach().cpu().tolist()
            seeds = {t: float(pi) for t, pi in
                     zip(sample["seed_triples"], probs)}
            store, nodes, weights = materialize_lineage(
                rules, seeds, sample.get("deterministic"))
            target = tuple(v & 0xFFFFFFFF for v in sample["target"])
            node = nodes.get(target, -1)
            seed_by_id = {}
            for i, (t, _pv) in enumerate(sorted(seeds.items())):
                seed_by_id[i + 1] = t
            if node == -1:
                p_q = 0.0
                grads: Dict[int, float] = {}
            else:
                manager = SddManager()
                sdd_node = store.to_sdd(node, manager, weights)
                p_q = manager.wmc(sdd_node)
                grads = wmc_gradient(manager, sdd_node)
            label = float(sample["label"])
            eps = 1e-6
            pq = min(1 - eps, max(eps, p_q))
            if loss_kind in ("bce", "ce", "nll"):
                loss_val = -(label * torch.log(torch.tensor(pq))
                             + (1 - label) * torch.log(torch.tensor(1 - pq)))
                dl_dpq = (pq - label) / (pq * (1 - pq))
            else:  # mse
                loss_val = torch.tensor((pq - label) ** 2)
                dl_dpq = 2 * (pq - label)
            # chain rule into each network output
            triple_to_idx = {t: i for i, t in
                             enumerate(sample["seed_triples"])}
            g = torch.zeros_like(p)
            for var, dwmc in grads.items():
                t = seed_by_id.get(var)
                if t is None:
                    continue
                i = triple_to_idx.get(tuple(v & 0xFFFFFFFF for v in t))
                if i is not None:
                    g[i] = dl_dpq * dwmc
            surrogate = (g.detach() * p).sum()
            surrogate.backward()
            optimizer.step()
            total += float(loss_val)
        losses.append(total / max(1, len(samples)))
    return losses
