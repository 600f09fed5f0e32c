"""MLSchema TTL emitter (ref: ml/src/mlschema.py — emits MLSchema metadata
for trained models so MLHandler can rank them)."""
from __future__ import annotations

from typing import Dict, Optional, Sequence

MLS = "http://www.w3.org/ns/mls#"


def emit_mlschema_ttl(model_name: str, features: Sequence[str],
                      measure_name: str, measure_value: float,
                      task: str = "classification",
                      extra: Optional[Dict[str, str]] = None) -> str:
    base = "http://kolibrie.amd/model/"
    lines = [
        f"@prefix mls: <{MLS}> .",
        f"@prefix km: <{base}> .",
        "",
        f"km:{model_name} a mls:Model ;",
        f"    mls:achieves \"{task}\" ;",
    ]
    for f in features:
        lines.append(f"    mls:hasInput \"{f}\" ;")
    lines.append(f"    mls:hasQuality km:{model_name}_q .")
    lines.append("")
    lines.append(f"km:{model_name}_q a mls:EvaluationMeasure ;")
    lines.append(f"    mls:hasMeasure \"{measure_name}\" ;")
    lines.append(f"    mls:hasValue \"{measure_value}\" .")
    for k, v in (extra or {}).items():
        lines.append(f"km:{model_name} km:{k} \"{v}\" .")
    return "\n".join(lines) + "\n"
