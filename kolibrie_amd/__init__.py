"""kolibrie_amd — an MI355X-native SPARQL/RDF stream-reasoning engine.

A from-scratch framework with the capabilities of StreamIntelligenceLab/Kolibrie
(reference surveyed in SURVEY.md), re-designed for AMD Instinct MI355X:

- dictionary-encoded int32 triple columns resident in HBM3E (288 GB/GPU),
  hash-partitioned across the 8 GPUs of one node,
- BGP scans/joins/filters/aggregates and the Datalog semi-naive fixpoint as
  hand-written CDNA4 HIP kernels (gfx950),
- distributed join shuffle and aggregate reduction as RCCL all-to-all /
  all-reduce over xGMI (torch.distributed, backend "nccl" == RCCL on ROCm),
- host orchestration (parsing, planning, provenance circuits) on CPU; strings
  never touch the GPU.

Public surface (mirrors the reference's crate layout):
  SparqlDatabase       — storage facade (ref: kolibrie/src/sparql_database.rs)
  execute_query        — SELECT/UPDATE entry (ref: kolibrie/src/execute_query.rs)
  Reasoner             — Datalog engine (ref: datalog/src/reasoning.rs)
  RSPBuilder/RSPEngine — streaming RSP (ref: kolibrie/src/rsp/*)
"""

__version__ = "0.1.0"

from .storage.database import SparqlDatabase  # noqa: F401
from .storage.terms import Term, TriplePattern, UNBOUND  # noqa: F401
from .engine.query import execute_query, execute_sparql_query  # noqa: F401
from .reasoning.reasoner import Reasoner  # noqa: F401
from .reasoning.rule import Rule  # noqa: F401


def _lazy(name):
    if name == "RSPBuilder":
        from .rsp.builder import RSPBuilder
        return RSPBuilder
    if name == "RSPEngine":
        from .rsp.engine import RSPEngine
        return RSPEngine
    if name == "QueryBuilder":
        from .engine.query_builder import QueryBuilder
        return QueryBuilder
    raise AttributeError(name)


def __getattr__(name):
    return _lazy(name)
