#!/usr/bin/env python3
"""RDF-star: quoted triples, annotation syntax, TRIPLE builtins
(ref: kolibrie/tests/rdf_star_test.rs shapes)."""
import sys
sys.path.insert(0, ".")
from kolibrie_amd import SparqlDatabase

db = SparqlDatabase()
db.parse_turtle("""
@prefix ex: <http://example.org/> .
ex:alice ex:knows ex:bob {| ex:certainty "0.9" ; ex:source ex:survey |} .
""")
print(db.query("""
    PREFIX ex: <http://example.org/>
    SELECT ?c WHERE { << ex:alice ex:knows ex:bob >> ex:certainty ?c }"""))
print(db.query("""
    PREFIX ex: <http://example.org/>
    SELECT ?s ?c WHERE { << ?s ex:knows ?o >> ex:certainty ?c }"""))
print(db.query("""
    PREFIX ex: <http://example.org/>
    SELECT ?t WHERE { ?t ex:certainty ?c . FILTER(isTRIPLE(?t)) }"""))
