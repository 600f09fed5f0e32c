#!/usr/bin/env python3
"""SPARQL basics: CRUD + SELECT composition (ref: examples/sparql_syntax/)."""
import sys
sys.path.insert(0, ".")
from kolibrie_amd import SparqlDatabase

db = SparqlDatabase()   # device="cuda:0" on an MI355X
db.query("""
    PREFIX ex: <http://example.org/>
    INSERT DATA {
        ex:alice ex:name "Alice" ; ex:age "30" ; ex:worksFor ex:acme .
        ex:bob   ex:name "Bob"   ; ex:age "45" ; ex:worksFor ex:acme .
        ex:carol ex:name "Carol" ; ex:age "22" ; ex:worksFor ex:initech .
    }""")

print(db.query("""
    PREFIX ex: <http://example.org/>
    SELECT ?n ?a WHERE { ?x ex:name ?n . ?x ex:age ?a . FILTER(?a > 25) }
    ORDER BY DESC(?a)"""))

print(db.query("""
    PREFIX ex: <http://example.org/>
    SELECT ?w (COUNT(?x) AS ?c) WHERE { ?x ex:worksFor ?w } GROUP BY ?w"""))

db.query("""
    PREFIX ex: <http://example.org/>
    DELETE { ?x ex:worksFor ex:acme } INSERT { ?x ex:worksFor ex:megacorp }
    WHERE { ?x ex:worksFor ex:acme }""")
print(db.query("PREFIX ex: <http://example.org/> SELECT ?x WHERE { ?x ex:worksFor ex:megacorp }"))
