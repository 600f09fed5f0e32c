"""SPARQL extension tour: OPTIONAL, ASK, CONSTRUCT, DESCRIBE, HAVING and
property paths (sequence / inverse / alternatives / closures).

Run:  python examples/sparql_extensions.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from kolibrie_amd import SparqlDatabase

EX = "http://ex/"


def main():
    db = SparqlDatabase()
    db.parse_turtle(f"""
    @prefix ex: <{EX}> .
    ex:alice ex:knows ex:bob . ex:bob ex:knows ex:carol .
    ex:carol ex:knows ex:dan .
    ex:alice ex:email "alice@x" .
    ex:alice ex:worksFor ex:acme . ex:acme ex:locatedIn ex:berlin .
    ex:bob ex:worksFor ex:acme .
    """)

    print("OPTIONAL:",
          db.query(f'SELECT ?p ?m WHERE {{ ?p <{EX}knows> ?q . '
                   f'OPTIONAL {{ ?p <{EX}email> ?m }} }}'))
    print("ASK:", db.query(f'ASK {{ <{EX}alice> <{EX}knows> ?x }}'))
    print("path seq:",
          db.query(f'SELECT ?c WHERE {{ <{EX}alice> '
                   f'<{EX}worksFor>/<{EX}locatedIn> ?c }}'))
    print("inverse:",
          db.query(f'SELECT ?e WHERE {{ <{EX}berlin> ^<{EX}locatedIn> ?e }}'))
    print("closure+:",
          sorted(db.query(f'SELECT ?x WHERE {{ <{EX}alice> <{EX}knows>+ ?x }}')))
    print("alts:",
          sorted(db.query(f'SELECT ?v WHERE {{ <{EX}alice> '
                          f'(<{EX}email>|<{EX}worksFor>) ?v }}')))
    print("HAVING:",
          db.query(f'SELECT ?d (COUNT(*) AS ?n) WHERE '
                   f'{{ ?e <{EX}worksFor> ?d }} GROUP BY ?d HAVING(?n > 1)'))
    print("CONSTRUCT:",
          db.query(f'CONSTRUCT {{ ?b <{EX}knownBy> ?a }} WHERE '
                   f'{{ ?a <{EX}knows> ?b }} LIMIT 2'))
    print("DESCRIBE:", db.query(f'DESCRIBE <{EX}acme>'))


if __name__ == "__main__":
    main()
