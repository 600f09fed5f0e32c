"""CLI frontend (ref: cli/src/main.rs:15-41 — `--file <RDF/XML>
--query <SPARQL>` -> parse, execute, print)."""
from __future__ import annotations

import argparse
import sys


def main(argv=None):
    ap = argparse.ArgumentParser(
        prog="kolibrie",
        description="MI355X-native SPARQL/RDF stream-reasoning engine")
    ap.add_argument("--file", "-f", help="RDF file to load (format-sniffed)")
    ap.add_argument("--query", "-q", help="SPARQL query text")
    ap.add_argument("--query-file", help="file containing the SPARQL query")
    ap.add_argument("--device", default="cpu",
                    help="cpu or cuda:N (MI355X)")
    ap.add_argument("--format", choices=["tsv", "json"], default="tsv")
    args = ap.parse_args(argv)

    from ..storage.database import SparqlDatabase
    db = SparqlDatabase(device=args.device)
    if args.file:
        db.load_file(args.file)
    query = args.query
    if args.query_file:
        with open(args.query_file, "r", encoding="utf-8") as f:
            query = f.read()
    if not query:
        ap.error("a query is required (--query or --query-file)")
    rows = db.query(query)
    if args.format == "json":
        import json
        print(json.dumps(rows))
    else:
        for r in rows:
            print("\t".join(r))
    return 0


if __name__ == "__main__":
    sys.exit(main())
