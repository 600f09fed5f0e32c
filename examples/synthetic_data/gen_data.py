#!/usr/bin/env python3
"""Synthetic employee RDF/XML generator (ref:
examples/synthetic_data/gen_data.rs — TOTAL_EMPLOYEES scales arbitrarily)."""
import argparse
import random
import sys

POSITIONS = ["Manager", "Developer", "Salesperson"]
DS = "https://data.cityofchicago.org/resource/xzkq-xp2w/"
FOAF = "http://xmlns.com/foaf/0.1/"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--employees", type=int, default=10_000)
    ap.add_argument("--out", default="synthetic_employee_data.rdf")
    args = ap.parse_args()
    rng = random.Random(42)
    with open(args.out, "w", encoding="utf-8") as f:
        f.write('<?xml version="1.0" encoding="UTF-8"?>\n')
        f.write(f'<rdf:RDF xmlns:rdf="http://www.w3.org/1999/02/22-rdf-syntax-ns#" '
                f'xmlns:foaf="{FOAF}" xmlns:ds="{DS}">\n')
        for i in range(args.employees):
            f.write(f'  <rdf:Description rdf:about="{DS}employee/{i}">\n')
            f.write(f'    <name xmlns="{FOAF}">Employee {i}</name>\n')
            f.write(f'    <position xmlns="{DS}">{rng.choice(POSITIONS)}</position>\n')
            f.write(f'    <annual_salary xmlns="{DS}">{rng.randint(30000, 150000)}</annual_salary>\n')
            f.write(f'    <workplaceHomepage xmlns="{FOAF}">http://company.example/{i % 100}</workplaceHomepage>\n')
            f.write('  </rdf:Description>\n')
        f.write('</rdf:RDF>\n')
    print(f"wrote {args.employees} employees to {args.out}")


if __name__ == "__main__":
    main()
