#!/usr/bin/env python3
"""RSP-QL streaming with reasoning in the window (ref: examples/rsp/)."""
import sys
sys.path.insert(0, ".")
from kolibrie_amd.rsp import RSPBuilder

EX = "http://example.org/"
q = f"""PREFIX ex: <{EX}>
REGISTER RSTREAM <http://out> AS
SELECT ?m
FROM NAMED WINDOW <http://w1> ON STREAM <http://s1> [RANGE 10 STEP 10]
WHERE {{ WINDOW <http://w1> {{ ?m ex:alert "hot" }} }}"""
rules = f"""RULE :Hot :- CONSTRUCT {{ ?m <{EX}alert> "hot" }}
WHERE {{ ?m <{EX}temp> ?t . FILTER(?t > 90) }} ."""

eng = (RSPBuilder().add_rsp_ql_query(q).add_sparql_rules(rules)
       .add_consumer(lambda rows: print("window fired:", rows)).build())
for ts, (m, t) in enumerate([("m1", 95), ("m2", 50), ("m1", 99), ("m3", 91)]):
    eng.add_to_stream("http://s1", (f"<{EX}{m}>", f"<{EX}temp>", f'"{t}"'), ts * 4)
eng.flush_windows()
