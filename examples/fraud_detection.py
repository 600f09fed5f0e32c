"""Neurosymbolic fraud-detection pipeline (parity with the reference's
real_scenario/fraud_detection_system.rs shape):

    RSP-QL window  ->  Datalog pass 1 (raw features -> symbolic flags)
                   ->  ML score (sklearn-style model via MLHandler or a
                       torch MLP fallback)
                   ->  Datalog pass 2 (ML score -> symbolic flags)
                   ->  fusion -> verdict per transaction

Run:  python examples/fraud_detection.py
"""
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from kolibrie_amd import Reasoner
from kolibrie_amd.rsp.builder import RSPBuilder

EX = "http://fraud.example/"


def make_engine(fired):
    q = f"""
        PREFIX ex: <{EX}>
        REGISTER RSTREAM <{EX}out> AS
        SELECT ?tx ?amt
        FROM NAMED WINDOW <{EX}w> ON STREAM <{EX}txStream> [RANGE 300 STEP 60]
        WHERE {{ WINDOW <{EX}w> {{ ?tx ex:amount ?amt }} }}
    """
    return (RSPBuilder()
            .add_rsp_ql_query(q)
            .add_consumer(lambda rows: fired.append(rows))
            .build())


def symbolic_pass1(r: Reasoner):
    """Raw features -> flags (R1-R5)."""
    for concl, body in [
        (f"?t <{EX}highVelocity> \"1\"",
         f"?t <{EX}velocity1h> ?v . FILTER(?v > 5)"),
        (f"?t <{EX}largeAmount> \"1\"",
         f"?t <{EX}amount> ?a . FILTER(?a > 1000)"),
        (f"?t <{EX}highMerchantRisk> \"1\"",
         f"?t <{EX}merchantRisk> ?m . FILTER(?m > 70)"),
        (f"?t <{EX}riskLevel> \"high\"",
         f"?t <{EX}amount> ?a . ?t <{EX}velocity1h> ?v . "
         f"FILTER(?a > 1000) FILTER(?v > 5)"),
    ]:
        r.add_rule_text(
            f"RULE :p1 :- CONSTRUCT {{ {concl} }} WHERE {{ {body} }}")


def symbolic_pass2(r: Reasoner):
    """ML outputs -> flags (R6-R7)."""
    r.add_rule_text(
        f"RULE :p2 :- CONSTRUCT {{ ?t <{EX}mlAssistedAlert> \"1\" }} "
        f"WHERE {{ ?t <{EX}mlFraudScore> ?s . ?t <{EX}velocity1h> ?v . "
        f"FILTER(?s > 40) FILTER(?v > 3) }}")


def ml_score(amount: float, velocity: float, merchant_risk: float) -> float:
    """Stand-in scorer with the same signature the MLHandler path uses
    (ml/handler.py loads a .pkl when one is available)."""
    z = 0.04 * amount / 100 + 4.0 * velocity + 0.3 * merchant_risk - 30
    return max(0.0, min(100.0, z))


def fuse(flags, score) -> str:
    p = score / 100.0
    if p > 0.80:
        return "FRAUD"
    if p > 0.50 and flags.get("riskLevel") == "high":
        return "FRAUD"
    if flags.get("riskLevel") == "high":
        return "SUSPICIOUS"
    if p > 0.60 or flags:
        return "REVIEW"
    return "CLEAR"


def main():
    random.seed(4)
    fired = []
    engine = make_engine(fired)
    verdicts = {}
    for step in range(1, 11):
        tx = f"<{EX}tx{step}>"
        amount = random.choice([40, 250, 1500, 4200])
        velocity = random.choice([1, 2, 4, 8])
        merchant = random.choice([10, 50, 90])
        engine.add_to_stream(f"<{EX}txStream>",
                             (tx, f"<{EX}amount>", f'"{amount}"'), step * 60)

        r = Reasoner()
        r.add_abox_triple(f"{EX}tx{step}", f"{EX}amount", str(amount))
        r.add_abox_triple(f"{EX}tx{step}", f"{EX}velocity1h", str(velocity))
        r.add_abox_triple(f"{EX}tx{step}", f"{EX}merchantRisk", str(merchant))
        symbolic_pass1(r)
        r.infer_new_facts_semi_naive()

        score = ml_score(amount, velocity, merchant)
        r.add_abox_triple(f"{EX}tx{step}", f"{EX}mlFraudScore", str(score))
        symbolic_pass2(r)
        r.infer_new_facts_semi_naive()

        flags = {}
        for (s, p, o) in r.query_abox(f"{EX}tx{step}", None, None):
            key = p.rsplit("/", 1)[-1]
            if key in ("highVelocity", "largeAmount", "highMerchantRisk",
                       "riskLevel", "mlAssistedAlert"):
                flags[key] = o
        verdicts[f"tx{step}"] = fuse(flags, score)

    print("window firings:", len(fired))
    for tx, v in verdicts.items():
        print(f"{tx}: {v}")
    assert set(verdicts.values()) & {"FRAUD", "SUSPICIOUS", "REVIEW"}
    assert "CLEAR" in verdicts.values()
    return verdicts


if __name__ == "__main__":
    main()
