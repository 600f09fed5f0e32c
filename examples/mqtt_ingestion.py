#!/usr/bin/env python3
"""MQTT-style stream ingestion (ref: examples/real_scenario/mqtt_example.rs —
subscribe -> parse -> add_to_stream).  The transport is pluggable; with no
broker in this environment a mock client replays recorded messages, and the
same `on_message` works with paho-mqtt's callback signature.
"""
import json
import sys
sys.path.insert(0, ".")
from kolibrie_amd.rsp import RSPBuilder

EX = "http://example.org/"
q = f"""PREFIX ex: <{EX}>
REGISTER RSTREAM <http://out> AS
SELECT ?m ?v
FROM NAMED WINDOW <http://w1> ON STREAM <http://sensors> [RANGE 10 STEP 10]
WHERE {{ WINDOW <http://w1> {{ ?m ex:reading ?v }} }}"""
eng = (RSPBuilder().add_rsp_ql_query(q)
       .add_consumer(lambda rows: print("fired:", rows)).build())


def on_message(client, userdata, msg):
    """paho-mqtt compatible callback: JSON payloads -> stream events."""
    data = json.loads(msg.payload)
    eng.add_to_stream(
        "http://sensors",
        (f"<{EX}{data['sensor']}>", f"<{EX}reading>", f'"{data["value"]}"'),
        int(data["ts"]))


class MockMsg:
    def __init__(self, payload):
        self.payload = payload


# with a real broker:  client = paho.mqtt.client.Client();
# client.on_message = on_message; client.connect(...); client.subscribe(...)
for ts in range(0, 22, 2):
    on_message(None, None, MockMsg(json.dumps(
        {"sensor": f"s{ts % 3}", "value": 20 + ts, "ts": ts})))
eng.flush_windows()
