"""Durable stream bus + WS bridge (VERDICT missing #3; reference
embedded NATS+JetStream, nats.go:119-163 WS listener, 608-699 durable
consumers): at-least-once delivery with ack leases and redelivery,
persistence across store reopen, cross-process access over one
WebSocket.
"""
import asyncio
import time

import pytest

from helix_amd.server.pubsub import PubSub, StreamBus
from helix_amd.store import Store


def test_publish_fetch_ack_floor(tmp_path):
    st = Store(str(tmp_path / "b.db"))
    bus = StreamBus(st, ack_wait_s=30)
    for i in range(5):
        bus.publish("jobs", "build", {"n": i})
    msgs = bus.fetch("jobs", "w1", batch=3)
    assert [m["payload"]["n"] for m in msgs] == [0, 1, 2]
    more = bus.fetch("jobs", "w1", batch=10)
    assert [m["payload"]["n"] for m in more] == [3, 4]
    for m in msgs + more:
        bus.ack("jobs", "w1", m["seq"])
    assert bus.pending("jobs", "w1") == 0
    # floor compacts: the consumer doc doesn't grow with acks
    doc = st.get("bus_consumers", "jobs:w1")
    assert doc["floor"] == 5 and doc["acked"] == []


def test_redelivery_after_lease_expiry(tmp_path):
    st = Store(str(tmp_path / "b.db"))
    bus = StreamBus(st, ack_wait_s=0.1)
    bus.publish("jobs", "build", {"n": 1})
    first = bus.fetch("jobs", "w1", batch=1)
    assert len(first) == 1
    assert bus.fetch("jobs", "w1", batch=1) == []      # leased
    time.sleep(0.15)
    again = bus.fetch("jobs", "w1", batch=1)           # lease expired
    assert len(again) == 1 and again[0]["seq"] == first[0]["seq"]
    bus.ack("jobs", "w1", again[0]["seq"])
    time.sleep(0.15)
    assert bus.fetch("jobs", "w1", batch=1) == []      # acked for good


def test_out_of_order_ack_and_subject_filter(tmp_path):
    st = Store(str(tmp_path / "b.db"))
    bus = StreamBus(st, ack_wait_s=30)
    bus.publish("jobs", "build.linux", {"n": 1})
    bus.publish("jobs", "test.linux", {"n": 2})
    bus.publish("jobs", "build.mac", {"n": 3})
    builds = bus.fetch("jobs", "w1", batch=10, subject_filter="build.*")
    assert [m["payload"]["n"] for m in builds] == [1, 3]
    bus.ack("jobs", "w1", 3)                            # out of order
    doc = st.get("bus_consumers", "jobs:w1")
    assert doc["floor"] == 0 and doc["acked"] == [3]
    bus.ack("jobs", "w1", 1)
    bus.ack("jobs", "w1", 2)
    doc = st.get("bus_consumers", "jobs:w1")
    assert doc["floor"] == 3 and doc["acked"] == []


def test_durability_across_restart(tmp_path):
    path = str(tmp_path / "b.db")
    st = Store(path)
    bus = StreamBus(st, ack_wait_s=30)
    bus.publish("jobs", "x", {"n": 1})
    bus.publish("jobs", "x", {"n": 2})
    got = bus.fetch("jobs", "w1", batch=1)
    bus.ack("jobs", "w1", got[0]["seq"])
    st.flush()
    st.close()
    # new process: same durable name resumes where it acked
    st2 = Store(path)
    bus2 = StreamBus(st2, ack_wait_s=30)
    rest = bus2.fetch("jobs", "w1", batch=10)
    assert [m["payload"]["n"] for m in rest] == [2]
    # seq allocation continues, no reuse
    assert bus2.publish("jobs", "x", {"n": 3}) == 3
    st2.close()


def test_two_consumers_independent(tmp_path):
    st = Store(str(tmp_path / "b.db"))
    bus = StreamBus(st, ack_wait_s=30)
    bus.publish("jobs", "x", {"n": 1})
    a = bus.fetch("jobs", "worker-a", batch=10)
    b = bus.fetch("jobs", "worker-b", batch=10)
    assert len(a) == 1 and len(b) == 1     # each durable gets the stream


def test_ws_bridge_pubsub_and_streams(tmp_path):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        me = auth.create_user("bus-user", admin=True)
        key = auth.create_api_key(me["id"])
        url = f"/api/v1/ws/bus?access_token={key}"
        with client.websocket_connect(url) as ws1, \
                client.websocket_connect(url) as ws2:
            # ephemeral pub/sub across connections
            ws1.send_json({"op": "sub", "pattern": "events.*"})
            ws1.send_json({"op": "ping"})
            assert ws1.receive_json()["op"] == "pong"   # sub registered
            ws2.send_json({"op": "pub", "topic": "events.deploy",
                           "payload": {"v": 7}})
            msg = ws1.receive_json()
            assert msg["topic"] == "events.deploy"
            assert msg["payload"] == {"v": 7}
            # durable stream: publish on ws2, fetch + ack on ws1
            ws2.send_json({"op": "stream_pub", "stream": "jobs",
                           "subject": "ci", "payload": {"job": 1}})
            assert ws2.receive_json()["seq"] == 1
            ws1.send_json({"op": "fetch", "stream": "jobs",
                           "durable": "ci-worker", "batch": 5})
            batch = ws1.receive_json()
            assert batch["op"] == "batch"
            assert batch["messages"][0]["payload"] == {"job": 1}
            ws1.send_json({"op": "ack", "stream": "jobs",
                           "durable": "ci-worker",
                           "seq": batch["messages"][0]["seq"]})
            ws1.send_json({"op": "fetch", "stream": "jobs",
                           "durable": "ci-worker", "batch": 5})
            assert ws1.receive_json()["messages"] == []
        # non-admin users are scoped to their own namespaces
        plain = auth.create_api_key(auth.create_user("plain")["id"])
        with client.websocket_connect(
                f"/api/v1/ws/bus?access_token={plain}") as ws:
            ws.send_json({"op": "sub", "pattern": "session.*"})
            assert ws.receive_json()["op"] == "error"
            ws.send_json({"op": "stream_pub", "stream": "jobs",
                          "subject": "x", "payload": {}})
            assert ws.receive_json()["op"] == "error"
            uid = auth.resolve(plain).id
            ws.send_json({"op": "stream_pub",
                          "stream": f"user-{uid}-tasks",
                          "subject": "x", "payload": {"n": 1}})
            assert ws.receive_json()["op"] == "pub_ack"
        # unauthenticated connections are rejected
        try:
            with client.websocket_connect("/api/v1/ws/bus") as ws:
                ws.receive_json()
            rejected = False
        except Exception:
            rejected = True
        assert rejected
