"""Unit tests for the document store and the in-process pub/sub
(the reference's store + NATS layers, SURVEY.md §4 unit tier)."""
import asyncio
import threading

import pytest

from helix_amd.server import pubsub as ps
from helix_amd.store import Store


def test_store_crud_roundtrip():
    st = Store(":memory:")
    st.put("apps", "a1", {"id": "a1", "x": 1}, owner="u1")
    assert st.get("apps", "a1")["x"] == 1
    st.put("apps", "a1", {"id": "a1", "x": 2}, owner="u1")   # upsert
    assert st.get("apps", "a1")["x"] == 2
    assert st.count("apps") == 1
    assert st.delete("apps", "a1") is True
    assert st.get("apps", "a1") is None
    assert st.delete("apps", "a1") is False


def test_store_list_filters_and_order():
    st = Store(":memory:")
    for i in range(5):
        st.put("sessions", f"s{i}", {"id": f"s{i}", "n": i},
               owner="alice" if i % 2 == 0 else "bob", parent="app1")
    assert len(st.list("sessions", owner="alice")) == 3
    assert len(st.list("sessions", parent="app1")) == 5
    asc = st.list("sessions", desc=False)
    assert asc[0]["id"] == "s0"
    assert len(st.list("sessions", limit=2)) == 2


def test_store_threaded_writes():
    st = Store(":memory:")

    def writer(n):
        for i in range(50):
            st.put("llm_calls", f"c{n}-{i}", {"id": f"c{n}-{i}"}, owner="u")
    threads = [threading.Thread(target=writer, args=(n,)) for n in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert st.count("llm_calls") == 400


def test_store_persistence(tmp_path):
    path = str(tmp_path / "t.db")
    st = Store(path)
    st.put("users", "u1", {"id": "u1", "username": "x"})
    st.close()
    st2 = Store(path)
    assert st2.get("users", "u1")["username"] == "x"


def test_pubsub_topics_and_wildcards():
    async def run():
        bus = ps.PubSub()
        sub = await bus.subscribe(ps.session_queue("alice", "*"))
        other = await bus.subscribe(ps.session_queue("bob", "*"))
        await bus.publish(ps.session_queue("alice", "s1"), {"v": 1})
        topic, msg = await sub.get(timeout=2)
        assert topic.endswith(".s1") and msg["v"] == 1
        with pytest.raises(asyncio.TimeoutError):
            await other.get(timeout=0.1)
        await sub.close()
        await bus.publish(ps.session_queue("alice", "s1"), {"v": 2})
        # closed sub gets nothing; no error raised
    asyncio.run(run())


def test_pubsub_request_reply():
    async def run():
        bus = ps.PubSub()
        responder = await bus.subscribe("svc.echo")

        async def serve():
            topic, msg = await responder.get(timeout=2)
            await bus.publish(msg["reply_to"],
                              {"echo": msg["data"]["x"] * 2})
        task = asyncio.ensure_future(serve())
        out = await bus.request("svc.echo", {"x": 21}, timeout=2)
        await task
        assert out["echo"] == 42
    asyncio.run(run())


def test_pubsub_slow_consumer_drops_oldest():
    import asyncio
    from helix_amd.server.pubsub import PubSub

    async def run():
        ps = PubSub()
        sub = await ps.subscribe("t.*")
        for i in range(1500):            # > maxsize
            await ps.publish("t.x", i)
        # newest survive; oldest dropped
        topic, first = await sub.get(timeout=1)
        assert first == 1500 - 1024
        for _ in range(1023):
            _, last = await sub.get(timeout=1)
        assert last == 1499
        await sub.close()
    asyncio.run(run())


def test_store_backup_restore(tmp_path):
    from helix_amd.store import Store
    s = Store(":memory:")
    s.put("apps", "a1", {"id": "a1", "x": 1}, owner="u")
    s.put("sessions", "s1", {"id": "s1"}, owner="u")
    path = str(tmp_path / "bak.db")
    s.backup(path)
    restored = Store(path)
    assert restored.get("apps", "a1") == {"id": "a1", "x": 1}
    assert restored.get("sessions", "s1") == {"id": "s1"}


def test_admin_backup_endpoint(tmp_path):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import ServerConfig
    from helix_amd.store import Store
    cfg = ServerConfig()
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg, store=Store(":memory:"))
    client = TestClient(app)
    r = client.post("/api/v1/admin/backup", json={},
                    headers={"Authorization": "Bearer admin-key"})
    assert r.status_code == 200
    import os as _os
    assert _os.path.exists(r.json()["path"])
    assert Store(r.json()["path"]).count("users") >= 0
