"""External-agent bridge (reference api/pkg/external-agent +
controller_external_agent.go): registry dispatch with idle-reset
timeouts, agent-runner routing for agent_type=zed_external, WS auth.
"""
import asyncio

import pytest

from helix_amd.server.external_agent import (ExternalAgentError,
                                             ExternalAgentRegistry)
from helix_amd.server.types import AssistantConfig


def test_turn_streams_until_done():
    async def scenario():
        reg = ExternalAgentRegistry()
        outbox = reg.attach("sess-1")

        async def agent():
            cmd = await outbox.get()
            assert cmd["type"] == "chat_message"
            assert cmd["messages"][0]["content"] == "hi"
            rid = cmd["request_id"]
            for part in ("Hel", "lo ", "world"):
                reg.deliver({"type": "delta", "request_id": rid,
                             "content": part})
                await asyncio.sleep(0)
            reg.deliver({"type": "done", "request_id": rid})

        task = asyncio.ensure_future(agent())
        chunks = []
        async for c in reg.run_turn("sess-1", {
                "messages": [{"role": "user", "content": "hi"}]}):
            chunks.append(c)
        await task
        assert "".join(chunks) == "Hello world"
        # response channel cleaned up
        assert reg._responses == {}

    asyncio.run(scenario())


def test_idle_timeout_resets_per_chunk():
    async def scenario():
        reg = ExternalAgentRegistry()
        outbox = reg.attach("s")

        async def slow_agent():
            cmd = await outbox.get()
            rid = cmd["request_id"]
            # three chunks each arriving just under the idle limit:
            # total time exceeds a single idle window, but the turn
            # must survive (idle RESETS per chunk — reference
            # controller_external_agent.go:16-24 rationale)
            for i in range(3):
                await asyncio.sleep(0.08)
                reg.deliver({"type": "delta", "request_id": rid,
                             "content": str(i)})
            reg.deliver({"type": "done", "request_id": rid})

        task = asyncio.ensure_future(slow_agent())
        chunks = [c async for c in reg.run_turn(
            "s", {}, idle_timeout_s=0.15)]
        await task
        assert chunks == ["0", "1", "2"]

        # and a genuinely idle agent trips the timeout
        outbox2 = reg.attach("s2")

        async def dead_agent():
            await outbox2.get()          # reads the command, says nothing

        t2 = asyncio.ensure_future(dead_agent())
        with pytest.raises(ExternalAgentError, match="idle"):
            async for _ in reg.run_turn("s2", {}, idle_timeout_s=0.1):
                pass
        await t2

    asyncio.run(scenario())


def test_error_and_disconnected():
    async def scenario():
        reg = ExternalAgentRegistry()
        with pytest.raises(ExternalAgentError, match="not connected"):
            async for _ in reg.run_turn("ghost", {}):
                pass
        outbox = reg.attach("s")

        async def failing_agent():
            cmd = await outbox.get()
            reg.deliver({"type": "error",
                         "request_id": cmd["request_id"],
                         "message": "compile failed"})

        task = asyncio.ensure_future(failing_agent())
        with pytest.raises(ExternalAgentError, match="compile failed"):
            async for _ in reg.run_turn("s", {}):
                pass
        await task

    asyncio.run(scenario())


def test_agent_runner_routes_external(tmp_path):
    from helix_amd.agent.runner import AgentRunner
    from helix_amd.server.config import load_config
    from helix_amd.store import Store

    async def scenario():
        ar = AgentRunner(load_config(), Store(":memory:"), None, None)
        reg = ExternalAgentRegistry()
        ar.external_agents = reg
        outbox = reg.attach("sess-9")

        async def agent():
            cmd = await outbox.get()
            rid = cmd["request_id"]
            reg.deliver({"type": "delta", "request_id": rid,
                         "content": "external says hi"})
            reg.deliver({"type": "done", "request_id": rid})

        task = asyncio.ensure_future(agent())
        asst = AssistantConfig(name="z", agent_type="zed_external")
        resp = await ar.run_blocking(
            asst, {"model": "m",
                   "messages": [{"role": "user", "content": "go"}]},
            "u1", {"session_id": "sess-9"})
        await task
        assert resp["choices"][0]["message"]["content"] == \
            "external says hi"

        # streaming variant
        outbox2 = reg.attach("sess-9")

        async def agent2():
            cmd = await outbox2.get()
            rid = cmd["request_id"]
            reg.deliver({"type": "delta", "request_id": rid,
                         "content": "st"})
            reg.deliver({"type": "delta", "request_id": rid,
                         "content": "ream"})
            reg.deliver({"type": "done", "request_id": rid})

        t2 = asyncio.ensure_future(agent2())
        parts = []
        async for chunk in ar.run_stream(
                asst, {"model": "m", "messages": []}, "u1",
                {"session_id": "sess-9"}):
            d = chunk["choices"][0]["delta"]
            if d.get("content"):
                parts.append(d["content"])
        await t2
        assert "".join(parts) == "stream"

    asyncio.run(scenario())


def test_ws_route_auth(tmp_path):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        key = auth.create_api_key(auth.create_user("x")["id"])
        # bad token rejected
        try:
            with client.websocket_connect(
                    "/api/v1/external-agents/ws?agent_id=a") as ws:
                ws.receive_json()
            ok = False
        except Exception:
            ok = True
        assert ok
        # a random agent_id not owned by the caller is rejected
        try:
            with client.websocket_connect(
                    f"/api/v1/external-agents/ws?agent_id=sess-7"
                    f"&access_token={key}") as ws:
                ws.receive_json()
            hijack_blocked = False
        except Exception:
            hijack_blocked = True
        assert hijack_blocked
        # owning a session grants the uplink; ping/pong works
        sess = app.state.controller.create_session(
            auth.resolve(key).id, name="mine")
        with client.websocket_connect(
                f"/api/v1/external-agents/ws?agent_id={sess.id}"
                f"&access_token={key}") as ws:
            ws.send_json({"type": "ping"})
            assert ws.receive_json() == {"type": "pong"}
