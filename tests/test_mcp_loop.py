"""Loop closure: helix_amd's MCP *client* consuming helix_amd's own MCP
*gateway* over the real HTTP route — one assistant's skills become
another assistant's remote tools (the reference's MCP story both ways:
server/mcp gateway + skill/mcp client).

Also: knowledge web-crawler link following with readability extraction
(reference Chrome-pool crawler behavior at the no-JS level).
"""
import asyncio

import pytest
from fastapi.testclient import TestClient

from helix_amd.agent.skills import build_mcp_skills


@pytest.fixture()
def server(tmp_path):
    from helix_amd.server.app import create_app
    from helix_amd.server.config import ServerConfig
    from helix_amd.server.providers import MockClient, ProviderManager
    from helix_amd.store import Store
    cfg = ServerConfig()
    cfg.inference.default_provider = "mock"
    cfg.inference.default_model = "mock-model"
    cfg.filestore.path = str(tmp_path / "fs")
    store = Store(":memory:")
    pm = ProviderManager(store)
    pm.register("mock", MockClient())
    app = create_app(cfg, store=store, providers=pm)
    tc = TestClient(app)
    tc.__enter__()
    r = tc.post("/api/v1/users", json={"username": "mcp-user"},
                headers={"Authorization": "Bearer admin-key"})
    key = r.json()["api_key"]
    yield app, tc, key
    tc.__exit__(None, None, None)


class _SyncAsAsync:
    """Adapts the sync TestClient to the MCPClient's async post."""

    def __init__(self, tc, key):
        self.tc = tc
        self.key = key

    async def post(self, url, headers=None, json=None):
        h = dict(headers or {})
        h["Authorization"] = f"Bearer {self.key}"
        return self.tc.post(url, headers=h, json=json)


def test_mcp_client_against_own_gateway(server):
    app, tc, key = server
    # an app whose assistant has the calculator skill
    r = tc.post("/api/v1/apps", headers={
        "Authorization": f"Bearer {key}"}, json={
        "config": {"name": "calc-app", "assistants": [
            {"name": "a", "calculator": {"enabled": True}}]}})
    assert r.status_code == 200, r.text
    app_id = r.json()["id"]
    skills = asyncio.run(build_mcp_skills(
        {"url": f"/api/v1/mcp/{app_id}"},
        http_client=_SyncAsAsync(tc, key)))
    names = {s.name for s in skills}
    assert "mcp_calculator" in names, names
    calc = next(s for s in skills if s.name == "mcp_calculator")
    out = asyncio.run(calc.execute({"expression": "6*7"}, {}))
    assert "42" in out


PAGES = {
    "https://docs.test/start": """
<html><head><title>Start</title></head><body>
<article><p>Start page content about installation and setup of the
framework, long enough to be kept by readability scoring.</p></article>
<a href="/guide">guide</a>
<a href="https://other.host/x">offsite</a>
</body></html>""",
    "https://docs.test/guide": """
<html><head><title>Guide</title></head><body>
<article><p>The guide page explains kernels and scheduling in enough
detail to pass the extraction threshold easily.</p></article>
</body></html>""",
}


def test_crawler_follows_same_host_links(tmp_path, monkeypatch):
    from helix_amd.server.config import load_config
    from helix_amd.server.knowledge import KnowledgeReconciler

    class FakeResp:
        def __init__(self, text):
            self.text = text

    class FakeAsyncClient:
        def __init__(self, *a, **kw):
            pass

        async def __aenter__(self):
            return self

        async def __aexit__(self, *a):
            return False

        async def get(self, url, follow_redirects=True):
            return FakeResp(PAGES.get(url, "<html><body>404</body></html>"))

    import httpx
    monkeypatch.setattr(httpx, "AsyncClient", FakeAsyncClient)
    kn = KnowledgeReconciler(load_config(), None, None,
                             filestore_path=str(tmp_path))
    docs = asyncio.run(kn._crawl({
        "urls": ["https://docs.test/start"],
        "max_pages": 5, "max_depth": 1}))
    sources = [d["metadata"]["source"] for d in docs]
    assert "https://docs.test/start" in sources
    assert "https://docs.test/guide" in sources      # same-host followed
    assert all("other.host" not in s for s in sources)  # offsite skipped
    start = next(d for d in docs
                 if d["metadata"]["source"].endswith("/start"))
    assert "installation and setup" in start["text"]
    assert start["metadata"]["title"] == "Start"
    # max_pages bound respected
    docs = asyncio.run(kn._crawl({
        "urls": ["https://docs.test/start"], "max_pages": 1}))
    assert len(docs) == 1


def test_mcp_gateway_guards_private_apps(server):
    app, tc, key = server
    r = tc.post("/api/v1/apps", headers={
        "Authorization": f"Bearer {key}"}, json={
        "config": {"name": "private-app", "assistants": [
            {"name": "a", "calculator": {"enabled": True}}]}})
    app_id = r.json()["id"]
    other = app.state.auth.create_api_key(
        app.state.auth.create_user("intruder")["id"])
    r = tc.post(f"/api/v1/mcp/{app_id}",
                headers={"Authorization": f"Bearer {other}"},
                json={"jsonrpc": "2.0", "id": 1,
                      "method": "tools/list", "params": {}})
    body = r.json()
    assert "error" in body, body
    # the owner still lists tools fine
    r = tc.post(f"/api/v1/mcp/{app_id}",
                headers={"Authorization": f"Bearer {key}"},
                json={"jsonrpc": "2.0", "id": 2,
                      "method": "tools/list", "params": {}})
    assert "result" in r.json()
