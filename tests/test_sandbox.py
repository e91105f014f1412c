"""Sandbox execution plane (reference api/pkg/hydra dev containers +
api/pkg/sandbox): workspace isolation, golden-template seeding, rlimited
exec with scrubbed env, timeout kill of the whole process group, file IO
containment, HTTP + WS terminal surface, and spec-task implement
verification running inside a sandbox.
"""
import asyncio
import json
import os
import time

import pytest

from helix_amd.server.sandbox import SandboxError, SandboxManager
from helix_amd.store import Store


@pytest.fixture()
def mgr(tmp_path):
    store = Store(":memory:")
    golden = tmp_path / "golden"
    golden.mkdir()
    (golden / "README.md").write_text("seeded")
    (golden / "bin").mkdir()
    (golden / "bin" / "hello.sh").write_text("echo hello-from-golden")
    m = SandboxManager(store, str(tmp_path / "sbx"),
                       golden_dir=str(golden), max_cpu_s=5,
                       max_mem_mb=512)
    return store, m


def test_create_seeds_golden_and_exec(mgr):
    store, m = mgr
    sbx = m.create("u1", name="dev")
    assert os.path.isfile(os.path.join(sbx["workspace"], "README.md"))
    r = m.exec(sbx["id"], "cat README.md && bash bin/hello.sh")
    assert r["exit_code"] == 0
    assert "seeded" in r["stdout"] and "hello-from-golden" in r["stdout"]
    # cwd is the workspace; env is scrubbed
    r = m.exec(sbx["id"], "pwd; echo PATH=$PATH; echo SECRET=$HOME_SECRET",
               env={"X": "1"})
    assert sbx["workspace"] in r["stdout"]
    assert "SECRET=\n" in r["stdout"] or r["stdout"].endswith("SECRET=\n")
    # LD_* injection rejected
    r = m.exec(sbx["id"], "echo $LD_PRELOAD",
               env={"LD_PRELOAD": "/evil.so"})
    assert r["stdout"].strip() == ""


def test_exec_timeout_kills_process_group(mgr):
    store, m = mgr
    sbx = m.create("u1")
    t0 = time.time()
    r = m.exec(sbx["id"], "sleep 30 & sleep 30", timeout_s=1)
    assert r["timed_out"] and time.time() - t0 < 10
    # workspace still usable afterwards
    assert m.exec(sbx["id"], "echo ok")["stdout"].strip() == "ok"


def test_memory_rlimit(mgr):
    store, m = mgr
    sbx = m.create("u1")
    r = m.exec(sbx["id"],
               "python3 -c \"x = bytearray(2 * 1024**3)\" 2>&1; echo rc=$?",
               timeout_s=30)
    assert "rc=0" not in r["stdout"]    # allocation must fail under RLIMIT_AS


def test_file_io_and_containment(mgr):
    store, m = mgr
    sbx = m.create("u1")
    m.write_file(sbx["id"], "src/app.py", b"print('hi')")
    assert m.read_file(sbx["id"], "src/app.py") == b"print('hi')"
    names = [f["name"] for f in m.list_files(sbx["id"], "src")]
    assert names == ["app.py"]
    with pytest.raises(SandboxError):
        m.write_file(sbx["id"], "../escape.txt", b"x")
    with pytest.raises(SandboxError):
        m.read_file(sbx["id"], "../../../../etc/passwd")
    # absolute paths are workspace-relative, never host paths
    with pytest.raises(FileNotFoundError):
        m.read_file(sbx["id"], "/etc/passwd")


def test_delete_removes_workspace(mgr):
    store, m = mgr
    sbx = m.create("u1")
    ws = sbx["workspace"]
    assert m.delete(sbx["id"])
    assert not os.path.exists(ws)
    assert m.get(sbx["id"]) is None


def test_http_surface_and_terminal(tmp_path):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        me = auth.create_user("sbx-user")
        other = auth.create_user("other-user")
        key = auth.create_api_key(me["id"])
        okey = auth.create_api_key(other["id"])
        H = {"Authorization": f"Bearer {key}"}
        r = client.post("/api/v1/sandboxes", json={"name": "dev"},
                        headers=H)
        assert r.status_code == 200, r.text
        sid = r.json()["id"]
        # exec
        r = client.post(f"/api/v1/sandboxes/{sid}/exec",
                        json={"command": "echo 40+2 | bc || echo 42"},
                        headers=H)
        assert "42" in r.json()["stdout"]
        # file IO
        client.put(f"/api/v1/sandboxes/{sid}/file",
                   json={"path": "a.txt", "content": "data"}, headers=H)
        r = client.get(f"/api/v1/sandboxes/{sid}/file",
                       params={"path": "a.txt"}, headers=H)
        assert r.json()["content"] == "data"
        # ownership guard
        r = client.post(f"/api/v1/sandboxes/{sid}/exec",
                        json={"command": "id"},
                        headers={"Authorization": f"Bearer {okey}"})
        assert r.status_code == 403
        # terminal over WS
        with client.websocket_connect(
                f"/api/v1/sandboxes/{sid}/terminal?access_token={key}"
                ) as ws:
            ws.send_bytes(b"echo T$((40+2))T\n")
            buf = b""
            for _ in range(20):
                buf += ws.receive_bytes()
                if b"T42T" in buf:
                    break
            assert b"T42T" in buf
        client.delete(f"/api/v1/sandboxes/{sid}", headers=H)


def test_spec_task_implement_verifies_in_sandbox(tmp_path):
    """The implement agent's manifest verify command runs inside a
    sandbox and its result lands on the task (reference: implementation
    agents in hydra sandboxes)."""
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        me = auth.create_user("dev")
        key = auth.create_api_key(me["id"])
        H = {"Authorization": f"Bearer {key}"}
        spec_tasks = app.state.spec_tasks
        proj = spec_tasks.create_project(me["id"], "proj1")
        task = spec_tasks.create_task(me["id"], proj["id"], "add script",
                                      "write ok.sh printing OK")
        task_doc = spec_tasks.get_task(task["id"])
        task_doc["spec"] = "write ok.sh"
        app.state.store.put("spec_tasks", task["id"], task_doc,
                            owner=me["id"], parent=proj["id"])

        manifest = {"message": "add ok.sh",
                    "files": {"ok.sh": "echo OK"},
                    "verify": "bash ok.sh"}

        class FakeController:
            cfg = app.state.cfg

            async def chat_completion(self, req, owner, ctx=None,
                                      **kw):
                return {"choices": [{"message": {
                    "content": json.dumps(manifest)}}]}

        spec_tasks.controller = FakeController()
        spec_tasks.transition(task["id"], "planning")
        spec_tasks.transition(task["id"], "spec_review")
        out = asyncio.run(spec_tasks.implement(task["id"]))
        doc = spec_tasks.get_task(task["id"])
        assert doc["verify"]["ran"] is True
        assert doc["verify"]["exit_code"] == 0
        assert "OK" in doc["verify"]["output"]
        # sandbox was cleaned up
        assert app.state.sandboxes.list(me["id"]) == []


def test_spec_task_merge_completes_flow(tmp_path):
    """pr -> merged performs a real git merge of the task branch into
    main (the forge-PR role, platform-native)."""
    import asyncio
    import json as _json

    from fastapi.testclient import TestClient

    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app) as client:
        auth = app.state.auth
        me = auth.create_user("dev2")
        key = auth.create_api_key(me["id"])
        H = {"Authorization": f"Bearer {key}"}
        tasks = app.state.spec_tasks
        proj = tasks.create_project(me["id"], "p")
        rid = app.state.store.get("projects",
                                  proj["id"]).get("repo_id")
        app.state.git.commit_files(rid, {"base.txt": "base"}, "init")
        task = tasks.create_task(me["id"], proj["id"], "feature", "d")
        doc = tasks.get_task(task["id"])
        doc["spec"] = "do it"
        app.state.store.put("spec_tasks", task["id"], doc,
                            owner=me["id"], parent=proj["id"])
        manifest = {"message": "add feature",
                    "files": {"feature.txt": "done"}}

        class FakeController:
            cfg = app.state.cfg

            async def chat_completion(self, *a, **kw):
                return {"choices": [{"message": {
                    "content": _json.dumps(manifest)}}]}

        tasks.controller = FakeController()
        tasks.transition(task["id"], "planning")
        tasks.transition(task["id"], "spec_review")
        asyncio.run(tasks.implement(task["id"]))
        assert tasks.get_task(task["id"])["state"] == "pr"
        r = client.post(f"/api/v1/spec-tasks/{task['id']}/merge",
                        headers=H)
        assert r.status_code == 200, r.text
        got = tasks.get_task(task["id"])
        assert got["state"] == "merged" and got["merge_commit"]
        # main now contains both files
        tree = app.state.git.ls_tree(rid)
        assert "feature.txt" in tree and "base.txt" in tree
