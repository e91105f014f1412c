"""Git smart-HTTP transport (reference git_http_server.go): a real
`git clone` and `git push` round-trip against the control plane over a
live socket, authenticated with an API key.
"""
import os
import socket
import subprocess
import threading
import time

import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture()
def live_server(tmp_path):
    import uvicorn

    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    port = _free_port()
    server = uvicorn.Server(uvicorn.Config(
        app, host="127.0.0.1", port=port, log_level="error"))
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    t0 = time.time()
    while not server.started and time.time() - t0 < 15:
        time.sleep(0.05)
    assert server.started
    yield app, f"http://127.0.0.1:{port}", port
    server.should_exit = True
    th.join(timeout=10)


def _git(cwd, *args, key=""):
    env = dict(os.environ,
               GIT_TERMINAL_PROMPT="0",
               GIT_AUTHOR_NAME="t", GIT_AUTHOR_EMAIL="t@t",
               GIT_COMMITTER_NAME="t", GIT_COMMITTER_EMAIL="t@t")
    extra = []
    if key:
        extra = ["-c", f"http.extraHeader=Authorization: Bearer {key}"]
    return subprocess.run(["git", *extra, *args], cwd=cwd, env=env,
                          capture_output=True, text=True, timeout=60)


@pytest.mark.timeout(120)
def test_clone_push_pull_roundtrip(live_server, tmp_path):
    app, base, port = live_server
    auth = app.state.auth
    me = auth.create_user("git-user")
    key = auth.create_api_key(me["id"])
    repo = app.state.git.create(me["id"], "demo")
    app.state.git.commit_files(repo["id"], {"README.md": "hello\n"},
                               "init")
    url = f"{base}/api/v1/git/repos/{repo['id']}.git"

    # clone
    r = _git(tmp_path, "clone", url, "work", key=key)
    assert r.returncode == 0, r.stderr
    wt = tmp_path / "work"
    assert (wt / "README.md").read_text() == "hello\n"

    # push a new commit
    (wt / "new.txt").write_text("pushed\n")
    assert _git(wt, "add", "new.txt", key=key).returncode == 0
    assert _git(wt, "commit", "-m", "add new").returncode == 0
    r = _git(wt, "push", "origin", "HEAD:main", key=key)
    assert r.returncode == 0, r.stderr
    # the platform sees the pushed file
    assert "new.txt" in app.state.git.ls_tree(repo["id"])

    # pull from a second clone
    r = _git(tmp_path, "clone", url, "work2", key=key)
    assert r.returncode == 0, r.stderr
    assert (tmp_path / "work2" / "new.txt").read_text() == "pushed\n"

    # unauthenticated clone rejected
    r = _git(tmp_path, "clone", url, "noauth")
    assert r.returncode != 0

    # another user's key rejected
    other_key = auth.create_api_key(auth.create_user("other")["id"])
    r = _git(tmp_path, "clone", url, "stolen", key=other_key)
    assert r.returncode != 0
