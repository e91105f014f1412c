"""Git repository service (parity with api/pkg/services git_*: server-side
bare repos under the filestore, branch listing, commit log, file access;
PR flows against external forges are config-gated and offline-safe)."""
from __future__ import annotations

import os
import subprocess
import time
from typing import List, Optional

from helix_amd.server.types import new_id

import re

_REF_RE = re.compile(r"[A-Za-z0-9][A-Za-z0-9._/\-]*")


def _safe_ref(ref: str) -> str:
    """Reject refs that could be parsed as git OPTIONS (argument
    injection: e.g. ref="--output=/tmp/x") or are malformed."""
    if not ref or ref.startswith("-") or not _REF_RE.fullmatch(ref):
        raise ValueError(f"invalid ref: {ref!r}")
    return ref


def _git(repo: str, *args: str, check: bool = True) -> str:
    res = subprocess.run(["git", *args], cwd=repo, capture_output=True,
                         text=True, timeout=60)
    if check and res.returncode != 0:
        raise RuntimeError(f"git {' '.join(args)} failed: {res.stderr}")
    return res.stdout


class GitService:
    def __init__(self, store, root: str):
        self.store = store
        self.root = os.path.join(os.path.abspath(root), "git-repositories")
        os.makedirs(self.root, exist_ok=True)

    def _path(self, repo_id: str) -> str:
        return os.path.join(self.root, f"{repo_id}.git")

    def create(self, owner: str, name: str, project_id: str = "") -> dict:
        rid = new_id("repo")
        path = self._path(rid)
        subprocess.run(["git", "init", "--bare", path], check=True,
                       capture_output=True)
        subprocess.run(["git", "symbolic-ref", "HEAD", "refs/heads/main"],
                       cwd=path, check=True, capture_output=True)
        doc = {"id": rid, "name": name, "owner": owner,
               "project_id": project_id, "path": path,
               "created": time.time()}
        self.store.put("git_repositories", rid, doc, owner=owner,
                       parent=project_id)
        return doc

    def get(self, rid: str) -> Optional[dict]:
        return self.store.get("git_repositories", rid)

    def list(self, owner: str) -> List[dict]:
        return self.store.list("git_repositories", owner=owner)

    def delete(self, rid: str) -> bool:
        import shutil
        path = self._path(rid)
        if os.path.isdir(path):
            shutil.rmtree(path)
        return self.store.delete("git_repositories", rid)

    # -- repo ops ----------------------------------------------------------
    def branches(self, rid: str) -> List[str]:
        out = _git(self._path(rid), "branch", "--list",
                   "--format=%(refname:short)")
        return [b for b in out.splitlines() if b]

    def log(self, rid: str, ref: str = "HEAD", n: int = 20) -> List[dict]:
        try:
            out = _git(self._path(rid), "log", _safe_ref(ref), f"-{int(n)}",
                       "--format=%H%x1f%an%x1f%at%x1f%s", "--")
        except ValueError:
            return []
        except RuntimeError:
            return []
        commits = []
        for line in out.splitlines():
            h, an, at, s = line.split("\x1f")
            commits.append({"hash": h, "author": an, "ts": int(at),
                            "subject": s})
        return commits

    def read_file(self, rid: str, path: str, ref: str = "HEAD") -> str:
        if path.startswith("-"):
            raise ValueError(f"invalid path: {path!r}")
        return _git(self._path(rid), "show",
                    f"{_safe_ref(ref)}:{path}")

    def ls_tree(self, rid: str, ref: str = "HEAD") -> List[str]:
        try:
            out = _git(self._path(rid), "ls-tree", "-r", "--name-only",
                       _safe_ref(ref), "--")
        except (RuntimeError, ValueError):
            return []
        return [p for p in out.splitlines() if p]

    def commit_files(self, rid: str, files: dict, message: str,
                     branch: str = "main") -> str:
        """Author a commit of {path: content} onto `branch` (used by the
        spec-task planner to write helix-specs, reference
        git_repository_service behavior)."""
        import tempfile
        bare = self._path(rid)
        _safe_ref(branch)
        with tempfile.TemporaryDirectory() as td:
            wt = os.path.join(td, "wt")
            subprocess.run(["git", "clone", "-q", bare, wt], check=True,
                           capture_output=True)
            _git(wt, "checkout", "-B", branch)
            wt_abs = os.path.abspath(wt)
            for rel, content in files.items():
                full = os.path.abspath(os.path.join(wt, str(rel)))
                # agent-authored manifests must not escape the worktree
                if not full.startswith(wt_abs + os.sep):
                    raise ValueError(f"path escapes worktree: {rel!r}")
                os.makedirs(os.path.dirname(full) or wt, exist_ok=True)
                with open(full, "w") as f:
                    f.write(str(content))
            _git(wt, "add", "-A")
            _git(wt, "-c", "user.email=agent@helix", "-c",
                 "user.name=helix-agent", "commit", "-m", message,
                 check=False)
            _git(wt, "push", "-q", "origin", branch)
            return _git(wt, "rev-parse", "HEAD").strip()

    def merge_branch(self, rid: str, branch: str,
                     into: str = "main",
                     message: str = "") -> dict:
        """Merge `branch` into `into` (the spec-task pr -> merged
        transition; reference forge-PR merge role). Returns
        {merged, commit} or raises ValueError on conflicts."""
        import tempfile
        bare = self._path(rid)
        _safe_ref(branch)
        _safe_ref(into)
        with tempfile.TemporaryDirectory() as td:
            wt = os.path.join(td, "wt")
            subprocess.run(["git", "clone", "-q", bare, wt], check=True,
                           capture_output=True)
            _git(wt, "checkout", "-B", into, f"origin/{into}",
                 check=False)
            out = subprocess.run(
                ["git", "-c", "user.email=agent@helix",
                 "-c", "user.name=helix-agent", "merge", "--no-ff",
                 "-m", message or f"merge {branch} into {into}",
                 f"origin/{branch}"],
                cwd=wt, capture_output=True, text=True)
            if out.returncode != 0:
                raise ValueError(
                    f"merge conflict: {out.stdout[-300:]}"
                    f"{out.stderr[-300:]}")
            _git(wt, "push", "-q", "origin", into)
            return {"merged": True,
                    "commit": _git(wt, "rev-parse", "HEAD").strip()}


# ---------------------------------------------------------------------------
# Smart-HTTP transport (reference api/pkg/server/git_http_server.go):
# real `git clone` / `git push` against platform repos, bridged to
# `git http-backend` (the stock CGI) with auth handled by the API.

def run_http_backend(repo_path: str, method: str, path_info: str,
                     query: str, content_type: str,
                     body: bytes) -> tuple:
    """Invoke git's CGI once; returns (status, headers, payload)."""
    env = {
        "GIT_PROJECT_ROOT": os.path.dirname(repo_path),
        "GIT_HTTP_EXPORT_ALL": "1",
        "PATH_INFO": path_info,
        "REQUEST_METHOD": method,
        "QUERY_STRING": query or "",
        "CONTENT_TYPE": content_type or "",
        "CONTENT_LENGTH": str(len(body)),
        "GATEWAY_INTERFACE": "CGI/1.1",
        "REMOTE_ADDR": "127.0.0.1",
        "REMOTE_USER": "helix",
        "PATH": os.environ.get("PATH", "/usr/bin:/bin"),
    }
    proc = subprocess.run(
        ["git", "http-backend"], input=body, env=env,
        capture_output=True, timeout=60)
    raw = proc.stdout
    sep = raw.find(b"\r\n\r\n")
    if sep < 0:
        sep = raw.find(b"\n\n")
        head, payload = raw[:sep], raw[sep + 2:]
    else:
        head, payload = raw[:sep], raw[sep + 4:]
    status = 200
    headers = {}
    for line in head.decode(errors="replace").splitlines():
        if ":" not in line:
            continue
        k, v = line.split(":", 1)
        if k.strip().lower() == "status":
            status = int(v.strip().split()[0])
        else:
            headers[k.strip()] = v.strip()
    return status, headers, payload
