"""Sandboxed execution plane (parity role of the reference's hydra
dev-container service, api/pkg/hydra/server.go:268-355 create/list/
get/delete + exec/terminal, and the sandbox controller's per-session
workspaces, api/pkg/sandbox). The reference provisions per-session
dockerd instances over ZFS golden-image zvols; this MI355X-native
deployment runs on the inference nodes themselves, so the equivalent is
process-level sandboxes: a per-sandbox workspace directory seeded from
a golden template, command execution under its own process group with
rlimits (CPU seconds, address space, open files) and a scrubbed
environment, and a PTY terminal over WebSocket.

Spec-task implementation agents run their shell steps through this
manager (spec_tasks implement flow), which is the behavior the
reference's "implementation agents in sandboxes" provides.
"""
from __future__ import annotations

import logging
import os
import shlex
import shutil
import signal
import subprocess
import time
from typing import Dict, List, Optional

from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.sandbox")

# Environment passed into sandboxed processes — nothing inherited.
_BASE_ENV = {
    "PATH": "/usr/local/bin:/usr/bin:/bin",
    "HOME": "/tmp",
    "LANG": "C.UTF-8",
    "TERM": "xterm-256color",
}


class SandboxError(Exception):
    pass


class SandboxManager:
    def __init__(self, store, root_dir: str,
                 golden_dir: str = "",
                 max_cpu_s: int = 120,
                 max_mem_mb: int = 2048,
                 max_output_bytes: int = 256 * 1024):
        self.store = store
        self.root = os.path.abspath(root_dir)
        self.golden = golden_dir          # template tree (ZFS golden analog)
        self.max_cpu_s = max_cpu_s
        self.max_mem_mb = max_mem_mb
        self.max_output = max_output_bytes
        os.makedirs(self.root, exist_ok=True)
        self._procs: Dict[str, subprocess.Popen] = {}

    # -- lifecycle ---------------------------------------------------------
    def create(self, owner: str, name: str = "",
               session_id: str = "", template: str = "") -> dict:
        sid = new_id("sbx")
        ws = os.path.join(self.root, sid)
        os.makedirs(ws, exist_ok=False)
        tpl = template or self.golden
        if tpl and os.path.isdir(tpl):
            # golden-image seed (reference golden_zvol.go role): copy
            # the template tree into the fresh workspace
            for entry in os.listdir(tpl):
                src = os.path.join(tpl, entry)
                dst = os.path.join(ws, entry)
                if os.path.isdir(src):
                    shutil.copytree(src, dst, symlinks=True)
                else:
                    shutil.copy2(src, dst)
        doc = {"id": sid, "name": name or sid, "owner": owner,
               "session_id": session_id, "workspace": ws,
               "state": "running", "created": time.time(),
               "exec_count": 0}
        self.store.put("sandboxes", sid, doc, owner=owner,
                       parent=session_id)
        return doc

    def get(self, sid: str) -> Optional[dict]:
        return self.store.get("sandboxes", sid)

    def list(self, owner: str) -> List[dict]:
        return self.store.list("sandboxes", owner=owner)

    def delete(self, sid: str) -> bool:
        doc = self.get(sid)
        if doc is None:
            return False
        proc = self._procs.pop(sid, None)
        if proc and proc.poll() is None:
            try:
                os.killpg(proc.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                pass
        ws = doc.get("workspace", "")
        if ws and os.path.isdir(ws) and \
                os.path.abspath(ws).startswith(self.root + os.sep):
            shutil.rmtree(ws, ignore_errors=True)
        return self.store.delete("sandboxes", sid)

    # -- path containment ---------------------------------------------------
    def _resolve(self, doc: dict, rel: str) -> str:
        ws = os.path.abspath(doc["workspace"])
        p = os.path.abspath(os.path.join(ws, rel.lstrip("/")))
        if p != ws and not p.startswith(ws + os.sep):
            raise SandboxError("path escapes sandbox workspace")
        return p

    def write_file(self, sid: str, rel: str, content: bytes):
        doc = self.get(sid)
        if doc is None:
            raise SandboxError("sandbox not found")
        p = self._resolve(doc, rel)
        os.makedirs(os.path.dirname(p), exist_ok=True)
        with open(p, "wb") as fh:
            fh.write(content)

    def read_file(self, sid: str, rel: str) -> bytes:
        doc = self.get(sid)
        if doc is None:
            raise SandboxError("sandbox not found")
        with open(self._resolve(doc, rel), "rb") as fh:
            return fh.read(self.max_output)

    def list_files(self, sid: str, rel: str = "") -> List[dict]:
        doc = self.get(sid)
        if doc is None:
            raise SandboxError("sandbox not found")
        base = self._resolve(doc, rel)
        out = []
        for entry in sorted(os.listdir(base)):
            p = os.path.join(base, entry)
            out.append({"name": entry,
                        "dir": os.path.isdir(p),
                        "size": os.path.getsize(p)
                        if os.path.isfile(p) else 0})
        return out

    # -- execution -----------------------------------------------------------
    def _preexec(self):
        import resource
        os.setsid()                    # own process group → killable tree
        resource.setrlimit(resource.RLIMIT_CPU,
                           (self.max_cpu_s, self.max_cpu_s))
        mem = self.max_mem_mb * 1024 * 1024
        resource.setrlimit(resource.RLIMIT_AS, (mem, mem))
        resource.setrlimit(resource.RLIMIT_NOFILE, (256, 256))
        resource.setrlimit(resource.RLIMIT_NPROC, (128, 128))

    def exec(self, sid: str, command: str, timeout_s: float = 60,
             cwd: str = "", env: Optional[dict] = None) -> dict:
        """Run a shell command inside the sandbox workspace; returns
        {exit_code, stdout, stderr, duration_ms, timed_out}."""
        doc = self.get(sid)
        if doc is None:
            raise SandboxError("sandbox not found")
        if doc.get("state") != "running":
            raise SandboxError(f"sandbox is {doc.get('state')}")
        workdir = self._resolve(doc, cwd) if cwd else doc["workspace"]
        full_env = dict(_BASE_ENV)
        for k, v in (env or {}).items():
            if isinstance(k, str) and isinstance(v, str) and \
                    not k.startswith("LD_"):
                full_env[k] = v
        t0 = time.time()
        timed_out = False
        proc = subprocess.Popen(
            ["/bin/bash", "-c", command],
            cwd=workdir, env=full_env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            stdin=subprocess.DEVNULL,
            preexec_fn=self._preexec)
        self._procs[sid] = proc
        try:
            out, err = proc.communicate(timeout=timeout_s)
        except subprocess.TimeoutExpired:
            timed_out = True
            try:
                os.killpg(proc.pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                pass
            out, err = proc.communicate()
        finally:
            self._procs.pop(sid, None)
        doc["exec_count"] = doc.get("exec_count", 0) + 1
        doc["last_exec"] = time.time()
        self.store.put("sandboxes", sid, doc, owner=doc["owner"],
                       parent=doc.get("session_id", ""))
        return {
            "exit_code": -9 if timed_out else proc.returncode,
            "stdout": out[:self.max_output].decode("utf-8",
                                                   errors="replace"),
            "stderr": err[:self.max_output].decode("utf-8",
                                                   errors="replace"),
            "duration_ms": int((time.time() - t0) * 1000),
            "timed_out": timed_out,
        }

    # -- PTY terminal ---------------------------------------------------------
    def open_terminal(self, sid: str):
        """Spawn an interactive bash on a PTY inside the workspace;
        returns (pid, master_fd). Caller pumps bytes both ways (the WS
        terminal route; reference hydra's dev-container terminal)."""
        doc = self.get(sid)
        if doc is None:
            raise SandboxError("sandbox not found")
        import pty
        pid, master = pty.fork()
        if pid == 0:                      # child
            try:
                os.chdir(doc["workspace"])
                for k in list(os.environ):
                    del os.environ[k]
                os.environ.update(_BASE_ENV)
                os.execv("/bin/bash", ["/bin/bash", "--norc", "-i"])
            finally:
                os._exit(1)
        return pid, master
