"""Spec-task system (parity with api/pkg/services spec_*: the kanban
pipeline Backlog -> Planning -> SpecReview -> InProgress -> PR -> Merged,
driven by planning/implementation agents; design docs written into the
project's git repo)."""
from __future__ import annotations

import logging
import time
from typing import List, Optional

from helix_amd.server.types import new_id

log = logging.getLogger("helix_amd.spec_tasks")

STATES = ["backlog", "planning", "spec_review", "in_progress", "pr",
          "merged", "failed"]
# allowed transitions
_NEXT = {
    "backlog": {"planning"},
    "planning": {"spec_review", "failed"},
    "spec_review": {"in_progress", "planning", "failed"},
    "in_progress": {"pr", "failed"},
    "pr": {"merged", "in_progress", "failed"},
    "merged": set(),
    "failed": {"backlog"},
}


class SpecTaskService:
    def __init__(self, store, controller=None, git=None, sandboxes=None):
        self.sandboxes = sandboxes
        self.store = store
        self.controller = controller
        self.git = git

    # -- projects ----------------------------------------------------------
    def create_project(self, owner: str, name: str) -> dict:
        pid = new_id("proj")
        doc = {"id": pid, "name": name, "owner": owner,
               "created": time.time()}
        self.store.put("projects", pid, doc, owner=owner)
        if self.git is not None:
            repo = self.git.create(owner, name, project_id=pid)
            doc["repo_id"] = repo["id"]
            self.store.put("projects", pid, doc, owner=owner)
        return doc

    def list_projects(self, owner: str) -> List[dict]:
        return self.store.list("projects", owner=owner)

    # -- tasks -------------------------------------------------------------
    def create_task(self, owner: str, project_id: str, title: str,
                    description: str = "") -> dict:
        tid = new_id("task")
        doc = {"id": tid, "project_id": project_id, "owner": owner,
               "title": title, "description": description,
               "state": "backlog", "spec": "", "comments": [],
               "created": time.time(), "updated": time.time()}
        self.store.put("spec_tasks", tid, doc, owner=owner,
                       parent=project_id)
        return doc

    def get_task(self, tid: str) -> Optional[dict]:
        return self.store.get("spec_tasks", tid)

    def list_tasks(self, project_id: str) -> List[dict]:
        return self.store.list("spec_tasks", parent=project_id, desc=False)

    def transition(self, tid: str, new_state: str) -> dict:
        doc = self.get_task(tid)
        if doc is None:
            raise KeyError(tid)
        if new_state not in STATES:
            raise ValueError(f"unknown state: {new_state}")
        if new_state not in _NEXT[doc["state"]]:
            raise ValueError(
                f"illegal transition {doc['state']} -> {new_state}")
        doc["state"] = new_state
        doc["updated"] = time.time()
        self.store.put("spec_tasks", tid, doc, owner=doc["owner"],
                       parent=doc["project_id"])
        return doc

    def add_comment(self, tid: str, author: str, text: str) -> dict:
        doc = self.get_task(tid)
        doc["comments"].append({"author": author, "text": text,
                                "ts": time.time()})
        self.store.put("spec_tasks", tid, doc, owner=doc["owner"],
                       parent=doc["project_id"])
        return doc

    # -- planning agent (reference spec_task_orchestrator planning phase) --
    async def plan(self, tid: str, model: str = "") -> dict:
        doc = self.transition(tid, "planning")
        if self.controller is None:
            raise RuntimeError("controller unavailable")
        req = {
            "model": model or self.controller.cfg.inference.default_model,
            "messages": [
                {"role": "system",
                 "content": "You are a software planning agent. Write a "
                            "concise implementation spec in markdown."},
                {"role": "user",
                 "content": f"Task: {doc['title']}\n\n{doc['description']}"},
            ],
        }
        try:
            resp = await self.controller.chat_completion(
                req, doc["owner"], ctx={"owner": doc["owner"],
                                        "step": "spec_plan"})
            spec = resp["choices"][0]["message"]["content"]
            doc = self.get_task(tid)
            doc["spec"] = spec
            self.store.put("spec_tasks", tid, doc, owner=doc["owner"],
                           parent=doc["project_id"])
            # write the spec into the project repo (helix-specs authoring)
            project = self.store.get("projects", doc["project_id"])
            if self.git is not None and project and project.get("repo_id"):
                self.git.commit_files(
                    project["repo_id"],
                    {f"specs/{tid}.md": spec},
                    f"spec: {doc['title']}", branch="helix-specs")
            return self.transition(tid, "spec_review")
        except Exception:
            log.exception("planning failed for %s", tid)
            return self.transition(tid, "failed")

    # -- implementation agent (reference spec_task_orchestrator impl
    # phase: implementation runs in sandboxes; here the process-level
    # SandboxManager replaces the per-session dockerd dev container) --
    async def implement(self, tid: str, model: str = "") -> dict:
        """Drive the approved spec to a commit on a task branch: the LLM
        emits a JSON file-manifest which is materialized into a sandbox
        workspace alongside the repo tree; the manifest's optional
        `verify` command runs inside the sandbox (rlimited, scrubbed
        env) and its outcome is recorded on the task before the files
        are committed to `helix/task-{id}`; the task moves
        in_progress -> pr."""
        import json as _json
        doc = self.transition(tid, "in_progress")
        if self.controller is None or self.git is None:
            raise RuntimeError("controller/git unavailable")
        project = self.store.get("projects", doc["project_id"])
        rid = (project or {}).get("repo_id")
        if not rid:
            raise RuntimeError("project has no repo")
        tree = []
        try:
            tree = self.git.ls_tree(rid)
        except Exception:
            pass
        req = {
            "model": model or self.controller.cfg.inference.default_model,
            "messages": [
                {"role": "system",
                 "content": "You are an implementation agent. Respond "
                            "with ONLY a JSON object: {\"message\": "
                            "\"commit message\", \"files\": {\"path\": "
                            "\"full file content\", ...}}."},
                {"role": "user",
                 "content": f"Spec:\n{doc['spec'] or doc['description']}"
                            f"\n\nExisting repo files: {tree[:100]}"},
            ],
        }
        try:
            resp = await self.controller.chat_completion(
                req, doc["owner"], ctx={"owner": doc["owner"],
                                        "step": "spec_implement"})
            text = resp["choices"][0]["message"]["content"]
            # tolerate fenced or prefixed JSON
            start = text.find("{")
            end = text.rfind("}")
            manifest = _json.loads(text[start:end + 1]) \
                if start >= 0 else {}
            files = manifest.get("files") or {}
            if not isinstance(files, dict) or not files:
                # the model produced no manifest: record the raw output
                files = {f"tasks/{tid}/output.md": text}
            msg = manifest.get("message") or f"task: {doc['title']}"
            verify = None
            if self.sandboxes is not None:
                verify = self._sandbox_verify(doc, rid, tree, files,
                                              manifest.get("verify", ""))
            branch = f"helix/task-{tid[-8:]}"
            self.git.commit_files(rid, files, msg, branch=branch)
            doc = self.get_task(tid)
            doc["branch"] = branch
            if verify is not None:
                doc["verify"] = verify
            self.store.put("spec_tasks", tid, doc, owner=doc["owner"],
                           parent=doc["project_id"])
            return self.transition(tid, "pr")
        except Exception:
            log.exception("implementation failed for %s", tid)
            return self.transition(tid, "failed")

    def _sandbox_verify(self, doc, rid, tree, files, verify_cmd):
        """Materialize repo + manifest into a fresh sandbox and run the
        verify command there (reference: implementation agents execute
        in hydra sandboxes, spec_task_orchestrator.go)."""
        sbx = self.sandboxes.create(doc["owner"],
                                    name=f"task-{doc['id'][-8:]}",
                                    session_id=doc["id"])
        try:
            for path in tree[:500]:
                try:
                    content = self.git.read_file(rid, path)
                except Exception:
                    continue
                self.sandboxes.write_file(sbx["id"], path,
                                          content.encode())
            for path, content in files.items():
                self.sandboxes.write_file(
                    sbx["id"], path,
                    content.encode() if isinstance(content, str)
                    else content)
            if not verify_cmd:
                return {"ran": False}
            result = self.sandboxes.exec(sbx["id"], verify_cmd,
                                         timeout_s=120)
            return {"ran": True, "command": verify_cmd,
                    "exit_code": result["exit_code"],
                    "timed_out": result["timed_out"],
                    "output": (result["stdout"] +
                               result["stderr"])[-2000:]}
        except Exception as e:
            log.warning("sandbox verify failed for %s: %s",
                        doc["id"], e)
            return {"ran": False, "error": str(e)}
        finally:
            self.sandboxes.delete(sbx["id"])

    def merge(self, tid: str) -> dict:
        """pr -> merged with a REAL git merge of the task branch
        (reference: PR merge completes the spec-task flow)."""
        doc = self.get_task(tid)
        if doc is None:
            raise KeyError(tid)
        if doc.get("state") != "pr":
            raise ValueError(f"task is {doc.get('state')}, not pr")
        branch = doc.get("branch", "")
        project = self.store.get("projects", doc["project_id"])
        rid = (project or {}).get("repo_id")
        if self.git is not None and rid and branch:
            result = self.git.merge_branch(
                rid, branch, message=f"task: {doc['title']} ({tid})")
            doc = self.get_task(tid)
            doc["merge_commit"] = result["commit"]
            self.store.put("spec_tasks", tid, doc, owner=doc["owner"],
                           parent=doc["project_id"])
        return self.transition(tid, "merged")
