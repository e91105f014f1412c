"""HelixClient — typed API client library (parity with the reference's
api/pkg/client HelixClient: apps, sessions, knowledge, secrets, orgs,
teams, projects, git, filestore, models, sandboxes, system settings —
client.go:190 makeRequest + the per-resource files). The CLI builds on
this, and it is the programmatic entry point for integrations.

Sync httpx under the hood; every method returns parsed JSON (dict/list)
and raises HelixAPIError with the server's status + message on non-2xx.
"""
from __future__ import annotations

import json
import os
from typing import Any, Dict, Iterator, List, Optional


class HelixAPIError(Exception):
    def __init__(self, status: int, message: str):
        super().__init__(f"HTTP {status}: {message}")
        self.status = status
        self.message = message


class HelixClient:
    def __init__(self, url: str = "", api_key: str = "",
                 timeout: float = 60.0, http_client=None):
        import httpx
        self.url = (url or os.environ.get("HELIX_URL",
                                          "http://localhost:8080")
                    ).rstrip("/")
        self.api_key = api_key or os.environ.get("HELIX_API_KEY", "")
        self._http = http_client or httpx.Client(timeout=timeout)

    # -- plumbing (client.go:190 makeRequest) ------------------------------
    def _headers(self) -> dict:
        h = {"Content-Type": "application/json"}
        if self.api_key:
            h["Authorization"] = f"Bearer {self.api_key}"
        return h

    def request(self, method: str, path: str,
                body: Any = None, params: Optional[dict] = None) -> Any:
        r = self._http.request(
            method, self.url + path, headers=self._headers(),
            json=body if body is not None else None, params=params)
        if r.status_code >= 400:
            try:
                msg = r.json().get("detail", r.text)
            except Exception:
                msg = r.text
            raise HelixAPIError(r.status_code, str(msg)[:500])
        if not r.content:
            return None
        try:
            return r.json()
        except json.JSONDecodeError:
            return r.text

    def _get(self, path, **kw):
        return self.request("GET", path, **kw)

    def _post(self, path, body=None, **kw):
        return self.request("POST", path, body=body, **kw)

    def _put(self, path, body=None, **kw):
        return self.request("PUT", path, body=body, **kw)

    def _delete(self, path, **kw):
        return self.request("DELETE", path, **kw)

    # -- apps (client/app.go) ----------------------------------------------
    def list_apps(self) -> List[dict]:
        return self._get("/api/v1/apps")

    def get_app(self, app_id: str) -> dict:
        return self._get(f"/api/v1/apps/{app_id}")

    def create_app(self, config: dict, global_: bool = False) -> dict:
        return self._post("/api/v1/apps",
                          {"config": config, "global": global_})

    def update_app(self, app_id: str, config: dict) -> dict:
        return self._put(f"/api/v1/apps/{app_id}", {"config": config})

    def delete_app(self, app_id: str) -> None:
        self._delete(f"/api/v1/apps/{app_id}")

    # -- chat / sessions (client/session.go) ---------------------------------
    def chat(self, messages: List[dict], model: str = "",
             app_id: str = "", stream: bool = False, **kw) -> dict:
        body = {"messages": messages, "stream": False, **kw}
        if model:
            body["model"] = model
        params = {"app_id": app_id} if app_id else None
        return self._post("/v1/chat/completions", body, params=params)

    def chat_stream(self, messages: List[dict], model: str = "",
                    app_id: str = "", **kw) -> Iterator[dict]:
        body = {"messages": messages, "stream": True, **kw}
        if model:
            body["model"] = model
        params = {"app_id": app_id} if app_id else {}
        with self._http.stream(
                "POST", self.url + "/v1/chat/completions",
                headers=self._headers(), json=body,
                params=params) as r:
            if r.status_code >= 400:
                raise HelixAPIError(r.status_code, r.read().decode())
            for line in r.iter_lines():
                if line.startswith("data: "):
                    data = line[6:]
                    if data.strip() == "[DONE]":
                        return
                    yield json.loads(data)

    def images_generate(self, prompt: str, model: str = "",
                        n: int = 1, size: str = "", steps: int = 0,
                        seed: Optional[int] = None) -> dict:
        """POST /v1/images/generations — returns b64_json PNGs."""
        body: dict = {"prompt": prompt, "n": n}
        if model:
            body["model"] = model
        if size:
            body["size"] = size
        if steps:
            body["steps"] = steps
        if seed is not None:
            body["seed"] = seed
        return self._post("/v1/images/generations", body)

    def list_sessions(self) -> List[dict]:
        return self._get("/api/v1/sessions")

    def get_session(self, sid: str) -> dict:
        return self._get(f"/api/v1/sessions/{sid}")

    def delete_session(self, sid: str) -> None:
        self._delete(f"/api/v1/sessions/{sid}")

    # -- knowledge (client/knowledge.go) -------------------------------------
    def list_knowledge(self) -> List[dict]:
        return self._get("/api/v1/knowledge")

    def create_knowledge(self, name: str, source: dict,
                         refresh_schedule: str = "") -> dict:
        return self._post("/api/v1/knowledge",
                          {"name": name, "source": source,
                           "refresh_schedule": refresh_schedule})

    def delete_knowledge(self, kid: str) -> None:
        self._delete(f"/api/v1/knowledge/{kid}")

    def refresh_knowledge(self, kid: str) -> dict:
        return self._post(f"/api/v1/knowledge/{kid}/refresh")

    # -- secrets (client/secret.go) ------------------------------------------
    def list_secrets(self) -> List[dict]:
        return self._get("/api/v1/secrets")

    def set_secret(self, name: str, value: str) -> dict:
        return self._post("/api/v1/secrets",
                          {"name": name, "value": value})

    def delete_secret(self, name: str) -> None:
        self._delete(f"/api/v1/secrets/{name}")

    # -- organizations / teams (client/organizations.go, team.go) ------------
    def list_organizations(self) -> List[dict]:
        return self._get("/api/v1/organizations")

    def create_organization(self, name: str) -> dict:
        return self._post("/api/v1/organizations", {"name": name})

    def add_org_member(self, org_id: str, user_id: str,
                       role: str = "member") -> dict:
        return self._post(f"/api/v1/organizations/{org_id}/members",
                          {"user_id": user_id, "role": role})

    def list_teams(self, org_id: str) -> List[dict]:
        return self._get(f"/api/v1/organizations/{org_id}/teams")

    def create_team(self, org_id: str, name: str) -> dict:
        return self._post(f"/api/v1/organizations/{org_id}/teams",
                          {"name": name})

    # -- projects / spec tasks (client/project.go) ----------------------------
    def list_projects(self) -> List[dict]:
        return self._get("/api/v1/projects")

    def create_project(self, name: str) -> dict:
        return self._post("/api/v1/projects", {"name": name})

    def list_tasks(self, project_id: str) -> List[dict]:
        return self._get(f"/api/v1/projects/{project_id}/tasks")

    def create_task(self, project_id: str, title: str,
                    description: str = "") -> dict:
        return self._post(f"/api/v1/projects/{project_id}/tasks",
                          {"title": title,
                           "description": description})

    def transition_task(self, task_id: str, state: str) -> dict:
        return self._post(f"/api/v1/spec-tasks/{task_id}/transition",
                          {"state": state})

    def plan_task(self, task_id: str) -> dict:
        return self._post(f"/api/v1/spec-tasks/{task_id}/plan")

    def implement_task(self, task_id: str) -> dict:
        return self._post(f"/api/v1/spec-tasks/{task_id}/implement")

    # -- git (client/git.go) ---------------------------------------------------
    def list_repos(self) -> List[dict]:
        return self._get("/api/v1/git/repos")

    def repo_log(self, repo_id: str) -> List[dict]:
        return self._get(f"/api/v1/git/repos/{repo_id}/log")

    def repo_files(self, repo_id: str) -> List[str]:
        return self._get(f"/api/v1/git/repos/{repo_id}/files")

    # -- filestore (client/fs.go) ----------------------------------------------
    def filestore_list(self, path: str = "") -> List[dict]:
        return self._get("/api/v1/filestore/list",
                         params={"path": path})

    def filestore_upload(self, path: str, content: bytes) -> dict:
        import httpx
        r = self._http.post(
            self.url + "/api/v1/filestore/upload",
            headers={"Authorization": self._headers().get(
                "Authorization", "")},
            params={"path": path},
            files={"file": (os.path.basename(path) or "file", content)})
        if r.status_code >= 400:
            raise HelixAPIError(r.status_code, r.text)
        return r.json()

    def filestore_delete(self, path: str) -> None:
        self._delete("/api/v1/filestore/delete",
                     params={"path": path})

    # -- models (client/helix_models.go) ----------------------------------------
    def list_models(self) -> Any:
        return self._get("/v1/models")

    # -- sandboxes (client/sandbox.go) --------------------------------------------
    def create_sandbox(self, name: str = "") -> dict:
        return self._post("/api/v1/sandboxes", {"name": name})

    def list_sandboxes(self) -> List[dict]:
        return self._get("/api/v1/sandboxes")

    def sandbox_exec(self, sid: str, command: str,
                     timeout_s: float = 60) -> dict:
        return self._post(f"/api/v1/sandboxes/{sid}/exec",
                          {"command": command, "timeout_s": timeout_s})

    def delete_sandbox(self, sid: str) -> None:
        self._delete(f"/api/v1/sandboxes/{sid}")

    # -- evaluations -----------------------------------------------------------
    def create_evaluation_suite(self, app_id: str, name: str,
                                cases: List[dict]) -> dict:
        return self._post(f"/api/v1/apps/{app_id}/evaluation-suites",
                          {"name": name, "cases": cases})

    def run_evaluation_suite(self, suite_id: str) -> dict:
        return self._post(f"/api/v1/evaluation-suites/{suite_id}/runs")

    def get_evaluation_run(self, run_id: str) -> dict:
        return self._get(f"/api/v1/evaluation-runs/{run_id}")

    # -- usage / billing ---------------------------------------------------------
    def usage(self) -> List[dict]:
        return self._get("/api/v1/usage")

    def wallet(self) -> dict:
        return self._get("/api/v1/wallet")

    def billing(self) -> dict:
        return self._get("/api/v1/billing")

    # -- system (client.go:248 system settings; /healthz) --------------------------
    def health(self) -> dict:
        return self._get("/healthz")

    def config(self) -> dict:
        return self._get("/api/v1/config")

    def close(self):
        self._http.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
