"""Authentication: API keys, users, runner token (parity-lite with
api/pkg/auth HelixAuthenticator + server/auth_middleware.go)."""
from __future__ import annotations

import hashlib
import secrets
from dataclasses import dataclass
from typing import Optional

from fastapi import Depends, HTTPException, Request


@dataclass
class AuthUser:
    id: str
    username: str = ""
    admin: bool = False


class Authenticator:
    def __init__(self, store, admin_api_key: str, runner_token: str):
        self.store = store
        self.admin_api_key = admin_api_key
        self.runner_token = runner_token

    # -- users / keys ------------------------------------------------------
    def create_user(self, username: str, admin: bool = False) -> dict:
        uid = f"user_{hashlib.sha1(username.encode()).hexdigest()[:16]}"
        doc = {"id": uid, "username": username, "admin": admin}
        self.store.put("users", uid, doc, owner=uid)
        return doc

    def create_api_key(self, owner: str, name: str = "default") -> str:
        key = f"hl-{secrets.token_hex(24)}"
        self.store.put("api_keys", key, {"id": key, "owner": owner,
                                         "name": name}, owner=owner)
        return key

    def resolve(self, token: str) -> Optional[AuthUser]:
        if not token:
            return None
        if token == self.admin_api_key:
            return AuthUser(id="admin", username="admin", admin=True)
        doc = self.store.get("api_keys", token)
        if doc is None:
            return None
        user = self.store.get("users", doc["owner"])
        return AuthUser(id=doc["owner"],
                        username=(user or {}).get("username", ""),
                        admin=(user or {}).get("admin", False))

    def is_runner(self, token: str) -> bool:
        return bool(token) and token == self.runner_token


def bearer_token(request: Request) -> str:
    h = request.headers.get("Authorization", "")
    if h.lower().startswith("bearer "):
        return h[7:]
    return request.headers.get("X-API-Key", "")


def make_auth_dep(app_state_attr: str = "auth"):
    def dep(request: Request) -> AuthUser:
        auth: Authenticator = getattr(request.app.state, app_state_attr)
        user = auth.resolve(bearer_token(request))
        if user is None:
            raise HTTPException(401, "invalid or missing API key")
        return user
    return dep


def make_admin_dep():
    base = make_auth_dep()

    def dep(user: AuthUser = Depends(base)) -> AuthUser:
        if not user.admin:
            raise HTTPException(403, "admin only")
        return user
    return dep


def make_runner_dep():
    def dep(request: Request) -> bool:
        auth: Authenticator = request.app.state.auth
        if not auth.is_runner(bearer_token(request)):
            raise HTTPException(401, "invalid runner token")
        return True
    return dep
