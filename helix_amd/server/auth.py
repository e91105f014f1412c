"""Authentication: API keys, users, runner token, self-issued HS256 JWTs
(parity with api/pkg/auth HelixAuthenticator + server/auth_middleware.go;
the reference validates Keycloak JWTs — with no external IdP in this
deployment the server is its own issuer, same bearer flow)."""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import secrets
import time
from dataclasses import dataclass
from typing import Optional

from fastapi import Depends, HTTPException, Request


def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _b64url_dec(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def jwt_encode(payload: dict, secret: str) -> str:
    header = _b64url(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
    body = _b64url(json.dumps(payload, separators=(",", ":")).encode())
    signing = f"{header}.{body}".encode()
    sig = _b64url(hmac.new(secret.encode(), signing,
                           hashlib.sha256).digest())
    return f"{header}.{body}.{sig}"


def jwt_decode(token: str, secret: str) -> Optional[dict]:
    """Verify signature + expiry; returns the claims or None."""
    try:
        header, body, sig = token.split(".")
        signing = f"{header}.{body}".encode()
        want = _b64url(hmac.new(secret.encode(), signing,
                                hashlib.sha256).digest())
        if not hmac.compare_digest(want, sig):
            return None
        hdr = json.loads(_b64url_dec(header))
        if hdr.get("alg") != "HS256":
            return None
        claims = json.loads(_b64url_dec(body))
        if claims.get("exp") is not None and claims["exp"] < time.time():
            return None
        return claims
    except Exception:
        return None


@dataclass
class AuthUser:
    id: str
    username: str = ""
    admin: bool = False
    app_id: str = ""      # set for app-scoped API keys (client/app.go
                          # GetAppAPIKeys): requests default to this app


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

    def create_api_key(self, owner: str, name: str = "default",
                       app_id: str = "") -> str:
        key = f"hl-{secrets.token_hex(24)}"
        self.store.put("api_keys", key, {"id": key, "owner": owner,
                                         "name": name,
                                         "app_id": app_id}, owner=owner)
        return key

    # -- JWT (self-issued HS256) --------------------------------------
    @property
    def jwt_secret(self) -> str:
        # derived, not the admin key itself (key rotation leaves JWTs
        # independent of direct admin-key comparison)
        return hashlib.sha256(
            f"jwt:{self.admin_api_key}".encode()).hexdigest()

    def issue_jwt(self, user: "AuthUser", ttl_s: int = 3600) -> str:
        now = int(time.time())
        return jwt_encode({"sub": user.id, "preferred_username":
                           user.username, "admin": user.admin,
                           "iat": now, "exp": now + ttl_s,
                           "iss": "helix_amd"}, self.jwt_secret)

    def resolve(self, token: str) -> Optional[AuthUser]:
        if not token:
            return None
        if hmac.compare_digest(token, self.admin_api_key):
            return AuthUser(id="admin", username="admin", admin=True)
        if token.count(".") == 2:                      # JWT shape
            claims = jwt_decode(token, self.jwt_secret)
            if claims is None:
                return None
            return AuthUser(id=claims.get("sub", ""),
                            username=claims.get("preferred_username", ""),
                            admin=bool(claims.get("admin")))
        doc = self.store.get("api_keys", token)
        if doc is None:
            return None
        user = self.store.get("users", doc["owner"])
        return AuthUser(id=doc["owner"],
                        username=(user or {}).get("username", ""),
                        admin=(user or {}).get("admin", False),
                        app_id=doc.get("app_id", ""))

    def is_runner(self, token: str) -> bool:
        return bool(token) and hmac.compare_digest(token,
                                                   self.runner_token)


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
