"""OIDC authentication (parity with the reference's OIDC mode,
api/pkg/auth/oidc.go: discovery, auth-URL with state+nonce, code
exchange, ID-token verification, userinfo, refresh, email-domain
allow-list, plus session cookies via api/pkg/auth/session_manager.go).

Offline-first implementation: RS256 signature verification is
pure-Python (PKCS#1 v1.5 over SHA-256 with JWKS n/e big-int math), so no
crypto package is required; deployments point it at any standard issuer
and tests run a self-contained FakeIssuer with a generated RSA key.
"""
from __future__ import annotations

import base64
import hashlib
import json
import secrets
import time
from typing import Dict, List, Optional

__all__ = ["OIDCClient", "OIDCError", "parse_email_domains",
           "email_domain_allowed", "rs256_verify", "b64url_decode"]


class OIDCError(Exception):
    pass


def b64url_decode(s: str) -> bytes:
    s = s.replace("-", "+").replace("_", "/")
    return base64.b64decode(s + "=" * (-len(s) % 4))


def b64url_encode(b: bytes) -> str:
    return base64.b64encode(b).decode().replace("+", "-").replace(
        "/", "_").rstrip("=")


# PKCS#1 v1.5 DigestInfo prefix for SHA-256 (RFC 8017 §9.2)
_SHA256_PREFIX = bytes.fromhex(
    "3031300d060960864801650304020105000420")


def rs256_verify(signing_input: bytes, signature: bytes, n: int,
                 e: int) -> bool:
    """RSASSA-PKCS1-v1_5 verification with SHA-256, by direct modexp —
    constant-time comparison of the recovered EM against the expected
    padding (verification is public-key math; no secrets involved)."""
    k = (n.bit_length() + 7) // 8
    if len(signature) != k:
        return False
    s = int.from_bytes(signature, "big")
    if s >= n:
        return False
    em = pow(s, e, n).to_bytes(k, "big")
    digest = hashlib.sha256(signing_input).digest()
    t = _SHA256_PREFIX + digest
    if k < len(t) + 11:
        return False
    expected = b"\x00\x01" + b"\xff" * (k - len(t) - 3) + b"\x00" + t
    return secrets.compare_digest(em, expected)


def parse_email_domains(s: str) -> List[str]:
    """Comma/space separated allow-list (reference oidc.go:472)."""
    out = []
    for part in s.replace(",", " ").split():
        part = part.strip().lstrip("@").lower()
        if part:
            out.append(part)
    return out


def email_domain_allowed(email: str, allowed: List[str]) -> bool:
    if not allowed:
        return True
    domain = email.rsplit("@", 1)[-1].lower() if "@" in email else ""
    return domain in allowed


class OIDCClient:
    """Code-flow client against a standard issuer. The http client is
    injected (httpx-compatible async interface) so air-gapped tests use
    a FakeIssuer; live deployments pass a real httpx.AsyncClient."""

    def __init__(self, issuer: str, client_id: str, client_secret: str,
                 redirect_url: str, http_client=None,
                 allowed_domains: str = "", scopes: Optional[List[str]] = None):
        self.issuer = issuer.rstrip("/")
        self.client_id = client_id
        self.client_secret = client_secret
        self.redirect_url = redirect_url
        self.scopes = scopes or ["openid", "profile", "email"]
        self.allowed_domains = parse_email_domains(allowed_domains)
        self._http = http_client
        self._discovery: Optional[dict] = None
        self._jwks: Optional[dict] = None
        self._jwks_fetched = 0.0

    def _client(self):
        if self._http is None:
            import httpx
            self._http = httpx.AsyncClient(timeout=15)
        return self._http

    async def discovery(self) -> dict:
        if self._discovery is None:
            r = await self._client().get(
                self.issuer + "/.well-known/openid-configuration")
            if r.status_code != 200:
                raise OIDCError(f"discovery failed: HTTP {r.status_code}")
            self._discovery = r.json()
        return self._discovery

    async def jwks(self, force: bool = False) -> dict:
        now = time.time()
        if self._jwks is None or force or now - self._jwks_fetched > 3600:
            disc = await self.discovery()
            r = await self._client().get(disc["jwks_uri"])
            if r.status_code != 200:
                raise OIDCError(f"jwks fetch failed: HTTP {r.status_code}")
            self._jwks = r.json()
            self._jwks_fetched = now
        return self._jwks

    async def get_auth_url(self, state: str, nonce: str) -> str:
        from urllib.parse import urlencode
        disc = await self.discovery()
        q = urlencode({
            "response_type": "code", "client_id": self.client_id,
            "redirect_uri": self.redirect_url,
            "scope": " ".join(self.scopes),
            "state": state, "nonce": nonce})
        return f"{disc['authorization_endpoint']}?{q}"

    async def exchange(self, code: str) -> dict:
        """Auth code -> token response (oidc.go:211)."""
        disc = await self.discovery()
        r = await self._client().post(disc["token_endpoint"], data={
            "grant_type": "authorization_code", "code": code,
            "redirect_uri": self.redirect_url,
            "client_id": self.client_id,
            "client_secret": self.client_secret})
        if r.status_code != 200:
            raise OIDCError(f"code exchange failed: HTTP {r.status_code}")
        return r.json()

    async def refresh(self, refresh_token: str) -> dict:
        disc = await self.discovery()
        r = await self._client().post(disc["token_endpoint"], data={
            "grant_type": "refresh_token",
            "refresh_token": refresh_token,
            "client_id": self.client_id,
            "client_secret": self.client_secret})
        if r.status_code != 200:
            raise OIDCError(f"refresh failed: HTTP {r.status_code}")
        return r.json()

    async def verify_id_token(self, id_token: str,
                              nonce: str = "") -> dict:
        """Signature (RS256 via JWKS kid), iss, aud, exp, optional
        nonce. Returns the claims (oidc.go:237)."""
        try:
            h_b64, p_b64, s_b64 = id_token.split(".")
            header = json.loads(b64url_decode(h_b64))
            claims = json.loads(b64url_decode(p_b64))
            sig = b64url_decode(s_b64)
        except (ValueError, json.JSONDecodeError):
            raise OIDCError("malformed id_token")
        if header.get("alg") != "RS256":
            raise OIDCError(f"unsupported alg {header.get('alg')!r}")
        signing_input = f"{h_b64}.{p_b64}".encode()
        keys = (await self.jwks()).get("keys", [])
        kid = header.get("kid")
        candidates = [k for k in keys if not kid or k.get("kid") == kid]
        if kid and not candidates:
            # key rotation: refetch once (go-oidc keyset behavior)
            keys = (await self.jwks(force=True)).get("keys", [])
            candidates = [k for k in keys if k.get("kid") == kid]
        ok = False
        for k in candidates:
            if k.get("kty") != "RSA":
                continue
            n = int.from_bytes(b64url_decode(k["n"]), "big")
            e = int.from_bytes(b64url_decode(k["e"]), "big")
            if rs256_verify(signing_input, sig, n, e):
                ok = True
                break
        if not ok:
            raise OIDCError("id_token signature verification failed")
        if claims.get("iss", "").rstrip("/") != self.issuer:
            raise OIDCError("issuer mismatch")
        aud = claims.get("aud")
        auds = aud if isinstance(aud, list) else [aud]
        if self.client_id not in auds:
            raise OIDCError("audience mismatch")
        if claims.get("exp", 0) < time.time():
            raise OIDCError("id_token expired")
        if nonce and claims.get("nonce") != nonce:
            raise OIDCError("nonce mismatch")
        email = claims.get("email", "")
        if email and not email_domain_allowed(email, self.allowed_domains):
            raise OIDCError(f"email domain not allowed: {email}")
        return claims

    async def userinfo(self, access_token: str) -> dict:
        disc = await self.discovery()
        r = await self._client().get(disc["userinfo_endpoint"], headers={
            "Authorization": f"Bearer {access_token}"})
        if r.status_code != 200:
            raise OIDCError(f"userinfo failed: HTTP {r.status_code}")
        return r.json()
