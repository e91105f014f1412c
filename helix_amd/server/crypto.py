"""At-rest secret encryption with stdlib primitives (no `cryptography`
wheel in the image): HMAC-SHA256 keystream in counter mode +
encrypt-then-MAC integrity tag. The reference encrypts secrets server-
side; key comes from HELIX_SECRETS_KEY (falls back to a key derived
from the admin API key so single-node deploys work out of the box —
rotate by setting the env var).
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import logging
import os
import secrets as _secrets


def _derive(key: str, purpose: bytes) -> bytes:
    return hashlib.sha256(purpose + b":" + key.encode()).digest()


def _keystream(key: bytes, nonce: bytes, n: int) -> bytes:
    out = b""
    counter = 0
    while len(out) < n:
        out += hmac.new(key, nonce + counter.to_bytes(8, "big"),
                        hashlib.sha256).digest()
        counter += 1
    return out[:n]


def encrypt_str(plaintext: str, key: str) -> str:
    """-> "enc1:<b64(nonce || ct || tag)>" (versioned format)."""
    ek = _derive(key, b"enc")
    mk = _derive(key, b"mac")
    nonce = _secrets.token_bytes(16)
    pt = plaintext.encode()
    ct = bytes(a ^ b for a, b in zip(pt, _keystream(ek, nonce, len(pt))))
    tag = hmac.new(mk, nonce + ct, hashlib.sha256).digest()[:16]
    return "enc1:" + base64.b64encode(nonce + ct + tag).decode()


def decrypt_str(blob: str, key: str) -> str:
    """Inverse of encrypt_str; raises ValueError on tamper/bad key.
    Plaintext values (pre-encryption rows) pass through unchanged."""
    if not blob.startswith("enc1:"):
        # Legacy plaintext row (pre-encryption). Accepting it silently
        # would let a store-level writer strip encryption (downgrade), so
        # log loudly; HELIX_STRICT_SECRETS=1 rejects outright (set it once
        # all rows are migrated via re-encryption on next write).
        if os.environ.get("HELIX_STRICT_SECRETS") == "1":
            raise ValueError("plaintext secret row rejected "
                             "(HELIX_STRICT_SECRETS=1)")
        logging.getLogger("helix_amd.crypto").warning(
            "decrypt_str: legacy plaintext secret row encountered; "
            "it will be re-encrypted on next write")
        return blob
    raw = base64.b64decode(blob[5:])
    nonce, ct, tag = raw[:16], raw[16:-16], raw[-16:]
    mk = _derive(key, b"mac")
    want = hmac.new(mk, nonce + ct, hashlib.sha256).digest()[:16]
    if not hmac.compare_digest(want, tag):
        raise ValueError("secret integrity check failed (wrong key?)")
    ek = _derive(key, b"enc")
    pt = bytes(a ^ b for a, b in zip(ct, _keystream(ek, nonce, len(ct))))
    return pt.decode()


def secrets_key(admin_api_key: str) -> str:
    return os.environ.get("HELIX_SECRETS_KEY") or \
        hashlib.sha256(f"secrets:{admin_api_key}".encode()).hexdigest()
