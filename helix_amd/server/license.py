"""License manager (parity with the reference api/pkg/license:
signature-verified license envelope with expiry/features/limits,
hashed-ID revocation denylist, hourly revalidation — license.go:40-120,
manager.go:21-60). The reference signs with ECDSA; this stack signs
with Ed25519 (helix_amd.server.ed25519), same envelope shape:

    {"license": "<base64 JSON>", "signature": "<base64 sig>"}

No license => development mode (reference behavior), with the user
count still reported so operators can see where they stand.
"""
from __future__ import annotations

import base64
import hashlib
import json
import time
from typing import Optional

from helix_amd.server import ed25519


class LicenseError(Exception):
    pass


# SHA-256 hashes of revoked license IDs (license.go:60: the offline
# validator's only revocation channel; hashed so the source does not
# enumerate them).
REVOKED_ID_HASHES = {
    # none issued for this stack yet
}


class License:
    def __init__(self, doc: dict):
        self.id = doc.get("id", "")
        self.organization = doc.get("organization", "")
        self.valid = bool(doc.get("valid", False))
        self.issued = float(doc.get("issued", 0))
        self.valid_until = float(doc.get("valid_until", 0))
        self.features = dict(doc.get("features", {}))
        self.limits = dict(doc.get("limits", {}))

    def expired(self, now: Optional[float] = None) -> bool:
        return (now if now is not None else time.time()) > \
            self.valid_until

    def to_dict(self) -> dict:
        return {"id": self.id, "organization": self.organization,
                "valid": self.valid, "issued": self.issued,
                "valid_until": self.valid_until,
                "features": self.features, "limits": self.limits}


def sign_license(doc: dict, secret_key: bytes) -> str:
    """Issue a license envelope (operator tooling / tests)."""
    blob = json.dumps(doc, sort_keys=True).encode()
    sig = ed25519.sign(blob, secret_key)
    return json.dumps({
        "license": base64.b64encode(blob).decode(),
        "signature": base64.b64encode(sig).decode()})


def validate_license(envelope_str: str, public_key: bytes,
                     now: Optional[float] = None) -> License:
    """Envelope -> verified License; raises LicenseError on any
    failure (bad json/signature, revoked, invalid, expired)."""
    try:
        env = json.loads(envelope_str)
        blob = base64.b64decode(env["license"])
        sig = base64.b64decode(env["signature"])
    except Exception:
        raise LicenseError("malformed license envelope")
    if not ed25519.verify(blob, sig, public_key):
        raise LicenseError("license signature verification failed")
    try:
        lic = License(json.loads(blob))
    except Exception:
        raise LicenseError("malformed license payload")
    id_hash = hashlib.sha256(lic.id.encode()).hexdigest()
    if id_hash in REVOKED_ID_HASHES:
        raise LicenseError(
            f"license revoked ({REVOKED_ID_HASHES[id_hash]})")
    if not lic.valid:
        raise LicenseError("license is not valid")
    if lic.expired(now):
        raise LicenseError("license has expired")
    return lic


class LicenseManager:
    """Holds the active license; periodically revalidated by the
    server's janitor loop (manager.go:21 hourly ticker)."""

    def __init__(self, store, public_key: bytes):
        self.store = store
        self.public_key = public_key
        self._license: Optional[License] = None
        self._error: str = ""
        self._load()

    def _load(self):
        row = self.store.get("system_settings", "license")
        if row and row.get("envelope"):
            try:
                self._license = validate_license(row["envelope"],
                                                 self.public_key)
                self._error = ""
            except LicenseError as e:
                self._license = None
                self._error = str(e)

    def install(self, envelope_str: str) -> License:
        lic = validate_license(envelope_str, self.public_key)
        self.store.put("system_settings", "license",
                       {"id": "license", "envelope": envelope_str,
                        "installed": time.time()})
        self._license = lic
        self._error = ""
        return lic

    def revalidate(self) -> None:
        """Janitor hook: expiry can trip between installs."""
        self._load()

    def status(self) -> dict:
        users = self.store.count("users")
        if self._license is None:
            return {"mode": "development", "license": None,
                    "error": self._error or None, "users": users}
        lic = self._license
        seats = int(lic.limits.get("users", 0))
        return {"mode": "licensed", "license": lic.to_dict(),
                "users": users,
                "seats_exceeded": bool(seats and users > seats),
                "expires_in_days":
                    max(0, int((lic.valid_until - time.time()) / 86400))}

    def check_seat(self) -> None:
        """Called before creating a user: enforce the seat limit
        (development mode is unrestricted, reference behavior)."""
        if self._license is None:
            return
        seats = int(self._license.limits.get("users", 0))
        if seats and self.store.count("users") >= seats:
            raise LicenseError(
                f"license seat limit reached ({seats} users)")
