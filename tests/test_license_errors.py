"""License manager (reference api/pkg/license: signed envelope,
expiry, revocation denylist, seat limits, development mode) and the
error-event reporting surface (Sentry role) + janitor sandbox GC.
"""
import hashlib
import json
import os
import time

import pytest

from helix_amd.server import ed25519
from helix_amd.server.license import (License, LicenseError,
                                      LicenseManager, sign_license,
                                      validate_license)
from helix_amd.store import Store

SK = bytes(range(1, 33))
PK = ed25519.public_from_secret(SK)


def _lic(**over):
    doc = {"id": "lic-1", "organization": "acme", "valid": True,
           "issued": time.time(), "valid_until": time.time() + 86400,
           "features": {"agents": True}, "limits": {"users": 3}}
    doc.update(over)
    return doc


def test_validate_roundtrip_and_failures():
    env = sign_license(_lic(), SK)
    lic = validate_license(env, PK)
    assert lic.organization == "acme" and lic.limits["users"] == 3
    # wrong key
    with pytest.raises(LicenseError, match="signature"):
        validate_license(env, ed25519.public_from_secret(b"x" * 32))
    # tampered payload
    bad = json.loads(env)
    import base64
    blob = json.loads(base64.b64decode(bad["license"]))
    blob["limits"]["users"] = 100000
    bad["license"] = base64.b64encode(
        json.dumps(blob, sort_keys=True).encode()).decode()
    with pytest.raises(LicenseError, match="signature"):
        validate_license(json.dumps(bad), PK)
    # expired
    with pytest.raises(LicenseError, match="expired"):
        validate_license(sign_license(
            _lic(valid_until=time.time() - 10), SK), PK)
    # invalid flag
    with pytest.raises(LicenseError, match="not valid"):
        validate_license(sign_license(_lic(valid=False), SK), PK)
    with pytest.raises(LicenseError, match="malformed"):
        validate_license("not json", PK)


def test_revocation_denylist(monkeypatch):
    import helix_amd.server.license as lm
    h = hashlib.sha256(b"lic-revoked").hexdigest()
    monkeypatch.setitem(lm.REVOKED_ID_HASHES, h, "R9")
    env = sign_license(_lic(id="lic-revoked"), SK)
    with pytest.raises(LicenseError, match="revoked"):
        validate_license(env, PK)


def test_manager_seats_and_dev_mode():
    store = Store(":memory:")
    mgr = LicenseManager(store, PK)
    assert mgr.status()["mode"] == "development"
    mgr.check_seat()                      # dev mode unrestricted
    mgr.install(sign_license(_lic(), SK))
    st = mgr.status()
    assert st["mode"] == "licensed" and not st["seats_exceeded"]
    for i in range(4):
        store.put("users", f"u{i}", {"id": f"u{i}"})
    with pytest.raises(LicenseError, match="seat"):
        mgr.check_seat()
    assert mgr.status()["seats_exceeded"]
    # persisted: a fresh manager reloads it
    mgr2 = LicenseManager(store, PK)
    assert mgr2.status()["mode"] == "licensed"


def test_http_license_errors_and_sandbox_gc(tmp_path, monkeypatch):
    from fastapi.testclient import TestClient
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    monkeypatch.setenv("HELIX_LICENSE_PUBKEY", PK.hex())
    cfg = load_config()
    cfg.store.path = str(tmp_path / "db.sqlite")
    cfg.filestore.path = str(tmp_path / "fs")
    app = create_app(cfg)
    with TestClient(app, raise_server_exceptions=False) as client:
        auth = app.state.auth
        admin = auth.create_user("root", admin=True)
        key = auth.create_api_key(admin["id"])
        H = {"Authorization": f"Bearer {key}"}
        r = client.get("/api/v1/license", headers=H)
        assert r.json()["mode"] == "development"
        r = client.post("/api/v1/license", headers=H, json={
            "envelope": sign_license(_lic(limits={"users": 2}), SK)})
        assert r.status_code == 200, r.text
        # seat limit enforced on user creation (1 user exists)
        r = client.post("/api/v1/users", headers=H,
                        json={"username": "second"})
        assert r.status_code == 200, r.text
        r = client.post("/api/v1/users", headers=H,
                        json={"username": "third"})
        assert r.status_code == 402
        # bad envelope rejected
        r = client.post("/api/v1/license", headers=H,
                        json={"envelope": "junk"})
        assert r.status_code == 400
        # error events: hit a route that raises; sweep the table
        app.state.record_error("/test/path", ValueError("boom"))
        app.state.record_error("/test/path", ValueError("boom again"))
        r = client.get("/api/v1/admin/errors", headers=H)
        rows = r.json()
        assert rows and rows[0]["count"] == 2
        assert rows[0]["type"] == "ValueError"
        # janitor GC: idle sandbox is reaped
        sbx = app.state.sandboxes.create(admin["id"], "idle")
        doc = app.state.store.get("sandboxes", sbx["id"])
        doc["created"] = time.time() - 100 * 3600
        app.state.store.put("sandboxes", sbx["id"], doc,
                            owner=admin["id"])
        r = client.post("/api/v1/admin/janitor", headers=H,
                        json={"sandbox_idle_hours": 24})
        assert r.json()["sandboxes"] == 1
        assert app.state.sandboxes.get(sbx["id"]) is None
