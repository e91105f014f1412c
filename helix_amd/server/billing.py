"""Stripe billing (parity with the reference api/pkg/stripe: webhook
signature verification + event routing, stripe.go:137-190; top-up
checkout flow crediting wallets, stripe_topups.go; subscription state
sync onto wallets, stripe_subscriptions.go).

Offline-first: webhook verification is plain HMAC-SHA256 over Stripe's
`t=<ts>,v1=<sig>` header scheme (no SDK needed), and outbound Stripe API
calls go through an injected client — the bundled FakeStripeAPI supports
air-gapped deployments and tests; a live deployment injects a thin
httpx-based client with the same three methods.
"""
from __future__ import annotations

import hashlib
import hmac
import json
import logging
import time
from typing import Dict, List, Optional

log = logging.getLogger("helix_amd.billing")


class WebhookError(Exception):
    pass


def verify_stripe_signature(payload: bytes, sig_header: str,
                            secret: str, tolerance_s: int = 300,
                            now: Optional[float] = None) -> None:
    """Stripe webhook scheme: header `t=<unix>,v1=<hex hmac>`; signed
    payload is `<t>.<body>` with HMAC-SHA256(secret). Raises
    WebhookError on any failure (reference uses the SDK's
    webhook.ConstructEvent, stripe.go:149)."""
    parts = dict(
        p.split("=", 1) for p in sig_header.split(",") if "=" in p)
    ts = parts.get("t", "")
    v1 = parts.get("v1", "")
    if not ts or not v1:
        raise WebhookError("malformed Stripe-Signature header")
    try:
        tsf = float(ts)
    except ValueError:
        raise WebhookError("bad timestamp")
    if abs((now if now is not None else time.time()) - tsf) > tolerance_s:
        raise WebhookError("timestamp outside tolerance")
    expected = hmac.new(secret.encode(),
                        f"{ts}.".encode() + payload,
                        hashlib.sha256).hexdigest()
    if not hmac.compare_digest(expected, v1):
        raise WebhookError("signature mismatch")


def sign_stripe_payload(payload: bytes, secret: str,
                        now: Optional[float] = None) -> str:
    """Produce a valid Stripe-Signature header (used by tests and the
    fake to exercise the real verification path)."""
    ts = str(int(now if now is not None else time.time()))
    sig = hmac.new(secret.encode(), f"{ts}.".encode() + payload,
                   hashlib.sha256).hexdigest()
    return f"t={ts},v1={sig}"


class FakeStripeAPI:
    """Offline stand-in for the Stripe REST API: customers, checkout
    sessions, subscriptions. Live deployments replace it with an httpx
    client exposing the same methods."""

    def __init__(self):
        self.customers: Dict[str, dict] = {}
        self.sessions: Dict[str, dict] = {}
        self.subscriptions: Dict[str, dict] = {}
        self._n = 0

    def _id(self, prefix):
        self._n += 1
        return f"{prefix}_{self._n:06d}"

    async def create_customer(self, email: str, metadata: dict) -> dict:
        cid = self._id("cus")
        self.customers[cid] = {"id": cid, "email": email,
                               "metadata": metadata}
        return self.customers[cid]

    async def create_checkout_session(self, customer: str,
                                      amount_cents: int,
                                      metadata: dict) -> dict:
        sid = self._id("cs")
        self.sessions[sid] = {
            "id": sid, "customer": customer,
            "amount_total": amount_cents, "metadata": metadata,
            "url": f"https://checkout.stripe.test/{sid}"}
        return self.sessions[sid]

    async def list_subscriptions(self, customer: str) -> List[dict]:
        return [s for s in self.subscriptions.values()
                if s["customer"] == customer]


class BillingService:
    """Wallet top-ups + subscription state driven by Stripe events
    (reference: topups credit wallets in USD; subscriptions set the
    wallet's subscription_status/plan)."""

    def __init__(self, store, usage, api=None,
                 webhook_secret: str = "", currency: str = "usd"):
        self.store = store
        self.usage = usage               # UsageService (wallets live there)
        self.api = api or FakeStripeAPI()
        self.webhook_secret = webhook_secret
        self.currency = currency

    def enabled(self) -> bool:
        return bool(self.webhook_secret)

    # -- customers ---------------------------------------------------------
    async def ensure_customer(self, user: dict) -> str:
        uid = user["id"]
        doc = self.store.get("billing_customers", uid)
        if doc:
            return doc["customer_id"]
        cust = await self.api.create_customer(
            user.get("email", ""), {"user_id": uid})
        self.store.put("billing_customers", uid,
                       {"id": uid, "customer_id": cust["id"],
                        "created": time.time()}, owner=uid)
        return cust["id"]

    # -- top-ups -----------------------------------------------------------
    async def create_topup_session(self, user: dict,
                                   amount_usd: float) -> dict:
        """GetTopUpSessionURL equivalent (stripe_topups.go:34)."""
        if amount_usd <= 0 or amount_usd > 10000:
            raise ValueError("top-up amount out of range")
        customer = await self.ensure_customer(user)
        sess = await self.api.create_checkout_session(
            customer, int(round(amount_usd * 100)),
            {"user_id": user["id"],
             "amount_cents": str(int(round(amount_usd * 100)))})
        return {"url": sess["url"], "session_id": sess["id"]}

    # -- webhook -----------------------------------------------------------
    def process_webhook(self, payload: bytes, sig_header: str) -> dict:
        """Verify + route (stripe.go:137): checkout.session.completed /
        payment_intent.succeeded credit the wallet exactly once per
        event id; customer.subscription.* sync subscription state."""
        verify_stripe_signature(payload, sig_header, self.webhook_secret)
        try:
            event = json.loads(payload)
        except json.JSONDecodeError:
            raise WebhookError("bad JSON payload")
        eid = event.get("id", "")
        if eid and self.store.get("billing_events", eid):
            return {"ok": True, "deduped": True}    # at-least-once safe
        etype = event.get("type", "")
        obj = (event.get("data") or {}).get("object") or {}
        handled = True
        if etype in ("checkout.session.completed",
                     "payment_intent.succeeded"):
            self._handle_topup(obj)
        elif etype.startswith("customer.subscription."):
            self._handle_subscription(etype, obj)
        elif etype == "invoice.paid":
            self._handle_invoice(obj)
        else:
            handled = False
        if eid:
            self.store.put("billing_events", eid,
                           {"id": eid, "type": etype,
                            "handled": handled, "ts": time.time()})
        return {"ok": True, "handled": handled}

    def _user_for_customer(self, obj: dict) -> Optional[str]:
        md = obj.get("metadata") or {}
        if md.get("user_id"):
            return md["user_id"]
        cust = obj.get("customer", "")
        for row in self.store.list("billing_customers", limit=100000):
            if row.get("customer_id") == cust:
                return row["id"]
        return None

    def _handle_topup(self, obj: dict):
        uid = self._user_for_customer(obj)
        if uid is None:
            log.warning("top-up event with unknown customer: %s",
                        obj.get("customer"))
            return
        # one PAYMENT can arrive as both checkout.session.completed and
        # payment_intent.succeeded (distinct event ids): credit once
        # per payment, keyed by the payment_intent (or the session id)
        pay_id = obj.get("payment_intent") or obj.get("id", "")
        if pay_id:
            guard = f"pay:{pay_id}"
            if self.store.get("billing_events", guard):
                return
            self.store.put("billing_events", guard,
                           {"id": guard, "type": "payment-credit",
                            "ts": time.time()})
        md = obj.get("metadata") or {}
        cents = int(md.get("amount_cents") or
                    obj.get("amount_total") or
                    obj.get("amount_received") or 0)
        if cents <= 0:
            return
        self.usage.topup(uid, cents / 100.0,
                         ref=f"stripe:{pay_id}")

    def _handle_subscription(self, etype: str, obj: dict):
        uid = self._user_for_customer(obj)
        if uid is None:
            return
        status = "canceled" if etype.endswith(".deleted") else \
            obj.get("status", "active")
        plan = ""
        items = (obj.get("items") or {}).get("data") or []
        if items:
            plan = ((items[0].get("price") or {}).get("lookup_key") or
                    (items[0].get("price") or {}).get("id", ""))
        w = self.usage.wallet(uid)
        w["subscription_status"] = status
        w["subscription_plan"] = plan
        w["subscription_id"] = obj.get("id", "")
        self.store.put("wallets", uid, w, owner=uid)

    def _handle_invoice(self, obj: dict):
        uid = self._user_for_customer(obj)
        if uid is None:
            return
        iid = obj.get("id", "")
        self.store.put("billing_invoices", iid or f"in_{time.time()}",
                       {"id": iid, "owner": uid,
                        "amount_paid": obj.get("amount_paid", 0),
                        "ts": time.time()}, owner=uid)
