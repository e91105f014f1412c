"""Notification senders (parity with api/pkg/notification: email +
Slack/Discord webhooks). Offline-safe: failures log, never raise."""
from __future__ import annotations

import logging
import os
import smtplib
from email.mime.text import MIMEText
from typing import List, Optional

log = logging.getLogger("helix_amd.notifications")


class EmailSender:
    def __init__(self, host: str = "", port: int = 587, username: str = "",
                 password: str = "", from_addr: str = "helix@localhost"):
        self.host = host or os.environ.get("SMTP_HOST", "")
        self.port = int(os.environ.get("SMTP_PORT", port))
        self.username = username or os.environ.get("SMTP_USERNAME", "")
        self.password = password or os.environ.get("SMTP_PASSWORD", "")
        self.from_addr = from_addr

    def send(self, to: str, subject: str, body: str) -> bool:
        if not self.host:
            log.info("email (no SMTP configured) to=%s subject=%s", to,
                     subject)
            return False
        try:
            msg = MIMEText(body)
            msg["Subject"] = subject
            msg["From"] = self.from_addr
            msg["To"] = to
            with smtplib.SMTP(self.host, self.port, timeout=15) as s:
                s.starttls()
                if self.username:
                    s.login(self.username, self.password)
                s.sendmail(self.from_addr, [to], msg.as_string())
            return True
        except Exception as e:
            log.warning("email send failed: %s", e)
            return False


class WebhookSender:
    """Slack/Discord-style JSON webhook."""

    def __init__(self, url: str = ""):
        self.url = url or os.environ.get("SLACK_WEBHOOK_URL", "")

    def send(self, text: str) -> bool:
        if not self.url:
            log.info("webhook (not configured): %s", text[:120])
            return False
        try:
            import httpx
            r = httpx.post(self.url, json={"text": text}, timeout=15)
            return r.status_code < 300
        except Exception as e:
            log.warning("webhook send failed: %s", e)
            return False


class NotificationService:
    def __init__(self, email: Optional[EmailSender] = None,
                 webhook: Optional[WebhookSender] = None, store=None):
        self.email = email or EmailSender()
        self.webhook = webhook or WebhookSender()
        self.store = store
        self.sent: List[dict] = []   # in-memory log (tests/introspection)

    def notify(self, owner: str, subject: str, body: str,
               email_to: str = "") -> dict:
        res = {"owner": owner, "subject": subject,
               "email": False, "webhook": False}
        if email_to:
            res["email"] = self.email.send(email_to, subject, body)
        res["webhook"] = self.webhook.send(f"*{subject}*\n{body}")
        self.sent.append(res)
        return res
