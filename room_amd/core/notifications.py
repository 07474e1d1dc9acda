"""Keeper notifications (reference: src/server/keeper-email.ts +
clerk-notifications.ts — email/telegram relay to the keeper).

Offline-first: every notification lands in a durable local outbox
(~/.roomamd/outbox.jsonl) and on the event bus; SMTP delivery is attempted
only when ROOMAMD_SMTP_URL is configured (fail-silent, like the reference's
relay)."""
from __future__ import annotations

import json
import os
import time
from pathlib import Path


def _outbox_path() -> Path:
    d = Path(os.environ.get("ROOMAMD_DATA_DIR", str(Path.home() / ".roomamd")))
    d.mkdir(parents=True, exist_ok=True)
    return d / "outbox.jsonl"


def notify_keeper(subject: str, body: str, room_id: int | None = None,
                  channel: str = "outbox", bus=None) -> dict:
    entry = {"ts": int(time.time()), "subject": subject, "body": body[:4000],
             "room_id": room_id, "channel": channel}
    try:
        with _outbox_path().open("a") as f:
            f.write(json.dumps(entry) + "\n")
    except OSError:
        pass
    if bus is not None:
        bus.emit("clerk", "keeper_notification", entry)
    tg_token = os.environ.get("ROOMAMD_TELEGRAM_BOT_TOKEN")
    tg_chat = os.environ.get("ROOMAMD_TELEGRAM_CHAT_ID")
    if tg_token and tg_chat:
        try:  # telegram relay (clerk-notifications.ts); fail-silent offline
            import urllib.parse
            import urllib.request
            data = urllib.parse.urlencode(
                {"chat_id": tg_chat,
                 "text": f"{subject}\n{body[:1000]}"}).encode()
            urllib.request.urlopen(
                f"https://api.telegram.org/bot{tg_token}/sendMessage",
                data=data, timeout=5)
            entry["telegram_delivered"] = True
        except Exception:
            entry["telegram_delivered"] = False
    smtp = os.environ.get("ROOMAMD_SMTP_URL")
    if smtp:
        try:  # fire-and-forget relay; never raises
            import smtplib
            from email.message import EmailMessage
            from urllib.parse import urlparse
            u = urlparse(smtp)
            msg = EmailMessage()
            msg["Subject"] = subject
            msg["From"] = u.username or "room-amd@localhost"
            msg["To"] = os.environ.get("ROOMAMD_KEEPER_EMAIL", "keeper@localhost")
            msg.set_content(body)
            with smtplib.SMTP(u.hostname, u.port or 25, timeout=5) as smtp_c:
                smtp_c.send_message(msg)
            entry["delivered"] = True
        except Exception:
            entry["delivered"] = False
    return entry


def read_outbox(limit: int = 50) -> list[dict]:
    p = _outbox_path()
    if not p.exists():
        return []
    lines = p.read_text().strip().splitlines()[-limit:]
    out = []
    for line in lines:
        try:
            out.append(json.loads(line))
        except ValueError:
            continue
    return out
