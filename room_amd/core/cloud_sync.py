"""Cloud sync (reference: src/shared/cloud-sync.ts — optional registration with
a cloud relay, per-room token store file, 5-min heartbeats, inter-room message
relay; everything fail-silent).

Offline default: disabled unless ROOMAMD_CLOUD_API is set. Token store format
(~/.roomamd/cloud-room-tokens.json) preserved.
"""
from __future__ import annotations

import json
import os
import time
import urllib.request
from pathlib import Path

from ..db import LockedDb
from ..db import queries as q
from .telemetry import get_machine_id

HEARTBEAT_INTERVAL_S = 300


def cloud_api() -> str | None:
    return os.environ.get("ROOMAMD_CLOUD_API") or None


def _tokens_path() -> Path:
    return Path(os.environ.get("ROOMAMD_DATA_DIR",
                               str(Path.home() / ".roomamd"))) / "cloud-room-tokens.json"


def load_room_tokens() -> dict:
    """Token store file format preserved from the reference:
    {"rooms": {"<roomId>": "<token>"}} (cloud-sync.ts:20-52; the on-disk
    contract in SURVEY §2d). A legacy flat dict is still read."""
    p = _tokens_path()
    if p.exists():
        try:
            data = json.loads(p.read_text())
            if isinstance(data, dict) and isinstance(data.get("rooms"), dict):
                return data["rooms"]
            return data if isinstance(data, dict) else {}
        except (ValueError, OSError):
            return {}
    return {}


def save_room_token(room_id: int, token: str) -> None:
    tokens = load_room_tokens()
    tokens[str(room_id)] = token
    p = _tokens_path()
    try:
        p.parent.mkdir(parents=True, exist_ok=True)
        p.write_text(json.dumps({"rooms": tokens}))
        p.chmod(0o600)
    except OSError:
        pass


def _post(path: str, payload: dict, token: str | None = None) -> dict | None:
    api = cloud_api()
    if not api:
        return None
    try:
        headers = {"Content-Type": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        req = urllib.request.Request(f"{api}{path}",
                                     data=json.dumps(payload).encode(),
                                     headers=headers)
        with urllib.request.urlopen(req, timeout=5) as resp:
            return json.load(resp)
    except Exception:
        return None  # fail-silent by design


def register_with_cloud(ldb: LockedDb, room_id: int) -> str | None:
    with ldb as db:
        room = q.get_room(db, room_id)
    if room is None:
        return None
    # endpoint paths mirror cloud-sync.ts (/rooms/register,
    # /rooms/<id>/heartbeat, /rooms/message)
    out = _post("/rooms/register",
                {"machine_id": get_machine_id(), "room_id": room_id,
                 "name": room["name"]})
    if out and out.get("token"):
        save_room_token(room_id, out["token"])
        return out["token"]
    return None


def send_heartbeat(ldb: LockedDb, room_id: int) -> bool:
    token = load_room_tokens().get(str(room_id))
    if not token:
        return False
    with ldb as db:
        usage = q.get_room_token_usage(db, room_id)
        workers = q.list_room_workers(db, room_id)
    out = _post(f"/rooms/{room_id}/heartbeat",
                {"room_id": room_id, "ts": int(time.time()),
                 "cycles": usage["cycles"], "workers": len(workers)},
                token=token)
    return out is not None


def send_cloud_room_message(ldb: LockedDb, room_id: int, to_room: str,
                            subject: str, body: str) -> bool:
    token = load_room_tokens().get(str(room_id))
    out = _post("/rooms/message",
                {"to": to_room, "subject": subject, "body": body}, token=token)
    with ldb as db:  # outbound row recorded regardless (durable intent)
        q.create_room_message(db, room_id, "outbound", subject, body,
                              to_room_id=to_room)
    return out is not None


def fetch_cloud_room_messages(ldb: LockedDb, room_id: int) -> int:
    """Pull relayed inbound messages into room_messages (runtime 60s loop)."""
    api = cloud_api()
    token = load_room_tokens().get(str(room_id))
    if not api or not token:
        return 0
    try:
        req = urllib.request.Request(
            f"{api}/rooms/{room_id}/messages",
            headers={"Authorization": f"Bearer {token}"})
        with urllib.request.urlopen(req, timeout=5) as resp:
            msgs = json.load(resp)
    except Exception:
        return 0
    n = 0
    with ldb as db:
        for m in msgs if isinstance(msgs, list) else []:
            q.create_room_message(db, room_id, "inbound",
                                  m.get("subject", ""), m.get("body", ""),
                                  from_room_id=m.get("from"))
            n += 1
    return n


# --- activity push (reference: src/server/cloud.ts — event-bus subscriber
# maps internal event types to cloud activity kinds and pushes them to the
# relay, max 1 push/sec per room, fail-silent)

# internal bus event type → cloud activity kind (the reference's map,
# cloud.ts CLOUD_EVENT_MAP, keyed by ITS event vocabulary; keys here are
# this framework's bus types for the same moments)
CLOUD_EVENT_MAP = {
    "decision": "decision_created",
    "escalation": "escalation",
    "escalation:resolved": "escalation_resolved",
    "message": "room_message",
    "worker_created": "worker_created",
    "room_started": "room_started",
    "cycle_finished": "cycle",
}


class ActivityPusher:
    """Wildcard bus subscriber that relays mapped room events to the cloud.
    Inert without ROOMAMD_CLOUD_API; a `sender` hook makes it testable."""

    def __init__(self, bus, ldb: LockedDb, sender=None,
                 min_gap_s: float = 1.0):
        self.ldb = ldb
        self.min_gap_s = min_gap_s
        self._last_push: dict[int, float] = {}
        self._sender = sender
        self._unsub = bus.on("*", self._on_event)

    def stop(self) -> None:
        if self._unsub:
            self._unsub()
            self._unsub = None

    def _on_event(self, channel: str, event: dict) -> None:
        if self._sender is None and not cloud_api():
            return
        kind = CLOUD_EVENT_MAP.get(event.get("type") or "")
        if kind is None or not channel.startswith("room:"):
            return
        try:
            room_id = int(channel.split(":", 1)[1])
        except ValueError:
            return
        now = time.time()
        if now - self._last_push.get(room_id, 0.0) < self.min_gap_s:
            return
        self._last_push[room_id] = now
        token = load_room_tokens().get(str(room_id))
        payload = {"type": kind, "data": event.get("data"),
                   "timestamp": event.get("timestamp")}
        try:
            if self._sender is not None:
                self._sender(room_id, payload)
            elif token:
                _post(f"/rooms/{room_id}/activity", payload, token=token)
        except Exception:
            pass
