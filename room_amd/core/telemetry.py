"""Telemetry (reference: src/shared/telemetry.ts — anonymous sha256 machine id,
crash reports and daily heartbeats; token-gated, fail-silent).

Offline environments record locally under the data dir; when an endpoint is
configured (ROOMAMD_TELEMETRY_URL) delivery is attempted fail-silently.
"""
from __future__ import annotations

import getpass
import hashlib
import json
import os
import socket
import time
from pathlib import Path


def get_machine_id() -> str:
    material = f"{socket.gethostname()}:{getpass.getuser()}"
    return hashlib.sha256(material.encode()).hexdigest()[:12]


def _telemetry_dir() -> Path:
    d = Path(os.environ.get("ROOMAMD_DATA_DIR",
                            str(Path.home() / ".roomamd"))) / "telemetry"
    d.mkdir(parents=True, exist_ok=True)
    return d


def _record(kind: str, payload: dict) -> None:
    payload = {"machine_id": get_machine_id(), "ts": int(time.time()),
               "kind": kind, **payload}
    try:
        path = _telemetry_dir() / f"{kind}.jsonl"
        with path.open("a") as f:
            f.write(json.dumps(payload) + "\n")
    except OSError:
        pass
    url = os.environ.get("ROOMAMD_TELEMETRY_URL")
    if url:
        try:
            import urllib.request
            req = urllib.request.Request(
                url, data=json.dumps(payload).encode(),
                headers={"Content-Type": "application/json"})
            urllib.request.urlopen(req, timeout=3)
        except Exception:
            pass  # fail-silent by design


def submit_crash_report(error: str, context: str = "") -> None:
    _record("crash", {"error": error[:2000], "context": context[:500]})


def submit_heartbeat(stats: dict | None = None) -> None:
    _record("heartbeat", {"stats": stats or {}})
