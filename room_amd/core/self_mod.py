"""Audited self-modification with rate limit, forbidden paths, and true revert
(reference: src/shared/self-mod.ts — 60s per-worker rate limit, forbidden
patterns :5-34, snapshot revert in a transaction :57-84)."""
from __future__ import annotations

import hashlib
import sqlite3
import time
from datetime import datetime

from ..db import queries as q
from .constants import SELF_MOD_RATE_LIMIT_MS

FORBIDDEN_PATTERNS = (
    ".env", "private_key", "id_rsa", "id_ed25519", ".ssh", ".aws",
    "self_mod.py", "secret", "credentials", "api.token", "auth.tokens",
)


def _hash(content: str | None) -> str | None:
    if content is None:
        return None
    return hashlib.sha256(content.encode()).hexdigest()[:16]


def can_modify(db: sqlite3.Connection, worker_id: int, target_path: str) -> tuple[bool, str]:
    lowered = target_path.lower()
    for pat in FORBIDDEN_PATTERNS:
        if pat in lowered:
            return False, f"Forbidden path pattern: {pat}"
    last = q.last_self_mod_time(db, worker_id)
    if last:
        try:
            last_dt = datetime.strptime(last, "%Y-%m-%d %H:%M:%S")
            elapsed_ms = (datetime.now() - last_dt).total_seconds() * 1000
            if elapsed_ms < SELF_MOD_RATE_LIMIT_MS:
                return False, (f"Rate limited: wait "
                               f"{int((SELF_MOD_RATE_LIMIT_MS - elapsed_ms) / 1000)}s")
        except ValueError:
            pass
    return True, "ok"


def perform_skill_modification(db: sqlite3.Connection, room_id: int | None,
                               worker_id: int, skill_id: int, new_content: str,
                               reason: str | None = None) -> dict:
    """Audited skill edit with snapshot for true revert."""
    target = f"skill:{skill_id}"
    ok, why = can_modify(db, worker_id, target)
    if not ok:
        raise PermissionError(why)
    skill = q.get_skill(db, skill_id)
    if skill is None:
        raise ValueError(f"Skill {skill_id} not found")
    old_content = skill["content"]
    audit_id = q.create_self_mod_audit(
        db, room_id, worker_id, target, _hash(old_content), _hash(new_content),
        reason, reversible=True)
    q.create_self_mod_snapshot(db, audit_id, "skill", skill_id, old_content, new_content)
    q.update_skill(db, skill_id, content=new_content)
    if room_id:
        q.log_room_activity(db, room_id, "self_mod",
                            f"Skill '{skill['name']}' modified", actor_id=worker_id)
    return {"audit_id": audit_id, "skill": q.get_skill(db, skill_id)}


def revert_modification(db: sqlite3.Connection, audit_id: int) -> dict:
    """Snapshot-based true revert, atomic (reference: self-mod.ts:57-84)."""
    audit = q.get_self_mod_audit(db, audit_id)
    if audit is None:
        raise ValueError(f"Audit {audit_id} not found")
    if not audit["reversible"]:
        raise ValueError("Modification is not reversible")
    if audit["reverted"]:
        raise ValueError("Already reverted")
    snap = q.get_self_mod_snapshot(db, audit_id)
    if snap is None:
        raise ValueError("No snapshot available for revert")
    with_tx = db
    try:
        with_tx.execute("BEGIN")
    except sqlite3.OperationalError:
        pass  # already in a transaction
    try:
        if snap["target_type"] == "skill" and snap["target_id"]:
            current = q.get_skill(with_tx, snap["target_id"])
            if current and current["content"] != snap["new_content"]:
                raise ValueError("Content changed since modification; refusing revert")
            with_tx.execute(
                "UPDATE skills SET content = ?, version = version + 1, updated_at = ?"
                " WHERE id = ?",
                (snap["old_content"], q.now_iso(), snap["target_id"]))
        q.mark_self_mod_reverted(with_tx, audit_id)
        with_tx.commit()
    except Exception:
        with_tx.rollback()
        raise
    return {"audit_id": audit_id, "reverted": True}
