"""Worker prompt markdown sync (reference: src/shared/worker-prompt-sync.ts).

Explicit export/import of worker system prompts as YAML-frontmatter markdown
under <base>/.roomamd/prompts/workers/room-<id>/worker-<id>.md with
newest-mtime-wins conflict policy and a `force` override.
"""
from __future__ import annotations

import os
import re
import sqlite3
from datetime import datetime
from pathlib import Path

from ..db import queries as q


def prompts_dir(base: str | None = None) -> Path:
    root = Path(base) if base else Path(
        os.environ.get("ROOMAMD_DATA_DIR", str(Path.home() / ".roomamd")))
    return root / "prompts" / "workers"


def _worker_file(base: Path, room_id: int, worker_id: int) -> Path:
    return base / f"room-{room_id}" / f"worker-{worker_id}.md"


def render_worker_md(worker: dict) -> str:
    return (f"---\n"
            f"worker_id: {worker['id']}\n"
            f"name: {worker['name']}\n"
            f"role: {worker.get('role') or ''}\n"
            f"model: {worker.get('model') or ''}\n"
            f"updated_at: {worker.get('updated_at') or ''}\n"
            f"---\n\n{worker['system_prompt']}\n")


_FM_RE = re.compile(r"^---\n(.*?)\n---\n\n?(.*)$", re.S)


def parse_worker_md(text: str) -> tuple[dict, str]:
    m = _FM_RE.match(text)
    if not m:
        return {}, text.strip()
    meta = {}
    for line in m.group(1).splitlines():
        if ":" in line:
            k, v = line.split(":", 1)
            meta[k.strip()] = v.strip()
    return meta, m.group(2).rstrip("\n")


def export_worker_prompts(db: sqlite3.Connection, room_id: int,
                          base: str | None = None) -> list[str]:
    root = prompts_dir(base)
    out = []
    for w in q.list_room_workers(db, room_id):
        path = _worker_file(root, room_id, w["id"])
        path.parent.mkdir(parents=True, exist_ok=True)
        path.write_text(render_worker_md(w))
        out.append(str(path))
    return out


def import_worker_prompts(db: sqlite3.Connection, room_id: int,
                          base: str | None = None,
                          force: bool = False) -> list[dict]:
    """Newest-mtime-wins: the file only overwrites the DB prompt if the file
    is newer than the worker row's updated_at (or force=True)."""
    root = prompts_dir(base) / f"room-{room_id}"
    results = []
    if not root.exists():
        return results
    for path in sorted(root.glob("worker-*.md")):
        meta, prompt = parse_worker_md(path.read_text())
        try:
            worker_id = int(meta.get("worker_id") or
                            path.stem.split("-", 1)[1])
        except (ValueError, IndexError):
            continue
        w = q.get_worker(db, worker_id)
        if w is None or w["room_id"] != room_id:
            results.append({"worker_id": worker_id, "action": "skipped-missing"})
            continue
        if not force:
            try:
                db_mtime = datetime.strptime(w["updated_at"],
                                             "%Y-%m-%d %H:%M:%S").timestamp()
            except (ValueError, TypeError):
                db_mtime = 0
            if path.stat().st_mtime <= db_mtime:
                results.append({"worker_id": worker_id, "action": "kept-db"})
                continue
        if prompt and prompt != w["system_prompt"]:
            q.update_worker(db, worker_id, system_prompt=prompt)
            results.append({"worker_id": worker_id, "action": "imported"})
        else:
            results.append({"worker_id": worker_id, "action": "unchanged"})
    return results
