"""File-watch path allowlist (reference: src/shared/watch-path.ts — home/temp
only, denies .ssh/.aws/etc., resolves symlinks)."""
from __future__ import annotations

import tempfile
from pathlib import Path

DENIED_SEGMENTS = (".ssh", ".aws", ".gnupg", ".roomamd", ".quoroom", ".config",
                   ".env", "id_rsa", "id_ed25519")


def validate_watch_path(path: str) -> tuple[bool, str]:
    try:
        resolved = Path(path).expanduser().resolve()
    except (OSError, RuntimeError):
        return False, "unresolvable path"
    home = Path.home().resolve()
    tmp = Path(tempfile.gettempdir()).resolve()
    if not (str(resolved).startswith(str(home))
            or str(resolved).startswith(str(tmp))):
        return False, "path must be under home or temp"
    for seg in resolved.parts:
        if seg.lower() in DENIED_SEGMENTS:
            return False, f"denied path segment: {seg}"
    return True, "ok"
