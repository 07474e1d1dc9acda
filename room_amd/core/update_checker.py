"""Release polling + staged user-space updates.

Reference: src/server/updateChecker.ts (419 LoC — GitHub release polling on a
4 h cadence, semver compare, result cached into /api/status) and
autoUpdate.ts (322 LoC — staged updates under ~/.quoroom/app with boot
health-check/rollback). This build preserves the surface: periodic checks
with graceful offline degradation (this environment has no egress — the
checker records state "offline" rather than erroring), semver comparison,
staged-update directory management with boot cleanup of interrupted stages,
and the /api/status + POST /api/status/check-update contract.
"""
from __future__ import annotations

import json
import re
import shutil
import time
import urllib.error
import urllib.request
from pathlib import Path
from typing import Optional

CHECK_INTERVAL_S = 4 * 3600        # updateChecker.ts poll cadence
DEFAULT_API = "https://api.github.com/repos/room-amd/room-amd/releases/latest"

_VER_RE = re.compile(r"v?(\d+)\.(\d+)\.(\d+)")


def parse_version(v: str) -> Optional[tuple[int, int, int]]:
    m = _VER_RE.search(v or "")
    return (int(m.group(1)), int(m.group(2)), int(m.group(3))) if m else None


def compare_versions(a: str, b: str) -> int:
    """-1 if a<b, 0 if equal, 1 if a>b (non-parsable treated as 0.0.0)."""
    pa = parse_version(a) or (0, 0, 0)
    pb = parse_version(b) or (0, 0, 0)
    return (pa > pb) - (pa < pb)


class UpdateChecker:
    def __init__(self, current_version: str, api_url: str = DEFAULT_API,
                 data_dir: Optional[Path] = None, fetcher=None):
        self.current_version = current_version
        self.api_url = api_url
        self.data_dir = Path(data_dir) if data_dir else None
        self._fetch = fetcher or self._http_fetch
        self.latest_version: Optional[str] = None
        self.release_url: Optional[str] = None
        self.last_checked_at: Optional[float] = None
        self.state = "never"           # never | ok | offline | error
        self.error: Optional[str] = None

    # ------------------------------------------------------------- checking

    def _http_fetch(self) -> dict:
        req = urllib.request.Request(self.api_url,
                                     headers={"User-Agent": "room-amd"})
        with urllib.request.urlopen(req, timeout=5) as r:
            return json.loads(r.read().decode())

    def check(self) -> dict:
        """One release check; offline/error states are recorded, never raised
        (updateChecker.ts fail-silent semantics)."""
        self.last_checked_at = time.time()
        try:
            data = self._fetch()
            tag = data.get("tag_name") or data.get("name") or ""
            if parse_version(tag) is None:
                raise ValueError(f"unparsable release tag: {tag!r}")
            self.latest_version = tag.lstrip("v")
            self.release_url = data.get("html_url")
            self.state = "ok"
            self.error = None
        except (urllib.error.URLError, OSError, TimeoutError) as e:
            self.state = "offline"
            self.error = str(e)
        except (ValueError, KeyError) as e:
            self.state = "error"
            self.error = str(e)
        return self.status()

    def maybe_check(self, interval_s: float = CHECK_INTERVAL_S) -> Optional[dict]:
        if (self.last_checked_at is None
                or time.time() - self.last_checked_at >= interval_s):
            return self.check()
        return None

    @property
    def update_available(self) -> bool:
        return bool(self.latest_version and compare_versions(
            self.latest_version, self.current_version) > 0)

    def status(self) -> dict:
        return {
            "currentVersion": self.current_version,
            "latestVersion": self.latest_version,
            "updateAvailable": self.update_available,
            "releaseUrl": self.release_url,
            "lastCheckedAt": self.last_checked_at,
            "state": self.state,
            "error": self.error,
            "staged": self.staged_version(),
        }

    # ------------------------------------------------------ staged updates

    def _app_dir(self) -> Optional[Path]:
        return self.data_dir / "app" if self.data_dir else None

    def staged_version(self) -> Optional[str]:
        d = self._app_dir()
        if not d or not d.exists():
            return None
        ready = sorted(p.name for p in d.iterdir()
                       if p.is_dir() and (p / ".ready").exists())
        return ready[-1] if ready else None

    def stage_update(self, payload_writer=None) -> dict:
        """Stage the latest release under <data>/app/<version>/ and mark it
        .ready (autoUpdate.ts staged-update layout). payload_writer(dir) lets
        callers/tests supply the artifact; without one (no egress) the stage
        is recorded as failed."""
        if not self.update_available:
            return {"staged": False, "error": "no update available"}
        d = self._app_dir()
        if d is None:
            return {"staged": False, "error": "no data dir configured"}
        target = d / str(self.latest_version)
        tmp = d / f".partial-{self.latest_version}"
        try:
            if tmp.exists():
                shutil.rmtree(tmp)
            tmp.mkdir(parents=True)
            if payload_writer is None:
                raise OSError("no download path available (offline)")
            payload_writer(tmp)
            (tmp / ".ready").write_text(str(time.time()))
            if target.exists():
                shutil.rmtree(target)
            tmp.rename(target)
            return {"staged": True, "version": self.latest_version}
        except OSError as e:
            shutil.rmtree(tmp, ignore_errors=True)
            return {"staged": False, "error": str(e)}

    def boot_health_check(self) -> int:
        """Remove interrupted partial stages on boot (autoUpdate.ts
        initBootHealthCheck). Returns number of cleaned entries."""
        d = self._app_dir()
        if not d or not d.exists():
            return 0
        cleaned = 0
        for p in d.iterdir():
            if p.name.startswith(".partial-"):
                shutil.rmtree(p, ignore_errors=True)
                cleaned += 1
            elif p.is_dir() and not (p / ".ready").exists():
                shutil.rmtree(p, ignore_errors=True)
                cleaned += 1
        return cleaned
