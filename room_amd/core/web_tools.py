"""Keyless web tools (reference: src/shared/web-tools.ts — webFetch via reader
services, webSearch, persistent headless-browser sessions).

Tool contract preserved (webFetch/webSearch/browser_action shapes); this
environment has no egress and no browser, so calls degrade to explicit
"network unavailable" results rather than hanging (the reference also falls
back through providers and reports failures). A urllib path serves networked
deployments.
"""
from __future__ import annotations

import re
import urllib.error
import urllib.parse
import urllib.request

FETCH_TIMEOUT_S = 10
_BROWSER_SESSIONS: dict[str, dict] = {}


def web_fetch(url: str, max_chars: int = 20_000) -> dict:
    if not re.match(r"^https?://", url):
        return {"ok": False, "error": "only http(s) urls"}
    try:
        req = urllib.request.Request(url, headers={"User-Agent": "room-amd/0.1"})
        with urllib.request.urlopen(req, timeout=FETCH_TIMEOUT_S) as resp:
            body = resp.read(max_chars * 4).decode("utf-8", "replace")
        text = re.sub(r"<script.*?</script>|<style.*?</style>", "", body,
                      flags=re.S | re.I)
        text = re.sub(r"<[^>]+>", " ", text)
        text = re.sub(r"\s+", " ", text).strip()
        return {"ok": True, "url": url, "content": text[:max_chars]}
    except (urllib.error.URLError, OSError, ValueError) as e:
        return {"ok": False, "url": url, "error": f"network unavailable: {e}"}


def web_search(query: str, limit: int = 5) -> dict:
    """DuckDuckGo lite search when egress exists; explicit failure otherwise."""
    try:
        url = ("https://html.duckduckgo.com/html/?q="
               + urllib.parse.quote(query))
        out = web_fetch(url, max_chars=40_000)
        if not out["ok"]:
            return out
        # crude result extraction from the lite page
        links = re.findall(r"(https?://[^\s\"<>]+)", out["content"])[:limit]
        return {"ok": True, "query": query, "results": links}
    except Exception as e:
        return {"ok": False, "query": query, "error": str(e)}


def browser_action(session_id: str, action: str, **kwargs) -> dict:
    """Persistent browser-session action (reference browserActionPersistent,
    web-tools.ts:456-599). Drives headless Chromium over CDP when a binary
    exists (core/browser.py: persistent user-data-dir sessions, 30-min idle
    GC, accessibility-style snapshots); otherwise degrades to an urllib
    fetch for goto/navigate and an explicit error for interactions — the
    reference's exact 'Chromium not installed' fallback semantics."""
    from . import browser as br
    url = kwargs.get("url", "")
    if br.find_chromium() is not None:
        acts = [] if action in ("goto", "navigate", "snapshot") else [
            {"type": action, **kwargs}]
        out = br.browser_action(url, acts, session_id=session_id or None)
        ok = not out["snapshot"].startswith(("browser error",
                                             "Chromium not installed"))
        return {"ok": ok, "session": out["sessionId"] or session_id,
                "action": action, "url": out["url"],
                "snapshot": out["snapshot"]}
    if action in ("goto", "navigate") and url:
        return web_fetch(url)
    return {"ok": False, "error": "no browser runtime in this environment "
                                  "(install chromium or set ROOMAMD_CHROMIUM)",
            "session": session_id, "action": action}


def close_browser(session_id: str) -> bool:
    from . import browser as br
    had = session_id in br._sessions
    br.close_session(session_id)
    return had or _BROWSER_SESSIONS.pop(session_id, None) is not None
