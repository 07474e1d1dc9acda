"""Persistent headless-browser automation (reference: src/shared/
web-tools.ts:44-116, 440-599 — Playwright Chromium sessions with 30-min idle
GC and accessibility-tree snapshots).

This build drives Chromium directly over the DevTools protocol (CDP) — no
Playwright in the image. Same contract as browserActionPersistent:

    browser_action(start_url, actions, session_id=None)
        -> {"snapshot": str, "sessionId": str, "url": str}

actions: [{"type": "navigate"|"click"|"type"|"press"|"scroll"|
           "waitForSelector"|"snapshot", ...}]

Sessions persist cookies/localStorage across calls (own --user-data-dir per
session), idle sessions are reaped after 30 minutes. When no Chromium binary
exists on the host (this image has none), the call degrades to the
reference's exact behavior: a result whose snapshot says the browser is not
installed — callers (web_tools.py) then fall back to urllib fetching.
"""
from __future__ import annotations

import json
import os
import re
import shutil
import subprocess
import tempfile
import time
import uuid
from typing import Optional

SESSION_IDLE_TIMEOUT_S = 30 * 60        # web-tools.ts:59
MAX_SNAPSHOT_CHARS = 8000
CHROMIUM_CANDIDATES = ["chromium", "chromium-browser", "google-chrome",
                       "google-chrome-stable", "chrome", "headless_shell"]

_sessions: dict[str, "BrowserSession"] = {}


def find_chromium() -> Optional[str]:
    env = os.environ.get("ROOMAMD_CHROMIUM")
    if env and os.path.exists(env):
        return env
    for c in CHROMIUM_CANDIDATES:
        p = shutil.which(c)
        if p:
            return p
    return None


# ---------------------------------------------------------------- session

class BrowserSession:
    def __init__(self, session_id: str, binary: str):
        self.id = session_id
        self.last_used = time.time()
        self.user_data_dir = tempfile.mkdtemp(prefix=f"roomamd-br-{session_id[:8]}-")
        self.proc = subprocess.Popen(
            [binary, "--headless=new", "--no-sandbox", "--disable-gpu",
             "--remote-debugging-port=0",
             f"--user-data-dir={self.user_data_dir}", "about:blank"],
            stdout=subprocess.DEVNULL, stderr=subprocess.PIPE)
        # Chromium forks renderers/zygotes: register for tree termination
        # on shutdown (reference process-supervisor.ts semantics)
        from .process_supervisor import register_managed_process
        register_managed_process(self.proc.pid, f"browser:{session_id[:8]}")
        self.ws_url = self._wait_devtools_url()
        self.http_base = re.sub(r"^ws://([^/]+)/.*$", r"http://\1",
                                self.ws_url)
        self._page_ws = None
        self._msg_id = 0

    def _wait_devtools_url(self, timeout: float = 15.0) -> str:
        deadline = time.time() + timeout
        buf = b""
        while time.time() < deadline:
            line = self.proc.stderr.readline()
            if not line:
                time.sleep(0.05)
                continue
            buf += line
            m = re.search(rb"DevTools listening on (ws://\S+)", buf)
            if m:
                return m.group(1).decode()
        raise RuntimeError("chromium did not expose a DevTools endpoint")

    # -------------------------------------------------------------- CDP

    def _page_target_ws(self) -> str:
        import urllib.request
        with urllib.request.urlopen(self.http_base + "/json/list",
                                    timeout=5) as r:
            targets = json.loads(r.read())
        for t in targets:
            if t.get("type") == "page":
                return t["webSocketDebuggerUrl"]
        # no page target: ask the browser to open one
        with urllib.request.urlopen(self.http_base + "/json/new?about:blank",
                                    timeout=5) as r:
            return json.loads(r.read())["webSocketDebuggerUrl"]

    def _ws(self):
        if self._page_ws is None:
            from ..utils.ws_client import WsClient
            self._page_ws = WsClient(self._page_target_ws())
        return self._page_ws

    def cdp(self, method: str, params: dict | None = None,
            timeout: float = 30.0) -> dict:
        self._msg_id += 1
        mid = self._msg_id
        ws = self._ws()
        ws.send_text(json.dumps({"id": mid, "method": method,
                                 "params": params or {}}))
        deadline = time.time() + timeout
        while time.time() < deadline:
            msg = ws.recv_text(timeout=deadline - time.time())
            if msg is None:
                break
            data = json.loads(msg)
            if data.get("id") == mid:
                if "error" in data:
                    raise RuntimeError(f"CDP {method}: {data['error']}")
                return data.get("result", {})
        raise TimeoutError(f"CDP {method} timed out")

    def evaluate(self, expr: str):
        out = self.cdp("Runtime.evaluate",
                       {"expression": expr, "returnByValue": True,
                        "awaitPromise": True})
        return out.get("result", {}).get("value")

    def navigate(self, url: str, timeout: float = 30.0) -> None:
        self.cdp("Page.enable")
        self.cdp("Page.navigate", {"url": url})
        deadline = time.time() + timeout
        while time.time() < deadline:
            state = self.evaluate("document.readyState")
            if state in ("interactive", "complete"):
                return
            time.sleep(0.1)

    def url(self) -> str:
        try:
            return self.evaluate("location.href") or ""
        except Exception:
            return ""

    def snapshot(self) -> str:
        """Accessibility-style outline of the page (role/name/text per
        interactive or labelled node — the ariaSnapshot analogue)."""
        js = r"""(() => {
  const out = [];
  const walk = (el, depth) => {
    if (!el || out.length > 400) return;
    const role = el.getAttribute && (el.getAttribute('role')
      || {A:'link',BUTTON:'button',INPUT:'textbox',SELECT:'combobox',
          TEXTAREA:'textbox',H1:'heading',H2:'heading',H3:'heading',
          NAV:'navigation',FORM:'form',TABLE:'table',IMG:'img'}[el.tagName]);
    let label = '';
    if (el.getAttribute) {
      label = el.getAttribute('aria-label') || el.getAttribute('alt')
        || el.getAttribute('placeholder') || el.getAttribute('name') || '';
    }
    const ownText = (el.children && el.children.length === 0 && el.textContent)
      ? el.textContent.trim().slice(0, 120) : '';
    if (role || (ownText && ownText.length > 1)) {
      out.push('  '.repeat(Math.min(depth, 8))
        + (role ? role + ': ' : '') + (label || ownText));
    }
    for (const c of (el.children || []))
      walk(c, depth + (role ? 1 : 0));
  };
  walk(document.body, 0);
  return out.join('\n');
})()"""
        try:
            text = self.evaluate(js) or ""
        except Exception:
            try:
                text = self.evaluate("document.body.innerText") or ""
            except Exception:
                text = "(could not read page)"
        return f"[{self.url()}]\n{text[:MAX_SNAPSHOT_CHARS]}"

    def close(self) -> None:
        try:
            if self._page_ws is not None:
                self._page_ws.close()
        except Exception:
            pass
        from .process_supervisor import terminate_tree, unregister_managed_process
        try:
            terminate_tree(self.proc.pid, grace=5.0)
            self.proc.wait(timeout=1)
        except Exception:
            pass
        unregister_managed_process(self.proc.pid)
        shutil.rmtree(self.user_data_dir, ignore_errors=True)


# ---------------------------------------------------------------- manager

def _gc_sessions() -> None:
    now = time.time()
    for sid in [s for s, sess in _sessions.items()
                if now - sess.last_used > SESSION_IDLE_TIMEOUT_S]:
        _sessions.pop(sid).close()


def close_session(session_id: str) -> None:
    s = _sessions.pop(session_id, None)
    if s:
        s.close()


def close_all_sessions() -> None:
    for s in list(_sessions.values()):
        s.close()
    _sessions.clear()


def _sel(action: dict) -> str:
    return json.dumps(action.get("selector") or "body")


def browser_action(start_url: str, actions: list[dict] | None = None,
                   session_id: str | None = None,
                   timeout_s: float = 60.0) -> dict:
    """browserActionPersistent contract (web-tools.ts:456-599)."""
    _gc_sessions()
    binary = find_chromium()
    if binary is None:
        return {"snapshot": "Chromium not installed. Install a chromium/"
                            "chrome binary or set ROOMAMD_CHROMIUM.",
                "sessionId": "", "url": ""}
    sess = _sessions.get(session_id) if session_id else None
    if sess is None:
        sid = session_id or uuid.uuid4().hex
        sess = BrowserSession(sid, binary)
        _sessions[sid] = sess
    sess.last_used = time.time()
    try:
        if start_url and sess.url() != start_url:
            sess.navigate(start_url, timeout=timeout_s)
        for action in actions or []:
            t = action.get("type")
            if t == "navigate":
                sess.navigate(action["url"], timeout=timeout_s)
            elif t == "click":
                if action.get("selector"):
                    sess.evaluate(
                        f"document.querySelector({_sel(action)})?.click()")
                elif action.get("text"):
                    txt = json.dumps(action["text"])
                    sess.evaluate(
                        "[...document.querySelectorAll('a,button,[role=button]')]"
                        f".find(e => e.textContent.includes({txt}))?.click()")
                time.sleep(0.5)
            elif t == "type":
                txt = json.dumps(action.get("text", ""))
                sess.evaluate(
                    f"(() => {{ const el = document.querySelector({_sel(action)});"
                    f" if (el) {{ el.focus(); el.value = {txt};"
                    " el.dispatchEvent(new Event('input', {bubbles:true}));"
                    " el.dispatchEvent(new Event('change', {bubbles:true})); } })()")
            elif t == "press":
                key = json.dumps(action.get("key", "Enter"))
                sess.evaluate(
                    "document.activeElement.dispatchEvent(new KeyboardEvent("
                    f"'keydown', {{key: {key}, bubbles: true}}))")
            elif t == "scroll":
                dy = int(action.get("dy", 600))
                sess.evaluate(f"window.scrollBy(0, {dy})")
            elif t == "waitForSelector":
                deadline = time.time() + float(action.get("timeout_s", 10))
                while time.time() < deadline:
                    if sess.evaluate(
                            f"!!document.querySelector({_sel(action)})"):
                        break
                    time.sleep(0.2)
        return {"snapshot": sess.snapshot(), "sessionId": sess.id,
                "url": sess.url()}
    except Exception as e:
        return {"snapshot": f"browser error: {e}", "sessionId": sess.id,
                "url": sess.url()}
