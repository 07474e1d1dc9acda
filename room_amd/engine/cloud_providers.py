"""Cloud-model providers + API-key resolution chain.

Reference: src/shared/model-provider.ts (model string → provider family,
key resolution chain room credential → any room's credential → clerk key →
env var, :87-141) and agent-executor.ts's OpenAI-compatible / Anthropic HTTP
executors (:316-362, 499-603). The perf path is the in-process engine
(SURVEY §2b scopes cloud models out of it); these providers preserve the
*semantics* — the same ChatEngine protocol and the same multi-turn tool
loop drive them — and degrade with explicit errors when there is no egress.
"""
from __future__ import annotations

import json
import os
import urllib.error
import urllib.request
from typing import Optional

from ..core.secret_store import decrypt_secret
from .types import AgentExecutionOptions, ToolDef

PROVIDER_ENV = {          # model-provider.ts env-var fallbacks
    "openai": "OPENAI_API_KEY",
    "anthropic": "ANTHROPIC_API_KEY",
    "gemini": "GEMINI_API_KEY",
}
PROVIDER_CRED = {         # credential row names per provider family
    "openai": "openai_api",
    "anthropic": "anthropic_api",
    "gemini": "gemini_api",
}
ENDPOINTS = {
    "openai": "https://api.openai.com/v1/chat/completions",
    "gemini": ("https://generativelanguage.googleapis.com/v1beta/openai/"
               "chat/completions"),          # OpenAI-compat surface
    "ollama": "http://127.0.0.1:11434/v1/chat/completions",
    "anthropic": "https://api.anthropic.com/v1/messages",
}


def provider_of(model: str) -> Optional[str]:
    """Cloud-provider family of a model string, None for local/stub
    (model-provider.ts:31-41)."""
    for p in ("openai", "anthropic", "gemini", "ollama"):
        if model.startswith(p + ":"):
            return p
    if model.startswith("claude-api:"):
        return "anthropic"
    return None


def resolve_api_key_for_model(ldb, model: str,
                              room_id: int | None = None) -> Optional[str]:
    """Key chain (model-provider.ts:87-141): this room's credential → any
    room's credential → clerk key setting → environment variable."""
    prov = provider_of(model)
    if prov in (None, "ollama"):
        return None
    cred_name = PROVIDER_CRED[prov]
    with ldb as db:
        if room_id is not None:
            row = db.execute(
                "SELECT value_encrypted FROM credentials WHERE room_id = ?"
                " AND name = ?", (room_id, cred_name)).fetchone()
            if row:
                return decrypt_secret(row["value_encrypted"])
        row = db.execute(
            "SELECT value_encrypted FROM credentials WHERE name = ?"
            " ORDER BY id LIMIT 1", (cred_name,)).fetchone()
        if row:
            return decrypt_secret(row["value_encrypted"])
        clerk = db.execute("SELECT value FROM settings WHERE key = ?",
                           (f"clerk_api_key_{cred_name}",)).fetchone()
        if clerk and clerk["value"]:
            try:
                return decrypt_secret(clerk["value"])
            except Exception:
                return clerk["value"]
    return os.environ.get(PROVIDER_ENV[prov]) or None


def _tooldefs_openai(tools: list[ToolDef]) -> list[dict]:
    return [{"type": "function",
             "function": {"name": t.name, "description": t.description,
                          "parameters": t.parameters}} for t in tools]


class HttpChatEngine:
    """ChatEngine over a provider HTTP API. One request per chat() turn; the
    shared execute_agent loop (providers.py) handles the multi-turn tool
    protocol, so tool calls come back as <tool_call> text via the prompt
    convention — identical observable semantics to the local engine."""

    def __init__(self, model: str, api_key: str | None = None,
                 endpoint: str | None = None):
        self.provider = provider_of(model) or "openai"
        self.model_name = model.split(":", 1)[1] if ":" in model else model
        self.api_key = api_key
        self.endpoint = endpoint or os.environ.get(
            f"ROOMAMD_{self.provider.upper()}_ENDPOINT",
            ENDPOINTS[self.provider])

    def _request(self, payload: dict, headers: dict) -> dict:
        req = urllib.request.Request(
            self.endpoint, data=json.dumps(payload).encode(),
            headers={"Content-Type": "application/json", **headers})
        with urllib.request.urlopen(req, timeout=120) as r:
            return json.loads(r.read())

    def chat(self, messages: list[dict], tools: list[ToolDef],
             options: AgentExecutionOptions) -> tuple[str, int, int]:
        try:
            if self.provider == "anthropic":
                sys_msg = next((m["content"] for m in messages
                                if m["role"] == "system"), "")
                payload = {
                    "model": self.model_name,
                    "max_tokens": options.max_new_tokens,
                    "system": sys_msg,
                    "messages": [
                        {"role": "user" if m["role"] in ("user", "tool")
                         else "assistant", "content": m["content"]}
                        for m in messages if m["role"] != "system"],
                }
                out = self._request(payload, {
                    "x-api-key": self.api_key or "",
                    "anthropic-version": "2023-06-01"})
                text = "".join(b.get("text", "")
                               for b in out.get("content", []))
                usage = out.get("usage", {})
                return (text, usage.get("input_tokens", 0),
                        usage.get("output_tokens", 0))
            payload = {
                "model": self.model_name,
                "messages": [{"role": ("assistant" if m["role"] == "tool"
                                       else m["role"]),
                              "content": m["content"]} for m in messages],
                "max_tokens": options.max_new_tokens,
                "temperature": options.temperature,
            }
            headers = {}
            if self.api_key:
                headers["Authorization"] = f"Bearer {self.api_key}"
            out = self._request(payload, headers)
            choice = (out.get("choices") or [{}])[0]
            text = (choice.get("message") or {}).get("content") or ""
            usage = out.get("usage", {})
            return (text, usage.get("prompt_tokens", 0),
                    usage.get("completion_tokens", 0))
        except urllib.error.URLError as e:
            raise RuntimeError(
                f"{self.provider} API unreachable ({e}); this host has no "
                "egress — use the in-process engine") from e


def get_model_auth_status(ldb) -> dict:
    """Readiness probe per provider family (model-provider.ts shape)."""
    out = {"local": {"ready": True, "reason": "in-process engine"}}
    for prov in ("openai", "anthropic", "gemini"):
        key = resolve_api_key_for_model(ldb, f"{prov}:probe")
        out[prov] = {"ready": bool(key),
                     "reason": "key configured" if key else "no API key"}
    out["ollama"] = {"ready": False,
                     "reason": "no sidecar (engine is in-process)"}
    return out
