"""Dual-token auth + RBAC (reference: src/server/auth.ts, access.ts).

- agent token: 64-hex persisted at ~/.roomamd/api.token (mode 0600)
- user token: issued by the localhost-only /api/auth/handshake, persisted in
  auth.tokens.json
- member tokens: read-only + whitelisted collaboration POSTs
- timing-safe comparison everywhere
"""
from __future__ import annotations

import hmac
import json
import os
import secrets
from pathlib import Path

ROLE_AGENT = "agent"
ROLE_USER = "user"
ROLE_MEMBER = "member"

# member whitelist: read-only plus these write endpoints (access.ts:13-24)
MEMBER_WRITE_WHITELIST = {
    ("POST", "/api/rooms/{room_id}/messages"),
    ("POST", "/api/rooms/{room_id}/escalations"),
    ("POST", "/api/decisions/{decision_id}/vote"),
    ("POST", "/api/decisions/{decision_id}/object"),
    ("POST", "/api/rooms/{room_id}/chat"),
}


def data_dir() -> Path:
    d = Path(os.environ.get("ROOMAMD_DATA_DIR", str(Path.home() / ".roomamd")))
    d.mkdir(parents=True, exist_ok=True)
    return d


class AuthManager:
    def __init__(self, skip_token_file: bool = False):
        self.skip_token_file = skip_token_file
        self.agent_token = secrets.token_hex(32)
        self.user_tokens: dict[str, str] = {}   # token → label
        self.member_tokens: dict[str, str] = {}
        if not skip_token_file:
            self._persist_agent_token()
            self._load_tokens()

    def _persist_agent_token(self) -> None:
        p = data_dir() / "api.token"
        try:
            if p.exists():
                self.agent_token = p.read_text().strip()
            else:
                p.write_text(self.agent_token)
                p.chmod(0o600)
        except OSError:
            pass

    def _tokens_path(self) -> Path:
        return data_dir() / "auth.tokens.json"

    def _load_tokens(self) -> None:
        p = self._tokens_path()
        if p.exists():
            try:
                data = json.loads(p.read_text())
                self.user_tokens = data.get("user", {})
                self.member_tokens = data.get("member", {})
            except (ValueError, OSError):
                pass

    def _save_tokens(self) -> None:
        if self.skip_token_file:
            return
        p = self._tokens_path()
        try:
            p.write_text(json.dumps({"user": self.user_tokens,
                                     "member": self.member_tokens}))
            p.chmod(0o600)
        except OSError:
            pass

    def issue_user_token(self, label: str = "ui") -> str:
        token = secrets.token_hex(32)
        self.user_tokens[token] = label
        self._save_tokens()
        return token

    def issue_member_token(self, label: str = "member") -> str:
        token = secrets.token_hex(32)
        self.member_tokens[token] = label
        self._save_tokens()
        return token

    def role_for(self, token: str | None) -> str | None:
        if not token:
            return None
        if hmac.compare_digest(token, self.agent_token):
            return ROLE_AGENT
        for t in self.user_tokens:
            if hmac.compare_digest(token, t):
                return ROLE_USER
        for t in self.member_tokens:
            if hmac.compare_digest(token, t):
                return ROLE_MEMBER
        return None


def member_can_write(method: str, route_template: str) -> bool:
    return (method, route_template) in MEMBER_WRITE_WHITELIST
