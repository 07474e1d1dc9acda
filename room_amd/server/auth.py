"""Dual-token auth + RBAC + cloud JWT (reference: src/server/auth.ts, access.ts).

- agent token: 64-hex persisted at ~/.roomamd/api.token (mode 0600)
- user token: issued by the localhost-only /api/auth/handshake, persisted in
  auth.tokens.json
- member tokens: read-only + whitelisted collaboration POSTs
- cloud HS256 JWTs: iss 'quoroom-cloud', aud 'quoroom-runtime', instanceId
  pinning, nbf/exp windows (auth.ts:106-165) — accepted when
  ROOMAMD_CLOUD_JWT_SECRET is configured
- timing-safe comparison everywhere
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import secrets
import time
from pathlib import Path

ROLE_AGENT = "agent"
ROLE_USER = "user"
ROLE_MEMBER = "member"

# member whitelist: read-only plus these write endpoints (access.ts:13-24),
# plus the harmless presence/typing heartbeats the SPA posts for any viewer
MEMBER_WRITE_WHITELIST = {
    ("POST", "/api/rooms/{room_id}/messages"),
    ("POST", "/api/rooms/{room_id}/escalations"),
    ("POST", "/api/decisions/{decision_id}/vote"),
    ("POST", "/api/decisions/{decision_id}/object"),
    ("POST", "/api/rooms/{room_id}/chat"),
    ("POST", "/api/clerk/presence"),
    ("POST", "/api/clerk/typing"),
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
        claims = validate_cloud_jwt(token)
        if claims is not None:
            return ROLE_MEMBER if claims.get("role") == "member" else ROLE_USER
        return None


# ------------------------------------------------------------- cloud JWT
# HS256 validation with issuer/audience/instanceId pinning (auth.ts:106-165).
# No external jwt dependency: header/payload are base64url JSON, signature is
# HMAC-SHA256 over "<header>.<payload>" with the shared cloud secret.

JWT_ISS = "quoroom-cloud"
JWT_AUD = "quoroom-runtime"


def _b64url_decode(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def _b64url_encode(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).decode().rstrip("=")


def cloud_jwt_secret() -> str | None:
    return os.environ.get("ROOMAMD_CLOUD_JWT_SECRET") or None


def cloud_instance_id() -> str | None:
    return os.environ.get("ROOMAMD_CLOUD_INSTANCE_ID") or None


def validate_cloud_jwt(token: str) -> dict | None:
    """Returns the claims dict (with normalized 'role') or None. Pinning:
    alg HS256 only, iss/aud fixed, sub required, optional instanceId must
    match this instance, nbf/exp enforced, timing-safe signature compare."""
    secret = cloud_jwt_secret()
    if not secret:
        return None
    parts = token.split(".")
    if len(parts) != 3:
        return None
    try:
        header = json.loads(_b64url_decode(parts[0]))
        payload = json.loads(_b64url_decode(parts[1]))
        provided = _b64url_decode(parts[2])
    except (ValueError, TypeError):
        return None
    if not isinstance(header, dict) or not isinstance(payload, dict):
        return None
    if header.get("alg") != "HS256":
        return None
    if payload.get("iss") != JWT_ISS or payload.get("aud") != JWT_AUD:
        return None
    sub = payload.get("sub")
    if not isinstance(sub, str) or not sub:
        return None
    expected_iid = cloud_instance_id()
    if expected_iid and payload.get("instanceId") != expected_iid:
        return None
    now = int(time.time())
    nbf = payload.get("nbf")
    if isinstance(nbf, (int, float)) and now < nbf:
        return None
    exp = payload.get("exp")
    if not isinstance(exp, (int, float)) or now >= exp:
        return None
    expected = hmac.new(secret.encode(),
                        f"{parts[0]}.{parts[1]}".encode(),
                        hashlib.sha256).digest()
    if not (len(expected) == len(provided)
            and hmac.compare_digest(expected, provided)):
        return None
    role = str(payload.get("role") or "").lower()
    payload["role"] = "member" if role == "member" else "user"
    return payload


def make_cloud_jwt(secret: str, sub: str = "user-1", role: str = "user",
                   instance_id: str | None = None, exp_in: int = 3600,
                   **extra) -> str:
    """Mint an HS256 cloud JWT (test helper + cloud-sync client side)."""
    header = {"alg": "HS256", "typ": "JWT"}
    payload = {"iss": JWT_ISS, "aud": JWT_AUD, "sub": sub, "role": role,
               "exp": int(time.time()) + exp_in, **extra}
    if instance_id is not None:
        payload["instanceId"] = instance_id
    h = _b64url_encode(json.dumps(header).encode())
    p = _b64url_encode(json.dumps(payload).encode())
    sig = hmac.new(secret.encode(), f"{h}.{p}".encode(), hashlib.sha256).digest()
    return f"{h}.{p}.{_b64url_encode(sig)}"


def member_can_write(method: str, route_template: str) -> bool:
    return (method, route_template) in MEMBER_WRITE_WHITELIST
