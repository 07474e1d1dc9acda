"""Auth failure-path matrix over EVERY route (reference pattern:
src/server/__tests__ exercises 401/403 per route class).

- no token → 401 on every /api route (except the public handshake and
  webhook pass-through)
- member token → 403 on every write route outside the collaboration
  whitelist (access.ts:13-24), and never 401/403 on reads
"""
import pytest
from fastapi.testclient import TestClient

from room_amd.db import LockedDb, init_test_db
from room_amd.server.app import create_app
from room_amd.server.auth import MEMBER_WRITE_WHITELIST

PUBLIC = {"/api/auth/handshake", "/api/auth/verify"}
PUBLIC_PREFIX = ("/api/hooks/",)


def _routes():
    app = create_app(LockedDb(init_test_db()))
    out = []
    for r in app.routes:
        if not hasattr(r, "methods") or not r.path.startswith("/api"):
            continue
        for m in r.methods:
            if m in ("HEAD", "OPTIONS"):
                continue
            out.append((m, r.path))
    return sorted(set(out))

ROUTES = _routes()
WRITES = [(m, p) for m, p in ROUTES if m in ("POST", "PUT", "PATCH", "DELETE")]


@pytest.fixture(scope="module")
def clients():
    ldb = LockedDb(init_test_db())
    app = create_app(ldb)
    # raise_server_exceptions=False: a handler 500 on an empty probe body is
    # fine here — this matrix only asserts the auth layer's 401/403 behavior
    anon = TestClient(app, raise_server_exceptions=False)
    member_tok = app.state.ctx.auth.issue_member_token("m")
    member = TestClient(app, raise_server_exceptions=False,
                        headers={"Authorization": f"Bearer {member_tok}"})
    return anon, member


def _probe_path(path: str) -> str:
    out = []
    for seg in path.split("/"):
        out.append("1" if seg.startswith("{") else seg)
    return "/".join(out)


@pytest.mark.parametrize("method,path", ROUTES,
                         ids=[f"{m} {p}" for m, p in ROUTES])
def test_unauthenticated_401(clients, method, path):
    anon, _ = clients
    if path in PUBLIC or path.startswith(PUBLIC_PREFIX):
        pytest.skip("public endpoint")
    r = anon.request(method, _probe_path(path),
                     json={} if method in ("POST", "PUT", "PATCH") else None)
    assert r.status_code == 401, f"{method} {path} -> {r.status_code}"


@pytest.mark.parametrize("method,path", WRITES,
                         ids=[f"{m} {p}" for m, p in WRITES])
def test_member_write_403_outside_whitelist(clients, method, path):
    _, member = clients
    if path in PUBLIC or path.startswith(PUBLIC_PREFIX):
        pytest.skip("public endpoint")
    whitelisted = any(m == method and w.replace("{room_id}", "{room_id}") == path
                      or (m == method and
                          w.split("/")[-1] == path.split("/")[-1] and
                          w.count("/") == path.count("/"))
                      for m, w in MEMBER_WRITE_WHITELIST)
    r = member.request(method, _probe_path(path), json={})
    if whitelisted:
        assert r.status_code != 403, f"{method} {path} wrongly forbidden"
    else:
        assert r.status_code == 403, f"{method} {path} -> {r.status_code}"


@pytest.mark.parametrize("method,path",
                         [(m, p) for m, p in ROUTES if m == "GET"],
                         ids=[p for m, p in ROUTES if m == "GET"])
def test_member_read_allowed(clients, method, path):
    """Members read everything (404/400/422 for bad ids are fine; 401/403
    are not)."""
    _, member = clients
    r = member.get(_probe_path(path))
    assert r.status_code not in (401, 403), f"{path} -> {r.status_code}"
