"""Cloud-model provider family + API-key resolution chain + HTTP executors
(reference model-provider.ts:31-141, agent-executor.ts:316-362/499-603)."""
import json
from unittest import mock

import pytest

from room_amd.core import room as room_mod
from room_amd.core.secret_store import encrypt_secret
from room_amd.db import LockedDb, init_test_db
from room_amd.db import queries as q
from room_amd.engine.cloud_providers import (HttpChatEngine,
                                             get_model_auth_status,
                                             provider_of,
                                             resolve_api_key_for_model)
from room_amd.engine.providers import get_model_provider, resolve_engine
from room_amd.engine.types import AgentExecutionOptions


def test_provider_of_families():
    assert provider_of("openai:gpt-4o") == "openai"
    assert provider_of("anthropic:claude-sonnet") == "anthropic"
    assert provider_of("claude-api:claude-sonnet") == "anthropic"
    assert provider_of("gemini:flash") == "gemini"
    assert provider_of("ollama:qwen3") == "ollama"
    assert provider_of("qwen3-coder-30b") is None
    assert provider_of("stub") is None
    assert get_model_provider("openai:gpt-4o") == "openai"
    assert get_model_provider("qwen3-coder-30b") == "local"


def test_key_resolution_chain(monkeypatch):
    ldb = LockedDb(init_test_db())
    monkeypatch.delenv("OPENAI_API_KEY", raising=False)
    with ldb as db:
        r1 = room_mod.create_room(db, "r1", worker_model="stub")
        r2 = room_mod.create_room(db, "r2", worker_model="stub")
    # 4) env fallback (lowest priority)
    assert resolve_api_key_for_model(ldb, "openai:gpt", r1["id"]) is None
    monkeypatch.setenv("OPENAI_API_KEY", "sk-env")
    assert resolve_api_key_for_model(ldb, "openai:gpt", r1["id"]) == "sk-env"
    # 3) clerk key beats env
    with ldb as db:
        q.set_setting(db, "clerk_api_key_openai_api", encrypt_secret("sk-clerk"))
    assert resolve_api_key_for_model(ldb, "openai:gpt", r1["id"]) == "sk-clerk"
    # 2) any room's credential beats clerk
    with ldb as db:
        q.set_credential(db, r2["id"], "openai_api", encrypt_secret("sk-r2"))
    assert resolve_api_key_for_model(ldb, "openai:gpt", r1["id"]) == "sk-r2"
    # 1) this room's credential wins
    with ldb as db:
        q.set_credential(db, r1["id"], "openai_api", encrypt_secret("sk-r1"))
    assert resolve_api_key_for_model(ldb, "openai:gpt", r1["id"]) == "sk-r1"
    # other rooms still get the chain fallback
    assert resolve_api_key_for_model(ldb, "openai:gpt", r2["id"]) == "sk-r2"


def _fake_urlopen(payload_holder, response):
    class R:
        def __enter__(self):
            return self
        def __exit__(self, *a):
            return False
        def read(self):
            return json.dumps(response).encode()
    def opener(req, timeout=0):
        payload_holder.append((req.full_url, json.loads(req.data),
                               dict(req.headers)))
        return R()
    return opener


def test_openai_compatible_executor():
    sent = []
    eng = HttpChatEngine("openai:gpt-4o", api_key="sk-x")
    resp = {"choices": [{"message": {"content": "hello back"}}],
            "usage": {"prompt_tokens": 11, "completion_tokens": 3}}
    with mock.patch("urllib.request.urlopen", _fake_urlopen(sent, resp)):
        text, itok, otok = eng.chat(
            [{"role": "system", "content": "sys"},
             {"role": "user", "content": "hi"}], [],
            AgentExecutionOptions(prompt="hi", model="openai:gpt-4o"))
    assert text == "hello back" and (itok, otok) == (11, 3)
    url, body, headers = sent[0]
    assert "api.openai.com" in url and body["model"] == "gpt-4o"
    assert headers["Authorization"] == "Bearer sk-x"


def test_anthropic_executor_shape():
    sent = []
    eng = HttpChatEngine("anthropic:claude-s", api_key="sk-a")
    resp = {"content": [{"type": "text", "text": "claude says"}],
            "usage": {"input_tokens": 7, "output_tokens": 2}}
    with mock.patch("urllib.request.urlopen", _fake_urlopen(sent, resp)):
        text, itok, otok = eng.chat(
            [{"role": "system", "content": "sys"},
             {"role": "user", "content": "hi"}], [],
            AgentExecutionOptions(prompt="hi", model="anthropic:claude-s"))
    assert text == "claude says" and (itok, otok) == (7, 2)
    url, body, headers = sent[0]
    assert "api.anthropic.com" in url
    assert body["system"] == "sys"
    assert headers["X-api-key"] == "sk-a" or headers.get("x-api-key") == "sk-a"


def test_offline_raises_explicit_error():
    eng = HttpChatEngine("openai:gpt-4o", api_key="sk",
                         endpoint="http://127.0.0.1:9/none")
    with pytest.raises(RuntimeError, match="unreachable"):
        eng.chat([{"role": "user", "content": "x"}], [],
                 AgentExecutionOptions(prompt="x", model="openai:gpt-4o"))


def test_resolve_engine_dispatch_and_auth_status():
    eng = resolve_engine("openai:some-model")
    assert isinstance(eng, HttpChatEngine)
    ldb = LockedDb(init_test_db())
    st = get_model_auth_status(ldb)
    assert st["local"]["ready"] is True
    assert st["ollama"]["ready"] is False
    assert st["openai"]["ready"] in (True, False)
