"""Global MCP registration into AI-client configs (index.ts:729-864):
existing configs are patched, missing ones are left alone, invalid JSON is
overwritten, Codex TOML sections are replaced idempotently."""
import json
from pathlib import Path

from room_amd.server.mcp_register import (patch_claude_code_permissions,
                                          patch_codex_config,
                                          patch_mcp_config,
                                          register_mcp_globally)


def test_patch_json_config_merges(tmp_path):
    cfg = tmp_path / "mcp.json"
    cfg.write_text(json.dumps({"mcpServers": {"other": {"command": "x"}},
                               "theme": "dark"}))
    assert patch_mcp_config(cfg, {"command": "python"}) is True
    out = json.loads(cfg.read_text())
    assert out["mcpServers"]["room-amd"]["command"] == "python"
    assert out["mcpServers"]["other"]["command"] == "x"   # preserved
    assert out["theme"] == "dark"


def test_patch_missing_and_invalid(tmp_path):
    assert patch_mcp_config(tmp_path / "nope.json", {}) is False
    bad = tmp_path / "bad.json"
    bad.write_text("{not json")
    assert patch_mcp_config(bad, {"command": "p"}) is True  # overwrite
    assert json.loads(bad.read_text())["mcpServers"]["room-amd"]


def test_codex_toml_idempotent(tmp_path):
    cfg = tmp_path / "config.toml"
    cfg.write_text("model = 'gpt'\n\n[other]\nx = 1\n")
    assert patch_codex_config(cfg, "/tmp/db") is True
    assert patch_codex_config(cfg, "/tmp/db2") is True     # replaces section
    text = cfg.read_text()
    assert text.count("[mcp_servers.room-amd]") == 1
    assert "ROOMAMD_DB_PATH = '/tmp/db2'" in text
    assert "model = 'gpt'" in text and "[other]" in text   # untouched


def test_claude_code_permissions(tmp_path):
    (tmp_path / ".claude").mkdir()
    sp = tmp_path / ".claude" / "settings.json"
    sp.write_text(json.dumps({"permissions": {"allow": ["Bash"]}}))
    assert patch_claude_code_permissions(tmp_path) is True
    allow = json.loads(sp.read_text())["permissions"]["allow"]
    assert "Bash" in allow and "mcp__room-amd__*" in allow
    # idempotent
    patch_claude_code_permissions(tmp_path)
    allow2 = json.loads(sp.read_text())["permissions"]["allow"]
    assert allow2.count("mcp__room-amd__*") == 1


def test_register_globally_only_touches_existing(tmp_path, monkeypatch):
    (tmp_path / ".cursor").mkdir()
    (tmp_path / ".cursor" / "mcp.json").write_text("{}")
    out = register_mcp_globally("/tmp/db", home=tmp_path)
    assert out["cursor"] is True
    assert out["claude-code"] is False                    # file didn't exist
    assert not (tmp_path / ".claude.json").exists()       # never created
    entry = json.loads((tmp_path / ".cursor" / "mcp.json").read_text())
    assert entry["mcpServers"]["room-amd"]["env"]["ROOMAMD_DB_PATH"] == "/tmp/db"
    # skip switch
    monkeypatch.setenv("ROOMAMD_SKIP_MCP_REGISTER", "1")
    assert register_mcp_globally("/tmp/db", home=tmp_path) == {}
