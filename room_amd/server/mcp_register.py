"""Global MCP registration into AI-client configs.

Reference: src/server/index.ts:729-864 — on boot the server patches the MCP
server entry into every AI client config it finds (Claude Code ~/.claude.json,
Claude Desktop, Cursor, Windsurf as JSON `mcpServers` maps; Codex as a TOML
section) and auto-approves the tools in Claude Code settings so headless
queen sessions never block on permission prompts. Same semantics here:
only EXISTING config files are patched (no new files are created), invalid
JSON is overwritten like the reference does, and ROOMAMD_SKIP_MCP_REGISTER=1
skips everything.
"""
from __future__ import annotations

import json
import os
import re
import sys
from pathlib import Path


def _entry(db_path: str, source: str) -> dict:
    return {"command": sys.executable,
            "args": ["-m", "room_amd.cli", "mcp"],
            "env": {"ROOMAMD_DB_PATH": db_path, "ROOMAMD_SOURCE": source}}


def patch_mcp_config(config_path: Path, entry: dict,
                     server_name: str = "room-amd") -> bool:
    """Merge mcpServers[server_name] into an existing JSON config."""
    try:
        if not config_path.exists():
            return False
        try:
            config = json.loads(config_path.read_text())
            if not isinstance(config, dict):
                config = {}
        except ValueError:
            config = {}          # invalid JSON — overwrite (index.ts:735)
        servers = config.get("mcpServers")
        if not isinstance(servers, dict):
            servers = {}
        servers[server_name] = entry
        config["mcpServers"] = servers
        config_path.write_text(json.dumps(config, indent=2) + "\n")
        return True
    except OSError:
        return False


def patch_codex_config(config_path: Path, db_path: str) -> bool:
    """Codex stores MCP servers as TOML sections (index.ts:751-783)."""
    try:
        if not config_path.exists():
            return False
        lines = config_path.read_text().split("\n")
        filtered, skipping = [], False
        for line in lines:
            if re.match(r"^\[mcp_servers\.room-amd[\].]", line):
                skipping = True
                continue
            if skipping and line.startswith("["):
                skipping = False
            if not skipping:
                filtered.append(line)
        content = "\n".join(filtered).rstrip()
        content += (f"\n\n[mcp_servers.room-amd]\n"
                    f"command = '{sys.executable}'\n"
                    f"args = ['-m', 'room_amd.cli', 'mcp']\n\n"
                    f"[mcp_servers.room-amd.env]\n"
                    f"ROOMAMD_DB_PATH = '{db_path}'\n"
                    f'ROOMAMD_SOURCE = "codex"\n')
        config_path.write_text(content)
        return True
    except OSError:
        return False


def patch_claude_code_permissions(home: Path) -> bool:
    """Auto-approve room-amd MCP tools in ~/.claude/settings.json so headless
    queen sessions don't block on prompts (index.ts:787-820)."""
    settings_path = home / ".claude" / "settings.json"
    try:
        if not settings_path.exists():
            return False
        try:
            settings = json.loads(settings_path.read_text())
            if not isinstance(settings, dict):
                settings = {}
        except ValueError:
            settings = {}
        perms = settings.get("permissions")
        if not isinstance(perms, dict):
            perms = {}
        allow = perms.get("allow")
        if not isinstance(allow, list):
            allow = []
        if "mcp__room-amd__*" not in allow:
            allow.append("mcp__room-amd__*")
        perms["allow"] = allow
        settings["permissions"] = perms
        settings_path.write_text(json.dumps(settings, indent=2) + "\n")
        return True
    except OSError:
        return False


def register_mcp_globally(db_path: str, home: Path | None = None) -> dict:
    """Patch every AI-client config found; returns {client: patched?}."""
    if os.environ.get("ROOMAMD_SKIP_MCP_REGISTER") == "1":
        return {}
    home = home or Path.home()
    out = {}
    out["claude-code"] = patch_mcp_config(home / ".claude.json",
                                          _entry(db_path, "claude-code"))
    out["claude-code-permissions"] = patch_claude_code_permissions(home)
    if sys.platform == "darwin":
        desktop = (home / "Library" / "Application Support" / "Claude"
                   / "claude_desktop_config.json")
    else:
        desktop = home / ".config" / "Claude" / "claude_desktop_config.json"
    out["claude-desktop"] = patch_mcp_config(desktop,
                                             _entry(db_path, "claude-desktop"))
    out["cursor"] = patch_mcp_config(home / ".cursor" / "mcp.json",
                                     _entry(db_path, "cursor"))
    out["windsurf"] = patch_mcp_config(
        home / ".codeium" / "windsurf" / "mcp_config.json",
        _entry(db_path, "windsurf"))
    out["codex"] = patch_codex_config(home / ".codex" / "config.toml", db_path)
    return out
