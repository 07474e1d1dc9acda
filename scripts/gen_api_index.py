"""Regenerate docs/API.md from the live route table and MCP registry."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from room_amd.db import LockedDb, init_test_db
from room_amd.mcp.server import McpServer
from room_amd.server.app import create_app
from room_amd.server.auth import AuthManager


def main() -> None:
    app = create_app(LockedDb(init_test_db()),
                     auth=AuthManager(skip_token_file=True))
    routes = sorted(
        (getattr(r, "path", ""), m)
        for r in app.router.routes
        for m in sorted((getattr(r, "methods", None) or set())
                        - {"HEAD", "OPTIONS"})
        if getattr(r, "path", "").startswith("/"))
    mcp = McpServer(LockedDb(init_test_db()), nudge=lambda w: True)
    tools = sorted(mcp.tools.items())

    lines = ["# API surface index", "",
             "Generated from the live route table and MCP registry "
             "(`python scripts/gen_api_index.py`). The shapes mirror the",
             "reference's REST/MCP contracts (SURVEY §2d); see "
             "`docs/COMPONENT_MAP.md` for the per-module mapping.", "",
             f"## HTTP REST — {len(routes)} routes", "",
             "| Method | Path |", "|---|---|"]
    lines += [f"| {m} | `{p}` |" for p, m in routes]
    lines += ["", f"## MCP tools — {len(tools)} `room_*` tools "
              "(reference: 76 `quoroom_*`)", "",
              "| Tool | Description |", "|---|---|"]
    lines += [f"| `{n}` | {s['description']} |" for n, (s, _) in tools]
    lines += ["", "## WebSocket", "",
              "`GET /ws?token=<token>` — subscribe/unsubscribe protocol "
              "`{type: 'subscribe', channel}`;",
              "channels: `room:<id>`, `run:<id>`, `tasks`, `runs`, `rooms`, "
              "`clerk`; events `{type, channel, data, timestamp}`.",
              "", "## CLI", "",
              "`python -m room_amd.cli serve [port] | mcp | status | "
              "uninstall`", ""]
    out = Path(__file__).resolve().parent.parent / "docs" / "API.md"
    out.write_text("\n".join(lines))
    print(f"wrote {out}: {len(routes)} routes, {len(tools)} tools")


if __name__ == "__main__":
    main()
