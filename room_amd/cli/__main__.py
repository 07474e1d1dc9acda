"""CLI entry: room-amd mcp | serve [port] | status | update | uninstall
(reference: src/cli/index.ts — quoroom mcp|serve|update|uninstall)."""
from __future__ import annotations

import argparse
import os
import shutil
import sys

VERSION = "0.1.0"


def _version_hint() -> None:
    """Print a one-line hint when a newer staged version exists (reference
    cli/version-hint.ts — non-blocking, fail-silent, suppressed for mcp
    whose stdout is the protocol stream)."""
    try:
        from ..core.update_checker import UpdateChecker, compare_versions
        from ..server.auth import data_dir
        uc = UpdateChecker(VERSION, data_dir=data_dir())
        staged = uc.staged_version()
        if staged and (compare_versions(staged, VERSION) or 0) > 0:
            print(f"[room-amd] v{staged} is staged — restart the server or "
                  f"run `room-amd update` to apply (current v{VERSION})",
                  file=sys.stderr)
    except Exception:
        pass


def main(argv: list[str] | None = None) -> int:
    p = argparse.ArgumentParser(prog="room-amd",
                                description="MI355X-native agent swarm runtime")
    sub = p.add_subparsers(dest="cmd")

    sp = sub.add_parser("serve", help="start the HTTP/WS server + runtime loops")
    sp.add_argument("port", nargs="?", type=int, default=3700)
    sp.add_argument("--host", default="127.0.0.1")
    sp.add_argument("--db", default=None)

    sub.add_parser("mcp", help="run the MCP stdio server")

    st = sub.add_parser("status", help="query the running server's status")
    st.add_argument("--port", type=int, default=None)

    sub.add_parser("update", help="check for (and stage) a newer release")

    sub.add_parser("uninstall", help="remove ~/.roomamd data (asks first)")

    args = p.parse_args(argv)

    if args.cmd in ("serve", "status", "update", None):
        _version_hint()

    if args.cmd == "serve":
        from ..server.bootstrap import serve
        serve(port=args.port, host=args.host, db_path=args.db)
        return 0

    if args.cmd == "mcp":
        from ..mcp.server import main as mcp_main
        mcp_main()
        return 0

    if args.cmd == "status":
        import json
        import urllib.request

        from ..server.auth import data_dir
        try:
            port = args.port or int((data_dir() / "api.port").read_text().strip())
            token = (data_dir() / "api.token").read_text().strip()
            req = urllib.request.Request(
                f"http://127.0.0.1:{port}/api/status",
                headers={"Authorization": f"Bearer {token}"})
            print(json.dumps(json.load(urllib.request.urlopen(req, timeout=5)),
                             indent=2))
            return 0
        except Exception as e:
            print(f"server not reachable: {e}", file=sys.stderr)
            return 1

    if args.cmd == "update":
        # reference: quoroom update (cli/update.ts) — poll the release feed,
        # report, and stage a user-space update when one is available
        from ..core.update_checker import UpdateChecker
        from ..server.auth import data_dir
        uc = UpdateChecker(VERSION, data_dir=data_dir())
        uc.boot_health_check()
        st = uc.check()
        if st["state"] in ("offline", "error"):
            print(f"update check failed ({st['state']}): {st['error']}",
                  file=sys.stderr)
            return 1
        if not st["updateAvailable"]:
            print(f"up to date (v{st['currentVersion']})")
            return 0
        print(f"update available: v{st['latestVersion']} ({st['releaseUrl']})")
        out = uc.stage_update()
        print(f"staged: {out}")
        return 0

    if args.cmd == "uninstall":
        from ..server.auth import data_dir
        d = data_dir()
        answer = input(f"Remove {d} (y/N)? ").strip().lower()
        if answer == "y":
            shutil.rmtree(d, ignore_errors=True)
            print("removed")
        return 0

    p.print_help()
    return 1


if __name__ == "__main__":
    sys.exit(main())
