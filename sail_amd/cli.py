"""Command-line entry points (ref: crates/sail-cli/src/runner.rs:19-26 —
`sail spark server|shell|mcp-server`, `sail flight server`).

    python -m sail_amd spark server [--port P]     Spark Connect server
    python -m sail_amd flight server [--port P]    Arrow Flight server
    python -m sail_amd mcp server                  MCP stdio server
    python -m sail_amd sql -e "SELECT 1"           run a statement
    python -m sail_amd shell                       interactive SQL shell
"""
from __future__ import annotations

import argparse
import sys


def main(argv=None):
    p = argparse.ArgumentParser(prog="sail_amd")
    sub = p.add_subparsers(dest="cmd")

    sp = sub.add_parser("spark")
    sp.add_argument("action", choices=["server"])
    sp.add_argument("--host", default="127.0.0.1")
    sp.add_argument("--port", type=int, default=15002)
    sp.add_argument("--device", default=None)

    fl = sub.add_parser("flight")
    fl.add_argument("action", choices=["server"])
    fl.add_argument("--host", default="127.0.0.1")
    fl.add_argument("--port", type=int, default=32010)
    fl.add_argument("--device", default=None)

    mc = sub.add_parser("mcp")
    mc.add_argument("action", choices=["server"])
    mc.add_argument("--device", default=None)

    sq = sub.add_parser("sql")
    sq.add_argument("-e", "--execute", required=True)
    sq.add_argument("--device", default=None)

    sh = sub.add_parser("shell")
    sh.add_argument("--device", default=None)

    args = p.parse_args(argv)
    if args.cmd == "spark":
        from .connect.server import SparkConnectServer

        srv = SparkConnectServer(host=args.host, port=args.port,
                                 device=args.device).start()
        print(f"Spark Connect server listening on {srv.address}", flush=True)
        import threading

        threading.Event().wait()  # serve until killed
    elif args.cmd == "flight":
        from .connect.flight_server import SailFlightServer

        srv = SailFlightServer(host=args.host, port=args.port, device=args.device)
        print(f"Flight server listening on {srv.address}", flush=True)
        srv.serve()
    elif args.cmd == "mcp":
        from .mcp.server import run_stdio_server

        run_stdio_server(device=args.device)
    elif args.cmd == "sql":
        from .engine.session import SessionContext

        SessionContext(device=args.device).sql(args.execute).show(100)
    elif args.cmd == "shell":
        _shell(args.device)
    else:
        p.print_help()
        return 1
    return 0


def _shell(device):
    from .engine.session import SessionContext

    s = SessionContext(device=device)
    print(f"sail-mi355x SQL shell (device={s.device}); end statements with ;")
    buf = []
    while True:
        try:
            line = input("... " if buf else "sql> ")
        except EOFError:
            break
        buf.append(line)
        if line.rstrip().endswith(";"):
            stmt = "\n".join(buf).rstrip().rstrip(";")
            buf = []
            if stmt.strip().lower() in ("exit", "quit"):
                break
            try:
                s.sql(stmt).show(50)
            except Exception as e:
                print(f"error: {e}")


if __name__ == "__main__":
    sys.exit(main())
