"""CLI entry points: `python -m audiomuse_amd <command>`.

Deployment shape (reference: deployment/docker-entrypoint.sh
SERVICE_TYPE=flask|worker; supervisord runs gunicorn + queue workers):
- web    : Flask app (dev server here; any WSGI server in production)
- worker : queue worker loop (one process per GPU rank)
- analyze: enqueue a full analysis run against a configured server
- bench  : shortcut to the flagship benchmark
"""

from __future__ import annotations

import argparse
import sys


def main() -> None:
    ap = argparse.ArgumentParser(prog="audiomuse_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    web = sub.add_parser("web")
    web.add_argument("--host", default="0.0.0.0")
    web.add_argument("--port", type=int, default=8000)
    web.add_argument("--db", default=None)
    web.add_argument("--no-auth", action="store_true")
    web.add_argument("--procs", type=int, default=1,
                    help="SO_REUSEPORT web processes sharing the port")
    web.add_argument("--reuse-port", action="store_true",
                    help="bind with SO_REUSEPORT (set by --procs children)")

    wk = sub.add_parser("worker")
    wk.add_argument("--db", default=None)
    wk.add_argument("--queues", default="high,default")
    wk.add_argument("--max-jobs", type=int, default=None)
    wk.add_argument("--idle-timeout", type=float, default=None,
                    help="exit after this many idle seconds (default: run forever)")

    an = sub.add_parser("analyze")
    an.add_argument("--db", default=None)
    an.add_argument("--server-type", default="synthetic")
    an.add_argument("--server-id", default="default")

    sa = sub.add_parser("standalone")
    sa.add_argument("--host", default="0.0.0.0")
    sa.add_argument("--port", type=int, default=8000)
    sa.add_argument("--db", default=None)
    sa.add_argument("--workers", type=int, default=None,
                    help="worker processes (default: one per GPU)")
    sa.add_argument("--no-auth", action="store_true")

    sub.add_parser("bench")

    args, rest = ap.parse_known_args()

    if args.cmd == "web":
        from audiomuse_amd.web.serve import serve, serve_procs

        if args.procs > 1:
            serve_procs(args.db, args.host, args.port, args.procs,
                        auth_disabled=args.no_auth)
        else:
            serve(args.db, args.host, args.port,
                  auth_disabled=args.no_auth, reuse_port=args.reuse_port)
    elif args.cmd == "worker":
        from audiomuse_amd.taskqueue.worker import Worker

        Worker(db_url=args.db, queues=tuple(args.queues.split(",")),
               max_jobs=args.max_jobs).run_forever(
            idle_timeout=args.idle_timeout)
    elif args.cmd == "analyze":
        from audiomuse_amd.db import get_db
        from audiomuse_amd.taskqueue import enqueue

        conn = get_db(args.db)
        tid = enqueue(conn, "run_analysis",
                      {"server_type": args.server_type,
                       "server_id": args.server_id}, queue="high")
        print(f"enqueued analysis task {tid}")
    elif args.cmd == "standalone":
        from audiomuse_amd.standalone import run_standalone

        run_standalone(db_url=args.db, host=args.host, port=args.port,
                       workers=args.workers, no_auth=args.no_auth)
    elif args.cmd == "bench":
        sys.argv = [sys.argv[0]] + rest
        import runpy

        runpy.run_path("bench.py", run_name="__main__")


if __name__ == "__main__":
    main()
