"""Web serving entry: one threaded werkzeug server per PROCESS, with
optional SO_REUSEPORT fan-out so several processes share one port and
the kernel load-balances accepted connections across them.

Why: one web process tops out at ~160-190 qps on a GPU host — its GPU
index queries serialize on one stream and its Python runs on one GIL
(profiles/r2_http_load*.log). The reference scales by adding worker
processes; the web tier scales the same way here, with no reverse
proxy needed on Linux (SO_REUSEPORT).
"""

from __future__ import annotations

import socket
import sys
from typing import Optional


def _reuseport_socket(host: str, port: int) -> socket.socket:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    if not hasattr(socket, "SO_REUSEPORT"):   # pragma: no cover — Linux has it
        raise RuntimeError("SO_REUSEPORT unavailable on this platform")
    s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEPORT, 1)
    s.bind((host, port))
    s.listen(128)
    return s


def serve(db_url: Optional[str], host: str, port: int,
          auth_disabled: bool = False, reuse_port: bool = False) -> None:
    """Serve forever in THIS process."""
    from werkzeug.serving import make_server

    from audiomuse_amd.web.app import create_app

    app = create_app(db_url, auth_disabled=auth_disabled)
    if reuse_port:
        sock = _reuseport_socket(host, port)
        srv = make_server(host, port, app, threaded=True,
                          fd=sock.fileno())
    else:
        srv = make_server(host, port, app, threaded=True)
    srv.serve_forever()


def serve_procs(db_url: Optional[str], host: str, port: int,
                procs: int, auth_disabled: bool = False):
    """Fork `procs` SO_REUSEPORT servers; block until any exits.
    Returns the list of child PIDs (for tests/supervisors)."""
    if procs <= 1:
        serve(db_url, host, port, auth_disabled=auth_disabled)
        return []
    import subprocess

    cmd = [sys.executable, "-m", "audiomuse_amd", "web",
           "--host", host, "--port", str(port), "--reuse-port"]
    if db_url:
        cmd += ["--db", db_url]
    if auth_disabled:
        cmd += ["--no-auth"]
    children = [subprocess.Popen(cmd) for _ in range(procs)]
    try:
        for c in children:
            c.wait()
    except KeyboardInterrupt:   # pragma: no cover — interactive stop
        for c in children:
            c.terminate()
    return children
