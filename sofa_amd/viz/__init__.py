"""sofa viz — serve the sofaboard dashboard from the logdir.

Parity: reference bin/sofa_viz.py:10-18.
"""

from __future__ import annotations

import http.server
import os
import socketserver

from .. import printing as p
from ..config import SofaConfig


def sofa_viz(cfg: SofaConfig) -> None:
    logdir = os.path.abspath(cfg.logdir)
    # the logdir carries kallsyms and full traces: bind loopback unless the
    # user deliberately exposes it with --viz_host (round-1 ADVICE)
    host = getattr(cfg, "viz_host", "127.0.0.1") or "127.0.0.1"
    shown = "localhost" if host in ("127.0.0.1", "localhost") else host
    p.print_hint(
        f"serving sofaboard at http://{shown}:{cfg.viz_port}/ from {logdir} "
        "(Ctrl-C to stop)"
    )
    handler = lambda *a, **kw: http.server.SimpleHTTPRequestHandler(  # noqa: E731
        *a, directory=logdir, **kw
    )
    with socketserver.TCPServer((host, cfg.viz_port), handler) as httpd:
        try:
            httpd.serve_forever()
        except KeyboardInterrupt:
            pass
