"""Debug/introspection HTTP server — the pprof-on-:6060 analog.

The reference starts a Go pprof server unconditionally on :6060
(main.go: pprof import + http.ListenAndServe). The equivalent here is a
stdlib threaded HTTP server exposing:

- /healthz          -> 200 "ok"
- /metrics          -> MetricsRegistry.snapshot() JSON
- /debug/stacks     -> all Python thread stacks (goroutine-dump analog)
- /debug/vars       -> process RSS / fds / uptime (expvar analog)

Enabled with --debug-port (0 = off, the default, so tests and parallel
ranks never fight over a port; pass 6060 for reference-identical
behavior).
"""
from __future__ import annotations

import json
import os
import sys
import threading
import time
import traceback
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional


def _stacks() -> str:
    lines = []
    frames = sys._current_frames()
    for t in threading.enumerate():
        lines.append(f"--- thread {t.name} (daemon={t.daemon}) ---")
        frame = frames.get(t.ident)
        if frame is not None:
            lines.extend(
                l.rstrip() for l in traceback.format_stack(frame)
            )
    return "\n".join(lines) + "\n"


def _vars(t0: float) -> dict:
    out = {"pid": os.getpid(), "uptime_s": round(time.monotonic() - t0, 1)}
    try:
        with open("/proc/self/status") as f:
            for line in f:
                if line.startswith(("VmRSS", "Threads")):
                    k, v = line.split(":", 1)
                    out[k.lower()] = v.strip()
        out["open_fds"] = len(os.listdir("/proc/self/fd"))
    except OSError:
        pass
    return out


class DebugServer:
    def __init__(self, port: int, metrics=None, host: str = "127.0.0.1"):
        self.port = port
        self.metrics = metrics
        self.t0 = time.monotonic()
        srv_self = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):  # quiet
                pass

            def _send(self, code: int, body: bytes,
                      ctype: str = "text/plain"):
                self.send_response(code)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                path = self.path.split("?")[0]
                if path == "/healthz":
                    self._send(200, b"ok\n")
                elif path == "/metrics":
                    snap = (srv_self.metrics.snapshot()
                            if srv_self.metrics is not None else {})
                    self._send(200,
                               (json.dumps(snap) + "\n").encode(),
                               "application/json")
                elif path == "/debug/stacks":
                    self._send(200, _stacks().encode())
                elif path == "/debug/vars":
                    self._send(200,
                               (json.dumps(_vars(srv_self.t0)) + "\n")
                               .encode(),
                               "application/json")
                else:
                    self._send(404, b"not found\n")

        self._httpd = ThreadingHTTPServer((host, port), Handler)
        self.port = self._httpd.server_address[1]  # resolve port 0
        self._thread: Optional[threading.Thread] = None

    def start(self):
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, daemon=True,
            name="debug-server",
        )
        self._thread.start()
        return self

    def stop(self):
        self._httpd.shutdown()
        self._httpd.server_close()
        if self._thread:
            self._thread.join(timeout=5)


def maybe_start(port: int, metrics=None) -> Optional[DebugServer]:
    """CLI hook: 0/None disables (reference always binds :6060; here it
    is opt-in so multi-rank launches don't collide)."""
    if not port:
        return None
    return DebugServer(port, metrics=metrics).start()
