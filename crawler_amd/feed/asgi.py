"""ASGI t.me mock — the tandem validator behind a production server.

tme_server.py puts the validator behind a stdlib threaded HTTP server;
this module adds the production-shaped hop: an ASGI app over the same
MockTMe, served by uvicorn in a background thread. The reference's
validator fetches https://t.me/<u> through a real TLS client stack
(telegramhelper/channelvalidator.go:64-103); running our validator
against uvicorn exercises the same kind of full server/event-loop path
without needing egress.
"""
from __future__ import annotations

import threading
from typing import Optional

from .tme import MockTMe


def tme_asgi_app(tme: MockTMe):
    """Minimal ASGI application serving GET /<username> from MockTMe."""

    async def app(scope, receive, send):
        if scope["type"] != "http":
            return
        username = scope["path"].lstrip("/").split("?")[0]
        status, body = tme(username)
        await send({
            "type": "http.response.start",
            "status": status,
            "headers": [
                (b"content-type", b"text/html; charset=utf-8"),
                (b"content-length", str(len(body)).encode()),
            ],
        })
        await send({"type": "http.response.body", "body": body})

    return app


class UvicornTMeServer:
    """MockTMe behind uvicorn on 127.0.0.1:<ephemeral>, run in a
    daemon thread. Raises RuntimeError if uvicorn is unavailable."""

    def __init__(self, tme: MockTMe, host: str = "127.0.0.1",
                 port: int = 0):
        try:
            import uvicorn
        except ImportError as e:  # pragma: no cover - installed here
            raise RuntimeError(f"uvicorn not available: {e}")
        config = uvicorn.Config(tme_asgi_app(tme), host=host, port=port,
                                log_level="error", lifespan="off")
        self.server = uvicorn.Server(config)
        self._thread = threading.Thread(target=self.server.run,
                                        daemon=True)
        self.host = host
        self.base_url: Optional[str] = None

    def start(self, timeout: float = 10.0) -> "UvicornTMeServer":
        self._thread.start()
        import time
        deadline = time.monotonic() + timeout
        while not self.server.started:
            if time.monotonic() > deadline:
                raise TimeoutError("uvicorn did not start")
            time.sleep(0.01)
        # the bound socket knows the ephemeral port
        sock = self.server.servers[0].sockets[0]
        self.base_url = f"http://{self.host}:{sock.getsockname()[1]}"
        return self

    def stop(self, timeout: float = 10.0):
        self.server.should_exit = True
        self._thread.join(timeout=timeout)
