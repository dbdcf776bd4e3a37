"""Local HTTP t.me mock server + fetcher.

Puts a real socket under the tandem validator (the reference fetches
https://t.me/<u> over uTLS, telegramhelper/channelvalidator.go:64-103):
a threaded stdlib HTTP server fronts MockTMe, and `http_fetcher` adapts
urllib to the validator's fetcher contract. Blocked-mode toggling flips
the underlying MockTMe so the IP-block state machine can be exercised
end-to-end over HTTP.
"""
from __future__ import annotations

import threading
import urllib.error
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Tuple

from .tme import MockTMe


class TMeServer:
    def __init__(self, tme: MockTMe, host: str = "127.0.0.1",
                 port: int = 0):
        self.tme = tme
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def do_GET(self):
                username = self.path.lstrip("/").split("?")[0]
                status, body = outer.tme(username)
                self.send_response(status)
                self.send_header("Content-Type",
                                 "text/html; charset=utf-8")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *args):  # quiet
                pass

        self.httpd = ThreadingHTTPServer((host, port), Handler)
        self.port = self.httpd.server_address[1]
        self.base_url = f"http://{host}:{self.port}"
        self._thread = threading.Thread(target=self.httpd.serve_forever,
                                        daemon=True)

    def start(self) -> "TMeServer":
        self._thread.start()
        return self

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()


def http_fetcher(base_url: str, timeout: float = 5.0):
    """fetcher(username) -> (status, body) over a real HTTP request."""

    def fetch(username: str) -> Tuple[int, bytes]:
        url = f"{base_url}/{username}"
        try:
            with urllib.request.urlopen(url, timeout=timeout) as resp:
                return resp.status, resp.read()
        except urllib.error.HTTPError as e:
            return e.code, e.read()
        except OSError as e:
            raise TimeoutError(str(e))

    return fetch
