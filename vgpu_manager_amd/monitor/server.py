"""Metrics HTTP server with optional TLS and scrape rate limiting
(reference pkg/metrics/server: TLS + limiter in front of promhttp).
"""
from __future__ import annotations

import logging
import ssl
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

from prometheus_client import REGISTRY, generate_latest
from prometheus_client.exposition import CONTENT_TYPE_LATEST

log = logging.getLogger("vgpu.monitor.server")


class RateLimiter:
    """Token bucket: `rate` scrapes/second, burst capacity `burst`."""

    def __init__(self, rate: float = 2.0, burst: int = 5):
        self.rate = rate
        self.burst = burst
        self._tokens = float(burst)
        self._last = time.monotonic()
        self._mu = threading.Lock()

    def allow(self) -> bool:
        with self._mu:
            now = time.monotonic()
            self._tokens = min(self.burst,
                               self._tokens + (now - self._last) *
                               self.rate)
            self._last = now
            if self._tokens >= 1.0:
                self._tokens -= 1.0
                return True
            return False


def serve_metrics(port: int, *, registry=REGISTRY,
                  certfile: Optional[str] = None,
                  keyfile: Optional[str] = None,
                  rate: float = 2.0, burst: int = 5,
                  bind: str = "0.0.0.0") -> ThreadingHTTPServer:
    limiter = RateLimiter(rate, burst)

    class Handler(BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path not in ("/metrics", "/healthz"):
                self.send_error(404)
                return
            if self.path == "/healthz":
                body = b"ok"
                ctype = "text/plain"
            else:
                if not limiter.allow():
                    self.send_error(429, "scrape rate limited")
                    return
                body = generate_latest(registry)
                ctype = CONTENT_TYPE_LATEST
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, fmt, *args):  # quiet access log
            log.debug("metrics: " + fmt, *args)

    srv = ThreadingHTTPServer((bind, port), Handler)
    if certfile:
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(certfile, keyfile)
        srv.socket = ctx.wrap_socket(srv.socket, server_side=True)
    t = threading.Thread(target=srv.serve_forever, daemon=True,
                         name="metrics-http")
    t.start()
    log.info("metrics on %s:%d (tls=%s, %.1f scrapes/s burst %d)",
             bind, port, bool(certfile), rate, burst)
    return srv
