"""Validator liveness/metrics endpoint + dummy-miner load client.

The reference exposes an HTTP endpoint on the validator that miners POST
signed loss metrics to, guarded by a signature check, a timestamp-nonce
freshness rule and a rate limiter/blacklist (axon serving helpers +
rate_limiter, /root/reference/hivetrain/btt_connector.py:99-260,454-480),
and ships a fake miner that signs timestamps and POSTs random losses to it
(utils/dummy_miner.py:25-82) as a liveness/auth-path test.

This module is the chain-free equivalent: a stdlib ThreadingHTTPServer
endpoint verifying HMAC envelopes (utils/keys.py) against a keyfile,
feeding accepted metrics into the Registry's anomaly tracker, and a
``dummy_miner`` client that exercises it.
"""

from __future__ import annotations

import json
import logging
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Dict, Optional

from ..registry import Registry
from . import keys as keymod

log = logging.getLogger(__name__)


class MetricsEndpoint:
    """POST /metrics {hotkey, nonce, payload:{loss,...}, signature}."""

    def __init__(self, registry: Registry, secrets: Dict[str, str],
                 host: str = "127.0.0.1", port: int = 0,
                 max_age_s: float = 300.0):
        self.registry = registry
        self.secrets = secrets          # hotkey -> shared secret
        self.max_age_s = max_age_s
        self.accepted = 0
        self.rejected = 0
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):  # quiet
                pass

            def do_POST(self):
                if self.path != "/metrics":
                    self.send_response(404)
                    self.end_headers()
                    return
                n = int(self.headers.get("Content-Length", 0))
                try:
                    env = json.loads(self.rfile.read(n))
                except json.JSONDecodeError:
                    return self._reply(400, "bad json")
                code, msg = outer.handle(env)
                self._reply(code, msg)

            def _reply(self, code, msg):
                body = json.dumps({"status": msg}).encode()
                self.send_response(code)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self.server = ThreadingHTTPServer((host, port), Handler)
        self.port = self.server.server_address[1]
        self._thread: Optional[threading.Thread] = None

    def handle(self, env: dict) -> tuple:
        """Auth pipeline: rate limit -> known hotkey -> signature+freshness
        (reference order: blacklist/rate_limiter first, then signature,
        btt_connector.py:170-220,454-480)."""
        hotkey = env.get("hotkey", "")
        if not self.registry.rate_limiter.allow(hotkey):
            self.rejected += 1
            return 429, "rate limited"
        secret = self.secrets.get(hotkey)
        if secret is None:
            self.rejected += 1
            return 403, "unknown hotkey"
        if not keymod.verify_envelope(env, secret, self.max_age_s):
            self.rejected += 1
            return 403, "bad signature"
        loss = env["payload"].get("loss")
        if isinstance(loss, (int, float)):
            self.registry.report_metric(hotkey, float(loss))
        self.accepted += 1
        return 200, "ok"

    def start(self) -> None:
        self._thread = threading.Thread(target=self.server.serve_forever,
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()


def dummy_miner_post(url: str, hotkey: str, secret: str,
                     loss: float) -> int:
    """One signed metrics POST (the reference dummy miner's loop body,
    dummy_miner.py:54-68). Returns the HTTP status."""
    import urllib.request
    env = keymod.signed_envelope(hotkey, secret, {"loss": loss})
    req = urllib.request.Request(url + "/metrics",
                                 data=json.dumps(env).encode(),
                                 headers={"Content-Type":
                                          "application/json"})
    try:
        with urllib.request.urlopen(req, timeout=10) as r:
            return r.status
    except Exception as e:  # HTTPError has .code
        return getattr(e, "code", 0)
