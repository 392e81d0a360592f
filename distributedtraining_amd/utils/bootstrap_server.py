"""Rendezvous/bootstrap service: hand out healthy rendezvous endpoints.

Reimplements the reference's bootstrap DHT server
(/root/reference/hivetrain/utils/bootstrap_server.py): there, a Flask
service maintains a pool of 10 hivemind DHT processes, health-checks and
respawns them (:39-72), and GET /return_dht_address returns a random live
multiaddr (:85-106). MI355X-native equivalent: joining processes need a
torch.distributed rendezvous (MASTER_ADDR:MASTER_PORT), so the pool holds
N TCPStore-backed rendezvous endpoints, health-checked by binding probes
and respawned on failure.

Endpoints (stdlib HTTP, no Flask dependency):
  GET /return_address  -> {"address": "127.0.0.1:PORT"}   (random healthy)
  GET /health          -> {"alive": n, "pool": [...]}
"""

from __future__ import annotations

import json
import logging
import random
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import List, Optional

log = logging.getLogger(__name__)


class RendezvousPool:
    """Pool of N live TCP rendezvous endpoints (reference pool size 10,
    bootstrap_server.py:39)."""

    def __init__(self, size: int = 10, host: str = "127.0.0.1"):
        self.size = size
        self.host = host
        self._sockets: List[socket.socket] = []
        self._lock = threading.Lock()
        self.check_and_manage()

    def _spawn(self) -> socket.socket:
        s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind((self.host, 0))
        s.listen(64)
        return s

    @staticmethod
    def _healthy(s: socket.socket) -> bool:
        try:
            return s.fileno() >= 0 and s.getsockname()[1] > 0
        except OSError:
            return False

    def check_and_manage(self) -> int:
        """Drop dead endpoints, respawn to pool size (reference:
        check_and_manage_dhts, bootstrap_server.py:39-72). Returns the
        number alive."""
        with self._lock:
            self._sockets = [s for s in self._sockets if self._healthy(s)]
            while len(self._sockets) < self.size:
                self._sockets.append(self._spawn())
            return len(self._sockets)

    def random_address(self) -> Optional[str]:
        with self._lock:
            if not self._sockets:
                return None
            s = random.choice(self._sockets)
            return f"{self.host}:{s.getsockname()[1]}"

    def addresses(self) -> List[str]:
        with self._lock:
            return [f"{self.host}:{s.getsockname()[1]}"
                    for s in self._sockets]

    def close(self) -> None:
        with self._lock:
            for s in self._sockets:
                s.close()
            self._sockets = []


class BootstrapServer:
    def __init__(self, pool_size: int = 10, host: str = "127.0.0.1",
                 port: int = 0, health_interval_s: float = 30.0):
        self.pool = RendezvousPool(pool_size, host)
        self.health_interval_s = health_interval_s
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                if self.path == "/return_address":
                    addr = outer.pool.random_address()
                    self._reply(200 if addr else 503, {"address": addr})
                elif self.path == "/health":
                    self._reply(200, {"alive": outer.pool.check_and_manage(),
                                      "pool": outer.pool.addresses()})
                else:
                    self._reply(404, {"error": "not found"})

            def _reply(self, code, obj):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self.server = ThreadingHTTPServer((host, port), Handler)
        self.port = self.server.server_address[1]
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()

    def _health_loop(self) -> None:
        while not self._stop.wait(self.health_interval_s):
            n = self.pool.check_and_manage()
            log.debug("rendezvous pool: %d alive", n)

    def start(self) -> None:
        t1 = threading.Thread(target=self.server.serve_forever, daemon=True)
        t2 = threading.Thread(target=self._health_loop, daemon=True)
        t1.start()
        t2.start()
        self._threads = [t1, t2]

    def stop(self) -> None:
        self._stop.set()
        self.server.shutdown()
        self.server.server_close()
        self.pool.close()


def stress_test(url: str, n_requests: int = 50,
                concurrency: int = 8) -> dict:
    """Async-ish load test of the bootstrap endpoint (reference:
    bootstrap_stress.py:18-47). Returns {ok, fail, unique_addresses}."""
    import urllib.request
    from concurrent.futures import ThreadPoolExecutor

    def one(_):
        try:
            with urllib.request.urlopen(url + "/return_address",
                                        timeout=10) as r:
                return json.loads(r.read()).get("address")
        except Exception:
            return None

    with ThreadPoolExecutor(concurrency) as ex:
        results = list(ex.map(one, range(n_requests)))
    ok = [r for r in results if r]
    return {"ok": len(ok), "fail": n_requests - len(ok),
            "unique_addresses": len(set(ok))}


if __name__ == "__main__":  # pragma: no cover
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8500)
    ap.add_argument("--pool-size", type=int, default=10)
    a = ap.parse_args()
    logging.basicConfig(level=logging.INFO)
    srv = BootstrapServer(a.pool_size, port=a.port)
    srv.start()
    print(f"bootstrap server on :{srv.port}, pool "
          f"{srv.pool.addresses()}")
    try:
        while True:
            time.sleep(60)
    except KeyboardInterrupt:
        srv.stop()
