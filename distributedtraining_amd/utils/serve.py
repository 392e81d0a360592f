"""Minimal HTTP inference endpoint over a deployed base model.

Deployment-side counterpart of the training roles: load a base model
(averaged_model.pt from the store, or random init), serve
POST /generate {"ids": [[...]], "max_new_tokens": N, "temperature": T,
"top_k": K} -> {"ids": [[...]]}, plus GET /health. Generation runs the
KV-cache decode path on GPU (models/generate.py); stdlib HTTP like the
other in-framework services (utils/liveness.py, utils/bootstrap_server.py).
"""

from __future__ import annotations

import json
import logging
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

import torch

from ..models import generate

log = logging.getLogger(__name__)


class InferenceServer:
    def __init__(self, model, device, host: str = "127.0.0.1",
                 port: int = 0, max_new_tokens_cap: int = 256):
        self.model = model.eval()
        self.device = device
        self.cap = max_new_tokens_cap
        self._lock = threading.Lock()   # one generation at a time
        self.served = 0
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                if self.path == "/health":
                    self._reply(200, {"status": "ok",
                                      "served": outer.served})
                else:
                    self._reply(404, {"error": "not found"})

            def do_POST(self):
                if self.path != "/generate":
                    return self._reply(404, {"error": "not found"})
                n = int(self.headers.get("Content-Length", 0))
                try:
                    req = json.loads(self.rfile.read(n))
                    out = outer.handle(req)
                except (json.JSONDecodeError, KeyError, ValueError,
                        TypeError) as e:
                    return self._reply(400, {"error": str(e)})
                self._reply(200, out)

            def _reply(self, code, obj):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self.server = ThreadingHTTPServer((host, port), Handler)
        self.port = self.server.server_address[1]

    def handle(self, req: dict) -> dict:
        ids = torch.tensor(req["ids"], dtype=torch.long,
                           device=self.device)
        if ids.dim() != 2 or ids.numel() == 0:
            raise ValueError("ids must be a non-empty [batch, seq]")
        vocab = self.model.cfg.vocab_size
        if int(ids.min()) < 0 or int(ids.max()) >= vocab:
            raise ValueError("token id out of range")
        mnt = min(int(req.get("max_new_tokens", 32)), self.cap)
        npos = getattr(self.model.cfg, "n_positions", None)
        if npos is not None and ids.shape[1] + mnt > npos:
            raise ValueError(
                f"prompt ({ids.shape[1]}) + max_new_tokens ({mnt}) exceeds "
                f"the model context length {npos}")
        temperature = float(req.get("temperature", 0.0))
        top_k = int(req.get("top_k", 0))
        top_p = float(req.get("top_p", 0.0))
        eos = req.get("eos_token_id")
        with self._lock, torch.no_grad():
            out = generate(self.model, ids, mnt, temperature=temperature,
                           top_k=top_k, top_p=top_p,
                           eos_token_id=None if eos is None else int(eos),
                           use_cache=hasattr(self.model, "prefill"))
        self.served += 1
        return {"ids": out.tolist()}

    def start(self) -> None:
        threading.Thread(target=self.server.serve_forever,
                         daemon=True).start()

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()


def post_generate(url: str, ids, max_new_tokens: int = 32,
                  **kw) -> Optional[list]:
    """Client helper: POST /generate; returns completed ids or None."""
    import urllib.request
    req = urllib.request.Request(
        url + "/generate",
        data=json.dumps({"ids": ids, "max_new_tokens": max_new_tokens,
                         **kw}).encode(),
        headers={"Content-Type": "application/json"})
    try:
        with urllib.request.urlopen(req, timeout=120) as r:
            return json.loads(r.read())["ids"]
    except Exception as e:
        log.warning("generate request failed: %s", e)
        return None
