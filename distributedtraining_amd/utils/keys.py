"""Hotkey identities: generation, signing, verification.

The reference identifies miners by Bittensor wallet hotkeys and signs
liveness messages with them (dummy_miner signs nonce timestamps,
/root/reference/hivetrain/utils/dummy_miner.py:38-52; generate_wallets
mass-creates test wallets, utils/generate_wallets.py:9-41). Chain-free
equivalent: an HMAC-SHA256 keypair per member — the registry stores the
public id, messages carry (hotkey, nonce, signature).
"""

from __future__ import annotations

import hashlib
import hmac
import json
import os
import secrets
import time
from typing import Dict, List, Tuple


def generate_keypair(name: str = "") -> Tuple[str, str]:
    """Returns (hotkey, secret). hotkey = sha256(secret)[:40] — the public
    identity registered with the registry."""
    secret = secrets.token_hex(32)
    hot = hashlib.sha256(bytes.fromhex(secret)).hexdigest()[:40]
    return (f"{name}-{hot}" if name else hot), secret


def sign(secret: str, message: str) -> str:
    return hmac.new(bytes.fromhex(secret), message.encode(),
                    hashlib.sha256).hexdigest()


def verify(secret: str, message: str, signature: str) -> bool:
    return hmac.compare_digest(sign(secret, message), signature)


def signed_envelope(hotkey: str, secret: str, payload: dict) -> dict:
    """Nonce-timestamped signed message (the dummy miner's wire format:
    message + timestamp nonce + signature, dummy_miner.py:54-61)."""
    nonce = f"{time.time():.6f}:{secrets.token_hex(8)}"
    body = json.dumps(payload, sort_keys=True)
    return {"hotkey": hotkey, "nonce": nonce, "payload": payload,
            "signature": sign(secret, nonce + body)}


def verify_envelope(env: dict, secret: str, max_age_s: float = 300.0) -> bool:
    try:
        ts = float(env["nonce"].split(":")[0])
        if abs(time.time() - ts) > max_age_s:
            return False
        body = json.dumps(env["payload"], sort_keys=True)
        return verify(secret, env["nonce"] + body, env["signature"])
    except (KeyError, ValueError):
        return False


def generate_keyfile(path: str, n: int, prefix: str = "miner") -> List[Dict]:
    """Mass-create test identities (reference: generate_wallets.py) into a
    JSON keyfile [{name, hotkey, secret}, ...]."""
    out = []
    for i in range(n):
        hot, sec = generate_keypair(f"{prefix}{i}")
        out.append({"name": f"{prefix}{i}", "hotkey": hot, "secret": sec})
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    return out
