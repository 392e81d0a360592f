"""Auxiliary subsystems: metrics, TTL, keys, liveness endpoint, bootstrap
server, rate limiter, stakes, auto-update watch, supervisor scripts.

Mirrors SURVEY.md §2.1 components #10 (rate limit/blacklist), #12 (MLflow
observability), #13 (bootstrap server), #14 (dummy miner / wallet
generation / auto-update), #15 (process supervisors).
"""

import json
import os
import subprocess
import time

import pytest
import torch

from distributedtraining_amd.registry import RateLimiter, Registry
from distributedtraining_amd.utils import keys as keymod
from distributedtraining_amd.utils.auto_update import read_version, watch
from distributedtraining_amd.utils.bootstrap_server import (BootstrapServer,
                                                            stress_test)
from distributedtraining_amd.utils.liveness import (MetricsEndpoint,
                                                    dummy_miner_post)
from distributedtraining_amd.utils.metrics import MetricsRun, system_metrics
from distributedtraining_amd.utils.ttl import TTLTimeout, run_with_ttl


# ---------------------------------------------------------------- rate limit
def test_rate_limiter_window_and_blacklist():
    rl = RateLimiter(max_requests=3, window_s=10.0, blacklist_after=2)
    t = 1000.0
    assert all(rl.allow("a", t + i) for i in range(3))
    assert not rl.allow("a", t + 3)          # over limit -> violation 1
    assert rl.allow("a", t + 20)             # window slid
    # force a second violation -> blacklist
    for i in range(3):
        rl.allow("a", t + 21 + i)
    assert not rl.allow("a", t + 24)
    assert "a" in rl.blacklist
    assert not rl.allow("a", t + 1000)       # blacklisted forever
    assert rl.allow("b", t)                  # others unaffected


def test_registry_stake_validator_threshold(tmp_path):
    r = Registry()
    r.set_stake("v1", 10_000)
    r.set_stake("m1", 5)
    assert r.validator_hotkeys() == ["v1"]   # reference: stake >= 1024
    assert r.get_stake("m1") == 5


# ---------------------------------------------------------------- ttl
def _slow():
    time.sleep(30)
    return 1


def _fast(x):
    return x * 2


def _boom():
    raise ValueError("child failure")


def test_run_with_ttl():
    assert run_with_ttl(_fast, 10.0, 21) == 42
    t0 = time.time()
    with pytest.raises(TTLTimeout):
        run_with_ttl(_slow, 1.0)
    assert time.time() - t0 < 10
    # a child exception is re-raised in the parent (reference semantics:
    # chain_manager.run_in_subprocess propagates failures)
    with pytest.raises(ValueError, match="child failure"):
        run_with_ttl(_boom, 10.0)


# ---------------------------------------------------------------- keys
def test_keys_sign_verify_envelope():
    hot, sec = keymod.generate_keypair("m")
    env = keymod.signed_envelope(hot, sec, {"loss": 1.5})
    assert keymod.verify_envelope(env, sec)
    bad = dict(env)
    bad["payload"] = {"loss": 0.0}
    assert not keymod.verify_envelope(bad, sec)
    stale = dict(env)
    stale["nonce"] = "100.0:deadbeef"
    assert not keymod.verify_envelope(stale, sec)


def test_generate_keyfile(tmp_path):
    ks = keymod.generate_keyfile(str(tmp_path / "keys.json"), 3)
    assert len(ks) == 3 and len({k["hotkey"] for k in ks}) == 3


# ---------------------------------------------------------------- liveness
def test_metrics_endpoint_auth_and_anomaly():
    reg = Registry()
    hot, sec = keymod.generate_keypair("m0")
    hot2, sec2 = keymod.generate_keypair("m1")
    ep = MetricsEndpoint(reg, {hot: sec, hot2: sec2})
    ep.start()
    url = f"http://127.0.0.1:{ep.port}"
    try:
        assert dummy_miner_post(url, hot, sec, 1.25) == 200
        assert dummy_miner_post(url, hot2, sec2, 1.30) == 200
        # wrong secret -> 403
        assert dummy_miner_post(url, hot, sec2, 1.0) == 403
        # unknown hotkey -> 403
        assert dummy_miner_post(url, "nobody", sec, 1.0) == 403
        assert ep.accepted == 2 and ep.rejected == 2
        # accepted metrics feed the MAD anomaly tracker
        anom = reg.detect_metric_anomaly()
        assert set(anom) == {hot, hot2}
    finally:
        ep.stop()


def test_metrics_endpoint_rate_limit():
    reg = Registry()
    reg.rate_limiter = RateLimiter(max_requests=2, window_s=60.0)
    hot, sec = keymod.generate_keypair("m0")
    ep = MetricsEndpoint(reg, {hot: sec})
    codes = [ep.handle(keymod.signed_envelope(hot, sec, {"loss": 1.0}))[0]
             for _ in range(4)]
    assert codes == [200, 200, 429, 429]


# ---------------------------------------------------------------- bootstrap
def test_bootstrap_server_pool_and_stress():
    srv = BootstrapServer(pool_size=4, health_interval_s=1000)
    srv.start()
    try:
        url = f"http://127.0.0.1:{srv.port}"
        res = stress_test(url, n_requests=20, concurrency=4)
        assert res["ok"] == 20 and res["fail"] == 0
        assert 1 <= res["unique_addresses"] <= 4
        # kill one endpoint; health check respawns to pool size
        srv.pool._sockets[0].close()
        assert srv.pool.check_and_manage() == 4
    finally:
        srv.stop()


# ---------------------------------------------------------------- metrics
def test_metrics_run_jsonl(tmp_path):
    with MetricsRun("miner", "hk", log_dir=str(tmp_path),
                    system_every=1) as run:
        run.log(1, loss=2.5)
        run.log(2, loss=2.4, perplexity=11.0)
    lines = [json.loads(ln) for ln in
             open(tmp_path / "miner_hk.jsonl")]
    assert lines[0]["event"] == "params" and "version" in lines[0]
    assert lines[1]["step"] == 1 and lines[1]["loss"] == 2.5
    assert lines[2]["loss"] == 2.4
    sm = system_metrics()
    assert "rss_gb" in sm


# ---------------------------------------------------------------- auto-update
def test_auto_update_watch(tmp_path):
    vf = tmp_path / "version.py"
    vf.write_text("__version__ = '1.0.0'\n")
    assert read_version(str(vf)) == "1.0.0"
    changes = []
    import threading

    def bump():
        time.sleep(0.15)
        vf.write_text("__version__ = '1.0.1'\n")

    threading.Thread(target=bump).start()
    watch(str(vf), lambda a, b: changes.append((a, b)), interval_s=0.1,
          max_iters=6)
    assert changes == [("1.0.0", "1.0.1")]


# ---------------------------------------------------------------- supervisor
@pytest.mark.timeout(120)
def test_supervisor_restarts_and_gives_up(tmp_path):
    # a role that always crashes: bad flag value -> fast exits; with
    # MAX_RESTARTS=1 and MIN_UPTIME large, supervisor should give up after
    # 2 fast crashes and exit nonzero.
    env = dict(os.environ, DTA_MAX_RESTARTS="1", DTA_MIN_UPTIME_S="9999")
    r = subprocess.run(
        ["bash", "scripts/supervise.sh", "miner", "--definitely-bad-flag-x",
         "--tiny", "--steps", "not_an_int"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        env=env, capture_output=True, text=True, timeout=110)
    assert r.returncode != 0
    assert "giving up" in r.stdout
