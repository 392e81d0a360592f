import math

import pytest

from distributedtraining_amd.registry import (FileRegistry, Registry,
                                              mad_outlier_scores)


def test_address_roundtrip():
    r = Registry()
    r.store_address("hk1", "/tmp/a")
    r.store_address("hk2", "/tmp/b")
    assert r.retrieve_address("hk1") == "/tmp/a"
    assert r.retrieve_address("missing") is None
    assert set(r.hotkeys) == {"hk1", "hk2"}
    r.deregister("hk1")
    assert r.retrieve_address("hk1") is None


def test_score_ema_alpha():
    # reference semantics: s' = a*new + (1-a)*old, a=0.333 (btt_connector.py:317)
    r = Registry(ema_alpha=0.333333)
    r.store_address("hk", "x")
    out1 = r.set_weights({"hk": 1.0})
    assert out1["hk"] == pytest.approx(0.333333)
    out2 = r.set_weights({"hk": 1.0})
    assert out2["hk"] == pytest.approx(0.333333 + 0.666667 * 0.333333, rel=1e-4)


def test_should_set_weights_gating():
    r = Registry(epoch_length=10**9)
    assert r.should_set_weights()  # never set before
    r.set_weights({})
    assert not r.should_set_weights()


def test_mad_outliers():
    metrics = {f"m{i}": [1.0 + 0.01 * i] for i in range(8)}
    metrics["bad"] = [100.0]
    scores = mad_outlier_scores(metrics, threshold=2.0)
    assert scores["bad"] == 0
    assert all(scores[f"m{i}"] == 1 for i in range(8))


def test_file_registry_cross_instance(tmp_path):
    a = FileRegistry(str(tmp_path))
    a.store_address("hk1", "addr1")
    b = FileRegistry(str(tmp_path))
    assert b.retrieve_address("hk1") == "addr1"
    b.store_address("hk2", "addr2")
    assert set(a.hotkeys) == {"hk1", "hk2"}


def test_file_registry_score_update_not_stale(tmp_path):
    """Concurrent-writer score merge: a reload must pick up NEWER persisted
    scores (round-1 advisory: setdefault pinned the first value forever)."""
    a = FileRegistry(str(tmp_path), ema_alpha=1.0)
    b = FileRegistry(str(tmp_path), ema_alpha=1.0)
    a.set_weights({"hk": 1.0})
    assert b.get_weights() == {}          # b hasn't reloaded yet
    b.store_address("hk", "addr")          # any reload path
    assert b.get_weights().get("hk") == 1.0
    a.set_weights({"hk": 0.25})
    b.retrieve_address("hk")               # reload again
    assert b.get_weights().get("hk") == 0.25   # newer value wins


def test_file_registry_set_weights_folds_latest(tmp_path):
    """set_weights EMA must fold into the latest persisted scores, not this
    process's stale copy."""
    a = FileRegistry(str(tmp_path), ema_alpha=0.5)
    b = FileRegistry(str(tmp_path), ema_alpha=0.5)
    a.set_weights({"hk": 1.0})             # file: 0.5
    out = b.set_weights({"hk": 1.0})       # folds into 0.5 -> 0.75
    assert abs(out["hk"] - 0.75) < 1e-9


def test_file_registry_concurrent_writers(tmp_path):
    """Many threads over separate FileRegistry instances hammering the same
    JSON file: no write may be lost, no exception may escape, and the file
    must stay parseable (atomic replace + merge-on-load under the lock)."""
    import threading

    N_WRITERS, N_KEYS = 8, 20
    errors = []

    def writer(wid):
        try:
            r = FileRegistry(str(tmp_path))
            for i in range(N_KEYS):
                r.store_address(f"hk{wid}_{i}", f"addr{wid}_{i}")
                r.set_stake(f"hk{wid}_{i}", float(wid))
        except Exception as e:  # pragma: no cover - failure path
            errors.append(e)

    threads = [threading.Thread(target=writer, args=(w,))
               for w in range(N_WRITERS)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors
    final = FileRegistry(str(tmp_path))
    hks = set(final.hotkeys)
    expect = {f"hk{w}_{i}" for w in range(N_WRITERS) for i in range(N_KEYS)}
    assert hks >= expect
    for w in range(N_WRITERS):
        assert final.retrieve_address(f"hk{w}_0") == f"addr{w}_0"


def test_file_registry_survives_corrupt_json(tmp_path):
    """A half-written or garbage registry.json must not crash readers or
    writers; the next clean save repairs the file."""
    a = FileRegistry(str(tmp_path))
    a.store_address("hk1", "addr1")
    with open(a.path, "w") as f:
        f.write('{"addresses": {"hk1"')     # truncated JSON
    b = FileRegistry(str(tmp_path))          # load of garbage -> tolerated
    assert b.retrieve_address("hk_missing") is None
    b.store_address("hk2", "addr2")          # write-through repairs
    c = FileRegistry(str(tmp_path))
    assert c.retrieve_address("hk2") == "addr2"
    # a still holds its in-memory copy and re-merges cleanly
    assert a.retrieve_address("hk1") == "addr1"
