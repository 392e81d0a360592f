import torch

from distributedtraining_amd.store import (DeltaCheckpoint, FileStore,
                                           tensor_sha256)


def _spec():
    return [("a", (4, 4), 16), ("b", (8,), 8)]


def test_delta_checkpoint_roundtrip(tmp_path):
    flat = torch.randn(24)
    ck = DeltaCheckpoint(flat, _spec(), base_hash="abc", step=7,
                         meta={"k": 1})
    p = str(tmp_path / "d.pt")
    ck.save(p)
    ck2 = DeltaCheckpoint.load(p)
    assert torch.equal(ck2.flat, flat)
    assert ck2.spec == _spec()
    assert ck2.base_hash == "abc" and ck2.step == 7 and ck2.meta == {"k": 1}


def test_shape_validation_and_nan():
    ck = DeltaCheckpoint(torch.randn(24), _spec(), "h")
    assert ck.validate_against(_spec())
    assert not ck.validate_against([("a", (4, 4), 16)])
    assert not ck.has_nan()
    bad = DeltaCheckpoint(torch.tensor([1.0, float("nan")]), [], "h")
    assert bad.has_nan()


def test_file_store_model_change_detection(tmp_path):
    s = FileStore(str(tmp_path), hotkey="m0")
    assert s.pull_model() is None
    assert not s.check_for_new_model()
    s.push_model({"flat_master": torch.ones(3)})
    assert s.check_for_new_model()
    sd = s.pull_model()
    assert torch.equal(sd["flat_master"], torch.ones(3))
    assert not s.check_for_new_model()  # seen
    s.push_model({"flat_master": torch.zeros(3)})
    assert s.check_for_new_model()


def test_file_store_delta_exchange(tmp_path):
    s = FileStore(str(tmp_path), hotkey="m0")
    ck = DeltaCheckpoint(torch.randn(24), _spec(), "h", step=3)
    addr = s.push_delta(ck)
    r = FileStore(str(tmp_path), hotkey="validator")
    got = r.receive_delta(addr)
    assert got is not None and torch.equal(got.flat, ck.flat)
    assert r.receive_delta(str(tmp_path / "nope")) is None


def test_tensor_sha256_stable():
    t = torch.arange(10, dtype=torch.float32)
    assert tensor_sha256(t) == tensor_sha256(t.clone())
    assert tensor_sha256(t) != tensor_sha256(t + 1)


def test_file_store_concurrent_push_pull(tmp_path):
    """Readers racing a writer must always see a COMPLETE checkpoint (old
    or new, never torn) — the atomic tmp-file + os.replace contract."""
    import threading

    import torch

    from distributedtraining_amd.store import FileStore

    store = FileStore(str(tmp_path))
    store.push_model({"format": "dta-base-v1", "v": 0,
                      "flat_master": torch.zeros(1000)})
    stop = threading.Event()
    errors = []

    def writer():
        v = 1
        while not stop.is_set():
            store.push_model({"format": "dta-base-v1", "v": v,
                              "flat_master": torch.full((1000,), float(v))})
            v += 1

    def reader():
        r = FileStore(str(tmp_path))
        try:
            while not stop.is_set():
                d = r.pull_model()
                assert d is not None and d["format"] == "dta-base-v1"
                # the payload must be internally consistent
                assert torch.all(d["flat_master"] == float(d["v"]))
        except Exception as e:  # pragma: no cover - failure path
            errors.append(e)

    tw = threading.Thread(target=writer)
    trs = [threading.Thread(target=reader) for _ in range(3)]
    tw.start()
    for t in trs:
        t.start()
    import time
    time.sleep(1.5)
    stop.set()
    tw.join()
    for t in trs:
        t.join()
    assert not errors


def test_pull_model_survives_corrupt_file(tmp_path):
    """A corrupt model file reads as absence (like the delta channel);
    a subsequent clean push is picked up again."""
    import torch

    from distributedtraining_amd.store import FileStore

    store = FileStore(str(tmp_path))
    with open(store.model_path, "wb") as f:
        f.write(b"not a torch file")
    assert store.pull_model() is None
    assert not store.check_for_new_model()    # corrupt hash recorded
    store.push_model({"format": "dta-base-v1",
                      "flat_master": torch.ones(4)})
    assert store.check_for_new_model()        # new content -> re-pull
    assert store.pull_model()["flat_master"].sum() == 4
