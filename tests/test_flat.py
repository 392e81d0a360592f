import torch
import torch.nn as nn

from distributedtraining_amd.config import ModelConfig
from distributedtraining_amd.models import build_model
from distributedtraining_amd.parallel.flat import FlatParams, FusedAdamW


def _tiny():
    torch.manual_seed(0)
    return build_model(ModelConfig.gpt2_tiny())


def test_flat_views_share_storage():
    m = _tiny()
    fp = FlatParams(m)
    for p in m.parameters():
        assert p.data_ptr() >= fp.work.data_ptr()
        assert p.grad is not None
    # mutation through flat is visible through the param view
    fp.work.zero_()
    assert float(next(m.parameters()).detach().abs().sum()) == 0.0


def test_tied_params_deduped():
    m = _tiny()
    n_unique = len({id(p) for p in m.parameters()})
    fp = FlatParams(m)
    assert len(fp.spec) == n_unique
    assert fp.offsets[-1].item() == fp.numel


def test_grad_accumulates_into_flat():
    m = _tiny()
    fp = FlatParams(m)
    ids = torch.randint(0, 512, (2, 16))
    out = m(input_ids=ids, labels=ids)
    out.loss.backward()
    assert float(fp.grad.abs().sum()) > 0
    for p in m.parameters():
        assert p.grad.data_ptr() >= fp.grad.data_ptr()


def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    ref = nn.Linear(16, 16)
    dup = nn.Linear(16, 16)
    dup.load_state_dict(ref.state_dict())
    fp = FlatParams(dup)
    opt_ref = torch.optim.AdamW(ref.parameters(), lr=1e-2, weight_decay=0.1)
    opt_fused = FusedAdamW(fp, lr=1e-2, weight_decay=0.1)
    x = torch.randn(8, 16)
    for _ in range(5):
        opt_ref.zero_grad()
        ref(x).pow(2).mean().backward()
        opt_ref.step()
        opt_fused.zero_grad()
        dup(x).pow(2).mean().backward()
        opt_fused.step()
    for a, b in zip(ref.parameters(), dup.parameters()):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_snapshot_delta_roundtrip():
    m = _tiny()
    fp = FlatParams(m)
    base = fp.snapshot()
    fp.master += 0.5
    ck = fp.make_delta(base)
    torch.testing.assert_close(ck.flat, torch.full_like(ck.flat, 0.5))
    fp.load_flat_master(base)
    torch.testing.assert_close(fp.master, base)


def test_all_gather_wire_dtype_single_rank():
    """bf16 wire-dtype exchange: halves gather bytes; single-rank path."""
    import torch
    from distributedtraining_amd.parallel.comm import CommPlane
    comm = CommPlane(device=torch.device("cpu"))
    flat = torch.randn(64, dtype=torch.float32)
    g32 = comm.all_gather_flat(flat)
    assert g32.dtype == torch.float32 and g32.shape == (1, 64)
    g16 = comm.all_gather_flat(flat, torch.bfloat16)
    assert g16.dtype == torch.bfloat16
    torch.testing.assert_close(g16[0].float(), flat, rtol=1e-2, atol=1e-2)
