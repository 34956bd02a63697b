"""Checkpoint spill/restore: training resumes bit-identically."""
import os

import pytest

pytestmark = pytest.mark.flaky_retry
import torch

from kubetorch_amd.models import Llama, llama_tiny
from kubetorch_amd.parallel import FlatDDP
from kubetorch_amd.utils import checkpoint as ckpt


def _setup(device="cpu"):
    torch.manual_seed(0)
    cfg = llama_tiny(n_layers=1, dim=128, intermediate=256, vocab_size=256,
                     n_heads=4, n_kv_heads=2)
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    try:
        with torch.device(device):
            model = Llama(cfg)
    finally:
        torch.set_default_dtype(prev)
    eng = FlatDDP(model, lr=1e-2, bucket_mb=1)
    x = torch.randint(0, cfg.vocab_size, (2, 32), device=device)
    y = torch.randint(0, cfg.vocab_size, (2, 32), device=device)
    return model, eng, x, y


def _roundtrip(device, tmp_path):
    model, eng, x, y = _setup(device)
    for _ in range(2):
        model.loss(x, y).backward()
        eng.step()
    path = str(tmp_path / "ck.pt")
    ckpt.save_engine_checkpoint(model, eng, path)
    # two more steps -> record result
    for _ in range(2):
        model.loss(x, y).backward()
        eng.step()
    after = torch.cat([b.flat_param.float().cpu().reshape(-1)
                       for b in eng.buckets])

    # fresh model, restore, repeat the same two steps
    model2, eng2, _, _ = _setup(device)
    ckpt.load_engine_checkpoint(model2, eng2, path)
    assert eng2.step_count == 2
    for _ in range(2):
        model2.loss(x, y).backward()
        eng2.step()
    after2 = torch.cat([b.flat_param.float().cpu().reshape(-1)
                        for b in eng2.buckets])
    torch.testing.assert_close(after, after2, rtol=0, atol=0)


def test_checkpoint_resume_cpu(tmp_path):
    _roundtrip("cpu", tmp_path)


@pytest.mark.gpu
def test_checkpoint_resume_gpu(tmp_path):
    """Exercises the pinned hipMemcpyAsync spill path."""
    _roundtrip("cuda", tmp_path)


def test_spill_to_store(tmp_path, monkeypatch):
    monkeypatch.setenv("KT_STORE_ROOT", str(tmp_path / "store"))
    monkeypatch.delenv("KT_STORE_URL", raising=False)
    import kubetorch_amd.data_store.commands as cmds

    monkeypatch.setattr(cmds, "LOCAL_STORE_ROOT", str(tmp_path / "store"))
    sd = {"w": torch.randn(8), "step": 3}
    ckpt.save_checkpoint(sd, str(tmp_path / "c.pt"), store_key="ckpts/c.pt")
    loaded = ckpt.load_checkpoint("ckpts/c.pt")
    torch.testing.assert_close(loaded["w"], sd["w"])
    assert loaded["step"] == 3


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_native_spill_roundtrip_gpu(tmp_path):
    """C++ pinned-ring spill engine: bit-exact roundtrip + bandwidth."""
    model, eng, x, y = _setup("cuda")
    model.loss(x, y).backward()
    eng.step()
    before = [b.flat_param.clone() for b in eng.buckets]
    before_m = [b.m.clone() for b in eng.buckets]
    path = str(tmp_path / "fast.ckpt")
    gbps_out = ckpt.save_engine_checkpoint_fast(eng, path)
    print(f"spill bandwidth: {gbps_out:.2f} GB/s")

    # clobber state, restore, verify bit-exact
    for b in eng.buckets:
        b.flat_param.zero_()
        b.m.fill_(7.0)
    eng.step_count = 0
    gbps_in = ckpt.load_engine_checkpoint_fast(eng, path)
    print(f"restore bandwidth: {gbps_in:.2f} GB/s")
    assert eng.step_count == 1
    for b, fp, m in zip(eng.buckets, before, before_m):
        assert torch.equal(b.flat_param, fp)
        assert torch.equal(b.m, m)


def test_profile_step_logs_table(capsys):
    """profile_step wraps a block and emits a kernel/op table + optional
    chrome trace through the log stream."""
    import torch

    from kubetorch_amd.utils.profiling import profile_step

    lines = []
    with profile_step("unit", top=5, printer=lines.append):
        a = torch.randn(64, 64)
        (a @ a).sum()
    out = "\n".join(lines)
    assert "[kt-profile] unit:" in out
    assert "Self CPU" in out  # the profiler table rendered


def test_profile_step_chrome_trace(tmp_path):
    import torch

    from kubetorch_amd.utils.profiling import profile_step

    lines = []
    with profile_step("tr", trace_dir=str(tmp_path), printer=lines.append):
        torch.randn(8, 8) @ torch.randn(8, 8)
    traces = list(tmp_path.glob("tr_*.json"))
    assert traces and traces[0].stat().st_size > 0
