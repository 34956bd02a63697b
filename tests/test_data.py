"""Token dataset / sharded loader tests (CPU)."""
import numpy as np
import torch

from kubetorch_amd.data import ShardedLoader, TokenDataset, synthetic_tokens


def _ident_dataset(n_samples=64, seq_len=8):
    # tokens = arange so a sample's first token identifies its index
    return TokenDataset(torch.arange(n_samples * seq_len + 1), seq_len)


def test_dataset_slicing():
    ds = _ident_dataset(4, 8)
    assert len(ds) == 4
    x, y = ds.sample(2)
    assert x.tolist() == list(range(16, 24))
    assert y.tolist() == list(range(17, 25))  # next-token shift


def test_sharding_disjoint_and_complete():
    ds = _ident_dataset(64, 8)
    world, batch = 4, 2
    seen = []
    for rank in range(world):
        ld = ShardedLoader(ds, batch=batch, rank=rank, world=world, seed=7)
        for x, _y in ld:
            seen.extend((x[:, 0] // 8).tolist())
    assert len(seen) == len(set(seen)), "ranks overlapped"
    assert len(seen) == world * len(ShardedLoader(ds, batch, 0, world)) * batch


def test_epoch_shuffle_and_determinism():
    ds = _ident_dataset(32, 8)
    ld = ShardedLoader(ds, batch=4, seed=3)
    e0 = [x[:, 0].tolist() for x, _ in ld]
    e0b = [x[:, 0].tolist() for x, _ in ld]
    assert e0 == e0b  # same epoch -> same order
    ld.set_epoch(1)
    e1 = [x[:, 0].tolist() for x, _ in ld]
    assert e0 != e1  # reshuffled across epochs
    # no-shuffle is sequential
    ld2 = ShardedLoader(ds, batch=4, shuffle=False)
    first_x, first_y = next(iter(ld2))
    assert first_x[0, 0].item() == 0


def test_memmap_bin_file(tmp_path):
    p = tmp_path / "toks.bin"
    arr = np.arange(1000, dtype=np.uint16)
    arr.tofile(p)
    ds = TokenDataset(str(p), seq_len=10)
    x, y = ds.sample(0)
    assert x.dtype == torch.int64 and x[0] == 0 and y[-1] == 10


def test_synthetic_tokens_and_train_smoke():
    from kubetorch_amd.models import Llama, llama_tiny
    from kubetorch_amd.parallel import FlatDDP

    toks = synthetic_tokens(512, 4096, seed=1)
    ds = TokenDataset(toks, seq_len=32)
    ld = ShardedLoader(ds, batch=2, seed=1)
    model = Llama(llama_tiny())
    eng = FlatDDP(model, lr=1e-3, bucket_mb=4)
    losses = []
    for i, (x, y) in enumerate(ld):
        assert x.shape == (2, 32) and y.shape == (2, 32)
        loss = model.loss(x, y)
        loss.backward()
        eng.step()
        losses.append(loss.item())
        if i == 3:
            break
    # smoke: training stepped over fresh shuffled batches without blowup
    # (monotone decrease needs repeated data — covered in test_model)
    assert len(losses) == 4 and all(np.isfinite(losses))


def test_loader_errors_and_edges():
    from kubetorch_amd.data import ShardedLoader, TokenDataset

    with np.testing.assert_raises(ValueError):
        TokenDataset(torch.arange(5), seq_len=10)  # too short
    ds = TokenDataset(torch.arange(65), 8)
    with np.testing.assert_raises(ValueError):
        ShardedLoader(ds, batch=100)  # not enough for one batch
    # 2-D input is flattened
    ds2 = TokenDataset(torch.arange(64).reshape(8, 8), 4)
    x, y = ds2.sample(0)
    assert x.tolist() == [0, 1, 2, 3]


def test_tokenizer_load_missing(tmp_path):
    import pytest as _pytest

    from kubetorch_amd.models.tokenizer import Tokenizer

    with _pytest.raises(FileNotFoundError):
        Tokenizer.load(str(tmp_path))  # dir without artifacts
