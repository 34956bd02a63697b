"""Token data loading for the training engine.

The reference ships no data loader (users bring their own); this is the
MI355X-native one the flagship workload uses off synthetic or memmapped
token shards:

  * `TokenDataset` — a flat token stream (torch tensor, numpy array, or a
    memmapped `.bin` file of uint16/uint32 tokens) sliced into (x, y)
    next-token pairs of `seq_len`.
  * `ShardedLoader` — rank-sharded, epoch-shuffled batches with pinned
    host staging and one-batch-ahead async H2D on a dedicated copy stream,
    so the transfer of batch N+1 overlaps the compute of batch N (HBM3E is
    fast; the PCIe/host side is what needs hiding).
"""
import numpy as np
import torch


class TokenDataset:
    def __init__(self, source, seq_len):
        if isinstance(source, str):
            arr = np.memmap(source, dtype=np.uint16, mode="r")
            self.tokens = torch.from_numpy(np.asarray(arr).astype(np.int64))
        elif isinstance(source, np.ndarray):
            self.tokens = torch.from_numpy(source.astype(np.int64))
        else:
            self.tokens = source.to(torch.int64)
        if self.tokens.dim() != 1:
            self.tokens = self.tokens.reshape(-1)
        self.seq_len = seq_len
        # each sample needs seq_len + 1 tokens (x and the shifted target)
        self.n_samples = (self.tokens.numel() - 1) // seq_len
        if self.n_samples <= 0:
            raise ValueError("token stream shorter than seq_len + 1")

    def __len__(self):
        return self.n_samples

    def sample(self, idx):
        s = idx * self.seq_len
        chunk = self.tokens[s:s + self.seq_len + 1]
        return chunk[:-1], chunk[1:]


class ShardedLoader:
    """Iterates (x, y) batches of [batch, seq_len] for this rank.

    Sharding: the epoch's shuffled sample order is split contiguously by
    rank (rank r takes samples [r*per_rank, (r+1)*per_rank)) — every sample
    is seen by exactly one rank per epoch and all ranks make the same
    number of steps (the tail that doesn't fill every rank is dropped, so
    collective-calling training loops stay in lockstep).
    """

    def __init__(self, dataset, batch, rank=0, world=1, seed=1234,
                 shuffle=True, device=None, prefetch=True):
        self.ds = dataset
        self.batch = batch
        self.rank = rank
        self.world = world
        self.seed = seed
        self.shuffle = shuffle
        self.device = torch.device(device) if device is not None else None
        self.epoch = 0
        per_rank = len(dataset) // world
        self.steps_per_epoch = per_rank // batch
        if self.steps_per_epoch == 0:
            raise ValueError("not enough samples for one batch per rank")
        self._use_prefetch = (prefetch and self.device is not None
                              and self.device.type == "cuda")
        self._copy_stream = (torch.cuda.Stream(self.device)
                             if self._use_prefetch else None)

    def set_epoch(self, epoch):
        self.epoch = epoch

    def _epoch_order(self):
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            order = torch.randperm(len(self.ds), generator=g)
        else:
            order = torch.arange(len(self.ds))
        per_rank = len(self.ds) // self.world
        return order[self.rank * per_rank:(self.rank + 1) * per_rank]

    def _host_batch(self, idxs):
        xs, ys = zip(*(self.ds.sample(int(i)) for i in idxs))
        x = torch.stack(xs)
        y = torch.stack(ys)
        if self._use_prefetch:
            x = x.pin_memory()
            y = y.pin_memory()
        return x, y

    def __iter__(self):
        order = self._epoch_order()
        n = self.steps_per_epoch
        if not self._use_prefetch:
            for i in range(n):
                x, y = self._host_batch(order[i * self.batch:(i + 1) * self.batch])
                if self.device is not None:
                    x, y = x.to(self.device), y.to(self.device)
                yield x, y
            return

        def stage(i):
            # async H2D on the copy stream; the consumer's stream waits on
            # the recorded event, not on the whole device
            hx, hy = self._host_batch(order[i * self.batch:(i + 1) * self.batch])
            with torch.cuda.stream(self._copy_stream):
                dx = hx.to(self.device, non_blocking=True)
                dy = hy.to(self.device, non_blocking=True)
                ev = torch.cuda.Event()
                ev.record(self._copy_stream)
            return dx, dy, ev, (hx, hy)  # keep host refs until the copy lands

        nxt = stage(0)
        for i in range(n):
            dx, dy, ev, _host = nxt
            nxt = stage(i + 1) if i + 1 < n else None
            torch.cuda.current_stream(self.device).wait_event(ev)
            yield dx, dy

    def __len__(self):
        return self.steps_per_epoch


def synthetic_tokens(vocab_size, n_tokens, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, vocab_size, (n_tokens,), generator=g)
