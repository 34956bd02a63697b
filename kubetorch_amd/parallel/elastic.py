"""In-step elastic re-join: survivors re-form a smaller process group and
continue training WITHOUT restarting the job process or reloading a
checkpoint.

This goes beyond the reference (SURVEY.md §7 stage 4): the reference's
elasticity is per-call — a failed distributed call aborts, the replacement
pod joins the NEXT call's rendezvous (spmd_supervisor.py:476-537). Here the
training loop itself recovers mid-job:

    stepper = ElasticStepper(engine, rendezvous)
    for batch in data:
        loss = stepper.step(lambda: forward_backward(batch))

On a collective failure (peer died: gloo raises a connection/timeout error;
RCCL raises under TORCH_NCCL_ASYNC_ERROR_HANDLING=1):
  1. abort engine comm state (drop in-flight bucket all-reduces),
  2. re-run rendezvous among survivors -> new (rank, world) + a fresh
     TCPStore generation (stale store keys can't collide),
  3. re-init the process group and re-sync params by broadcast from the
     new rank 0 (a dying peer can leave ranks mid-step with partially
     reduced buckets — broadcast restores bitwise consistency without a
     checkpoint),
  4. retry the step with the new world size (grad mean re-scaled).

ZeRO-1 engines (zero=True) size optimizer shards by world and cannot
re-form in place; ElasticStepper refuses them at construction.
"""
import os
import time
import uuid

import torch
import torch.distributed as dist


def reform_process_group(rank, world, master_addr, master_port,
                         backend=None, timeout_s=60):
    """Destroy the current default group (if any) and re-init with an
    explicit TCPStore. Safe to call repeatedly; each call must use a fresh
    (addr, port) generation so no stale store state survives."""
    if dist.is_initialized():
        try:
            dist.destroy_process_group()
        except Exception:
            pass
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    import datetime

    store = dist.TCPStore(master_addr, master_port, world, rank == 0,
                          timeout=datetime.timedelta(seconds=timeout_s))
    dist.init_process_group(backend, store=store, rank=rank,
                            world_size=world,
                            timeout=datetime.timedelta(seconds=timeout_s))


class FileRendezvous:
    """Directory-based survivor rendezvous for single-host worlds (the
    local driver, tests, one-node jobs). Each reform generation g uses
    fresh files `g{g}-{uid}` and a fresh master port (base + g).

    `settle` is how long the member set must be stable before concluding
    — it MUST exceed the worst-case spread between survivors detecting
    the failure (≈ the process-group timeout plus scheduling jitter), or
    a fast detector can conclude a smaller world before slow detectors
    arrive and the group splits. ElasticStepper wires this automatically
    (settle >= pg_timeout_s + 2)."""

    def __init__(self, root, uid=None, master_addr="127.0.0.1",
                 port_base=29700, settle=2.0, timeout=60.0):
        self.root = root
        self.uid = uid or uuid.uuid4().hex[:8]
        self.master_addr = master_addr
        self.port_base = port_base
        self.settle = settle
        self.timeout = timeout
        self.generation = 0
        os.makedirs(root, exist_ok=True)

    def __call__(self):
        self.generation += 1
        g = self.generation
        my = os.path.join(self.root, f"g{g}-{self.uid}")
        with open(my, "w") as f:
            f.write(str(os.getpid()))
        deadline = time.time() + self.timeout
        members = None
        stable_since = time.time()
        while time.time() < deadline:
            cur = sorted(n for n in os.listdir(self.root)
                         if n.startswith(f"g{g}-"))
            if cur != members:
                members = cur
                stable_since = time.time()
            elif time.time() - stable_since >= self.settle and members:
                uids = [m.split("-", 1)[1] for m in members]
                rank = uids.index(self.uid)
                return rank, len(uids), self.master_addr, self.port_base + g
            time.sleep(0.1)
        raise TimeoutError(f"rendezvous generation {g} did not settle")


class PeersRendezvous:
    """In-pod rendezvous over the control plane's live peer list
    (discovery.current_peers — reflects pod death and respawn). Rank order
    is the sorted host list; master is rank 0's host."""

    def __init__(self, service_name=None, namespace=None, port_base=29700,
                 quorum_timeout=60.0, settle=3.0):
        self.service_name = service_name
        self.namespace = namespace
        self.port_base = port_base
        self.quorum_timeout = quorum_timeout
        self.settle = settle
        self.generation = 0

    def __call__(self):
        from kubetorch_amd.serving import discovery
        from kubetorch_amd.serving.supervisors import _self_host

        self.generation += 1
        deadline = time.time() + self.quorum_timeout
        prev = None
        stable_since = time.time()
        while time.time() < deadline:
            cur = sorted(discovery.current_peers(self.service_name,
                                                 self.namespace))
            if cur != prev:
                prev = cur
                stable_since = time.time()
            elif cur and time.time() - stable_since >= self.settle:
                me = _self_host()
                if me not in cur:
                    raise RuntimeError(f"self {me} not in peer list {cur}")
                rank = cur.index(me)
                addr = cur[0].split(":")[0]
                return rank, len(cur), addr, self.port_base + self.generation
            time.sleep(0.25)
        raise TimeoutError("peer rendezvous did not settle")


class ElasticStepper:
    """Wraps a FlatDDP engine's train step with in-step fault recovery."""

    def __init__(self, engine, rendezvous, max_reforms=4, backend=None,
                 pg_timeout_s=60):
        if getattr(engine, "zero", False):
            raise ValueError(
                "ZeRO-1 shards are sized by world at construction; in-step "
                "re-join needs zero=False (re-create the engine and "
                "load_state_dict to resume a ZeRO run elastically)")
        self.engine = engine
        self.rendezvous = rendezvous
        self.max_reforms = max_reforms
        self.backend = backend
        self.pg_timeout_s = pg_timeout_s
        self.reforms = 0
        # survivors detect a failure up to ~pg_timeout apart; the
        # rendezvous must wait at least that long for stragglers
        min_settle = pg_timeout_s + 2.0
        if getattr(rendezvous, "settle", None) is not None                 and rendezvous.settle < min_settle:
            rendezvous.settle = min_settle
        if getattr(rendezvous, "timeout", 0) < 4 * min_settle:
            rendezvous.timeout = 4 * min_settle

    def step(self, forward_backward, lr=None):
        """Run forward_backward() + engine.step(); on a collective failure
        re-form the group among survivors and retry. Returns the loss."""
        for attempt in range(self.max_reforms + 1):
            try:
                loss = forward_backward()
                self.engine.step(lr=lr)
                return loss
            except RuntimeError as e:
                if attempt >= self.max_reforms:
                    raise
                self._reform(e)

    def _reform(self, cause):
        self.reforms += 1
        self.engine.abort_comm()
        rank, world, addr, port = self.rendezvous()
        reform_process_group(rank, world, addr, port, backend=self.backend,
                             timeout_s=self.pg_timeout_s)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        self.engine.set_world(world)
        # partially reduced buckets can differ across survivors: restore
        # bitwise-identical params from the new rank 0 (no checkpoint load)
        self.engine.broadcast_params(src=0)
