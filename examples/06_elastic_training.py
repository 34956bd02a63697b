"""Elastic training, both levels (BASELINE config 4):

1. Per-call elasticity (the control plane's job): kill a worker pod
   mid-call and the in-flight distributed call aborts with
   WorkerMembershipChanged, the controller's pod monitor re-provisions
   the pod automatically, and the NEXT call's rendezvous re-forms the
   group — so the client-side loop is simply "catch, retry".

2. In-step re-join (beyond the reference): inside the training function
   itself, ElasticStepper recovers from a collective failure WITHOUT
   restarting the job process — survivors re-rendezvous, re-form a
   smaller process group, re-sync params by broadcast and retry the
   step. No checkpoint reload.

Run: KT_LOCAL_MODE=true PYTHONPATH=. python examples/06_elastic_training.py
"""
import kubetorch_amd as kt


def train_elastically(steps: int = 20):
    import os

    import torch

    from kubetorch_amd.parallel import (ElasticStepper, FlatDDP,
                                        PeersRendezvous, init_distributed)

    rank, world, _ = init_distributed(backend="gloo")
    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(), torch.nn.Linear(128, 8))
    engine = FlatDDP(model, lr=1e-2, bucket_mb=1, overlap_optimizer=False)
    engine.broadcast_params(src=0)
    # survivors re-discover each other through the control plane's live
    # peer list when a rank dies mid-step
    stepper = ElasticStepper(engine, PeersRendezvous(), pg_timeout_s=30)

    x = torch.randn(32, 64)
    y = torch.randn(32, 8)

    def fb():
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        return loss

    losses = [float(stepper.step(fb)) for _ in range(steps)]
    return {"rank": int(os.environ.get("RANK", 0)),
            "first": losses[0], "last": losses[-1],
            "reforms": stepper.reforms}


def main():
    from kubetorch_amd.exceptions import WorkerMembershipChanged

    f = kt.fn(train_elastically).to(
        kt.Compute(cpus=1).distribute("pytorch", workers=2, num_proc=1))
    try:
        # level 1: the client-side retry loop around pod death
        for attempt in range(3):
            try:
                results = f(steps=20, kt_timeout=300)
                break
            except WorkerMembershipChanged:
                print("membership changed mid-call; controller is "
                      "re-provisioning — retrying")
        for r in results:
            print(f"rank {r['rank']}: loss {r['first']:.4f} -> {r['last']:.4f} "
                  f"(in-step reforms: {r['reforms']})")
    finally:
        f.teardown()


if __name__ == "__main__":
    main()
