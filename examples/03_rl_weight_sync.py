"""RL-style two-service job: a trainer publishes policy weights into the
tensor store (hipIpc registration, zero copy); inference pods pull them
(same-node device copy, or a per-transfer RCCL broadcast across nodes).
BASELINE config 5."""
import kubetorch_amd as kt

# Requires a GPU (hipIpc tensor store). On a CPU-only machine see
# tests/test_rl_two_services.py for the CPU analog of this flow.


class Trainer:
    def __init__(self):
        import torch

        from kubetorch_amd.models import Llama, llama_tiny

        self.model = Llama(llama_tiny()).cuda().bfloat16()
        self.opt = torch.optim.AdamW(self.model.parameters(), lr=1e-4)

    def train_and_publish(self, steps=10):
        import torch

        cfg = self.model.cfg
        x = torch.randint(0, cfg.vocab_size, (2, 256), device="cuda")
        for _ in range(steps):
            loss = self.model.loss(x[:, :-1], x[:, 1:])
            loss.backward()
            self.opt.step()
            self.opt.zero_grad()
        kt.put("policy/weights", dict(self.model.state_dict()),
               window=kt.BroadcastWindow(pack=True))
        return loss.item()


class Inference:
    def __init__(self):
        import torch

        from kubetorch_amd.models import Llama, llama_tiny

        self.model = Llama(llama_tiny()).cuda().bfloat16()

    def sync(self):
        import torch

        sd = {k: torch.empty_like(v) for k, v in self.model.state_dict().items()}
        kt.get("policy/weights", sd)
        self.model.load_state_dict(sd)

    def generate(self, tokens):
        import torch

        with torch.no_grad():
            return self.model(torch.tensor([tokens], device="cuda")).argmax(-1).tolist()


if __name__ == "__main__":
    import torch

    if not torch.cuda.is_available():
        raise SystemExit("example 03 needs a GPU (hipIpc tensor store) — "
                         "see tests/test_rl_two_services.py for the CPU "
                         "analog")
    trainer = kt.cls(Trainer).to(kt.Compute(gpus=4, memory="64Gi"))
    infer = kt.cls(Inference).to(
        kt.Compute(gpus=1).autoscale(min_scale=1, max_scale=8, target=4))
    for round_ in range(5):
        loss = trainer.train_and_publish(10)
        infer.sync()
        print(f"round {round_}: loss={loss:.4f}")
