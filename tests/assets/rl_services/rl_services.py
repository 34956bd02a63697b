"""RL-style two-service assets (BASELINE config 5 analog): a trainer class
that publishes weights to the tensor store, and an inference class that
pulls them — the documented trainer -> inference weight-sync path
(reference: data_store/design.md:349-405)."""
import torch
import torch.nn as nn


def _mlp():
    torch.manual_seed(7)
    return nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))


class Trainer:
    def __init__(self):
        self.model = _mlp()
        self.opt = torch.optim.SGD(self.model.parameters(), lr=0.05)
        self.version = 0

    def train_step(self, n=4):
        x = torch.randn(16, 8)
        y = torch.randn(16, 4)
        for _ in range(n):
            loss = ((self.model(x) - y) ** 2).mean()
            self.opt.zero_grad()
            loss.backward()
            self.opt.step()
        self.version += 1
        from kubetorch_amd.data_store import gpu_store

        gpu_store.put("rl/policy", dict(self.model.state_dict()))
        return {"version": self.version, "loss": loss.item()}

    def weight_sum(self):
        return sum(p.sum().item() for p in self.model.parameters())


class InferenceServer:
    def __init__(self):
        self.model = _mlp()
        self.version = -1

    def sync_weights(self):
        from kubetorch_amd.data_store import gpu_store

        sd = {k: torch.zeros_like(v) for k, v in self.model.state_dict().items()}
        gpu_store.get("rl/policy", sd)
        self.model.load_state_dict(sd)
        self.version += 1
        return self.version

    def weight_sum(self):
        return sum(p.sum().item() for p in self.model.parameters())

    def act(self, obs):
        with torch.no_grad():
            return self.model(torch.tensor(obs, dtype=torch.float32)).tolist()
