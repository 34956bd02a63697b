"""Serve Llama decode as a kubetorch_amd service: KV-cache generation on
one MI355X pod, weights fanned out to replicas through the filesystem
tree broadcast (the store uploads once; pods feed each other).

Run: python examples/05_serving_decode.py   (local driver — no cluster)
"""
import kubetorch_amd as kt


class Generator:
    """Deployed as a kt.cls service: loads weights once, serves decode."""

    def __init__(self, weights_key=None):
        import torch

        from kubetorch_amd.models import Llama, llama_tiny

        dev = "cuda" if torch.cuda.is_available() else "cpu"
        self.model = Llama(llama_tiny()).to(dev).eval()
        if dev == "cuda":
            self.model = self.model.bfloat16()
        if weights_key:
            # replicas join the rolling tree: O(1) store egress at any W
            path = kt.get_broadcast(weights_key, dest="/tmp/weights")
            sd = torch.load(f"{path}/model.pt", map_location=dev)
            self.model.load_state_dict(sd)
        self.device = dev

    def generate(self, prompt_ids, max_new_tokens=16, temperature=0.0):
        import torch

        toks = torch.tensor([prompt_ids], device=self.device)
        out = self.model.generate(toks, max_new_tokens,
                                  temperature=temperature)
        return out[0].tolist()


if __name__ == "__main__":
    gen = kt.cls(Generator).to(kt.Compute(gpus=0))  # gpus=1 on a cluster
    try:
        ids = gen.generate([1, 2, 3, 4], max_new_tokens=8)
        print("generated:", ids)
    finally:
        gen.teardown()
