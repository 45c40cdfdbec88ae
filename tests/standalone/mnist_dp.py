"""Standalone DP smoke (reference tests/standalone/mnist.py): torchrun
--nproc-per-node 2 --master-addr 127.0.0.1 tests/standalone/mnist_dp.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch  # noqa: E402

import torchacc_amd as ta  # noqa: E402


class MLP(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.net = torch.nn.Sequential(
            torch.nn.Linear(784, 128), torch.nn.ReLU(),
            torch.nn.Linear(128, 10))

    def forward(self, x, labels=None):
        logits = self.net(x)
        if labels is not None:
            return torch.nn.functional.cross_entropy(logits, labels)
        return logits


def main():
    cfg = ta.Config()
    cfg.dist.dp.size = int(os.environ.get("WORLD_SIZE", 1))
    torch.manual_seed(0)
    model = ta.accelerate(MLP(), config=cfg)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    rank = int(os.environ.get("RANK", 0))
    torch.manual_seed(100 + rank)
    # fixed batch: random labels are learnable only by memorization
    x = torch.randn(32, 784)
    y = torch.randint(0, 10, (32,))
    losses = []
    for step in range(20):
        loss = model(x, labels=y)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses
    if rank == 0:
        print(f"mnist_dp OK: {losses[0]:.3f} -> {losses[-1]:.3f}")


if __name__ == "__main__":
    main()
