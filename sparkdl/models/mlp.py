"""2-layer MLP for the MNIST-shaped CPU plumbing config
(BASELINE.json config 1)."""

import torch
import torch.nn as nn


class MnistMLP(nn.Module):
    """784 → hidden → 10 classifier."""

    def __init__(self, hidden=128):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(784, hidden),
            nn.ReLU(),
            nn.Linear(hidden, 10),
        )

    def forward(self, x):
        return self.net(x.flatten(1))


def train_step_fn(seed=0, steps=4, batch=32, lr=1e-2, device="cpu"):
    """A self-contained Horovod-idiom ``main`` for tests and the plumbing
    config: synthetic MNIST-shaped batches, DistributedOptimizer,
    returns the loss curve (rank 0's value is returned by the runner)."""
    import sparkdl.torch as hvd

    hvd.init()
    torch.manual_seed(seed)  # same init on all ranks
    model = MnistMLP().to(device)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=lr))
    hvd.broadcast_parameters(model, root_rank=0)

    # Fixed per-rank synthetic shard: loss must decrease as the model
    # memorizes it, which the end-to-end tests assert.
    g = torch.Generator().manual_seed(seed + 1000 + hvd.rank())
    x = torch.randn(batch, 784, generator=g).to(device)
    y = torch.randint(0, 10, (batch,), generator=g).to(device)
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        loss = nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(hvd.allreduce(loss.detach())))
    return losses
