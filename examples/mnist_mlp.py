"""HorovodRunner quick-start: distributed MLP on synthetic MNIST-shaped
data (BASELINE.json config 1 — runs on CPU).

    python examples/mnist_mlp.py --np -2
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa
import argparse


def main(epochs=2, lr=0.05, batch=64):
    import torch
    import sparkdl.torch as hvd
    from sparkdl.horovod import log_to_driver
    from sparkdl.models.mlp import MnistMLP

    hvd.init()
    torch.manual_seed(0)
    model = MnistMLP(hidden=128)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=lr))
    hvd.broadcast_parameters(model, root_rank=0)
    cb = hvd.LogCallback(per_batch_log=False)

    g = torch.Generator().manual_seed(1 + hvd.rank())
    x = torch.randn(512, 784, generator=g)
    y = torch.randint(0, 10, (512,), generator=g)

    for epoch in range(epochs):
        if hvd.rank() == 0:
            cb.on_epoch_begin(epoch)
        total = 0.0
        for i in range(0, len(x), batch):
            opt.zero_grad()
            loss = torch.nn.functional.cross_entropy(
                model(x[i:i + batch]), y[i:i + batch])
            loss.backward()
            opt.step()
            total += float(loss)
        avg = hvd.metric_average(total / (len(x) // batch))
        if hvd.rank() == 0:
            cb.on_epoch_end(epoch, {"loss": avg})
    if hvd.rank() == 0:
        log_to_driver("training finished, final loss %.4f" % avg)
    return avg


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--np", type=int, default=-2)
    ap.add_argument("--epochs", type=int, default=2)
    args = ap.parse_args()

    from sparkdl import HorovodRunner
    hr = HorovodRunner(np=args.np)
    final = hr.run(main, epochs=args.epochs)
    print("rank-0 returned final loss: %.4f" % final)
