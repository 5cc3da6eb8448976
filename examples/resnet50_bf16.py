"""ResNet-50 bf16 training on an 8x MI355X node via HorovodRunner
(BASELINE.json configs 2-3).

    python examples/resnet50_bf16.py --np 8
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa
import argparse


def main(steps=50, batch=512):
    import torch
    import sparkdl.torch as hvd
    import sparkdl.ops as ops
    from sparkdl.models.resnet import ResNet50
    from sparkdl.utils import StepTimer, save_checkpoint

    hvd.init()
    torch.manual_seed(1234)
    torch.backends.cudnn.benchmark = True
    model = ResNet50().cuda().to(memory_format=torch.channels_last)
    opt = hvd.DistributedOptimizer(
        ops.FusedSGD(model.parameters(), lr=0.256 * hvd.size() / 8,
                     momentum=0.875, weight_decay=1 / 32768))
    hvd.broadcast_parameters(model, root_rank=0)

    x = torch.randn(batch, 3, 224, 224, device="cuda") \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device="cuda")

    timer = StepTimer()
    for step in range(steps):
        with timer:
            opt.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = torch.nn.functional.cross_entropy(model(x), y)
            loss.backward()
            opt.step()
    save_checkpoint("/tmp/resnet50_ck.pt", model, opt, step=steps)
    if hvd.rank() == 0:
        s = timer.summary()
        img_s = batch * hvd.size() / (s["p50_ms"] / 1000)
        print("p50 %.1f ms/step -> %.0f images/sec (whole job)"
              % (s["p50_ms"], img_s))
    return timer.summary()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--np", type=int, default=8)
    args = ap.parse_args()
    from sparkdl import HorovodRunner
    print(HorovodRunner(np=args.np).run(main))
