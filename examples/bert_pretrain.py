"""BERT-base bf16 MLM pretraining on MI355X via HorovodRunner
(BASELINE.json config 4).

    python examples/bert_pretrain.py --np 8
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa
import argparse


def main(steps=50, batch=64, seq=512):
    import torch
    import sparkdl.torch as hvd
    import sparkdl.ops as ops
    from sparkdl.models.bert import BertBase, bert_pretrain_step
    from sparkdl.utils import StepTimer

    hvd.init()
    torch.manual_seed(1234)
    model = BertBase().cuda()
    ops.convert_bf16_training(model)   # bf16 weights, fp32 masters
    opt = hvd.DistributedOptimizer(
        ops.FusedAdamW(model.parameters(), lr=1e-4, weight_decay=0.01))
    hvd.broadcast_parameters(model, root_rank=0)

    step = bert_pretrain_step(model, opt, batch, seq, "cuda", True)
    timer = StepTimer()
    for _ in range(steps):
        with timer:
            step()
    if hvd.rank() == 0:
        s = timer.summary()
        print("p50 %.1f ms/step -> %.0f sequences/sec (whole job)"
              % (s["p50_ms"], batch * hvd.size() / (s["p50_ms"] / 1000)))
    return timer.summary()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--np", type=int, default=8)
    args = ap.parse_args()
    from sparkdl import HorovodRunner
    print(HorovodRunner(np=args.np).run(main))
