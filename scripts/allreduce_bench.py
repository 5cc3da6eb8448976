"""RCCL all-reduce micro-benchmark with bus-bandwidth report vs the
xGMI roofline (BASELINE.md last row).

Launch:  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
             --master-addr 127.0.0.1 scripts/allreduce_bench.py
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa
import json

import torch

import sparkdl.torch as hvd
from sparkdl.utils.profiling import CommTimer


def main():
    hvd.init()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    timer = CommTimer()
    # ResNet-50-sized gradient buckets and the sweep around them
    for mb in (1, 4, 16, 32, 64, 102, 128):
        t = torch.randn(mb * 1024 * 1024 // 4, device=dev)
        rec = timer.allreduce(t, iters=20, warmup=5)
        if hvd.rank() == 0:
            print(json.dumps({"MB": mb, **{k: round(v, 3) if
                  isinstance(v, float) else v for k, v in rec.items()}}),
                  flush=True)
    hvd.shutdown()


if __name__ == "__main__":
    main()
