"""Microbenchmark: fused BatchNormAct2d vs torch BN+ReLU on ResNet-50
shapes (bf16 channels_last, training fwd+bwd)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

import sparkdl.ops as ops

shapes = [  # (N, C, H, W) representative ResNet-50 BN sites
    (256, 64, 112, 112),
    (256, 256, 56, 56),
    (256, 512, 28, 28),
    (256, 1024, 14, 14),
    (256, 2048, 7, 7),
]


def bench(fn, x, iters=10):
    for _ in range(3):
        y = fn(x)
        y.backward(torch.ones_like(y))
        x.grad = None
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        y = fn(x)
        y.backward(torch.ones_like(y))
        x.grad = None
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


for N, C, H, W in shapes:
    x = torch.randn(N, C, H, W, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_(True)

    m_ours = ops.BatchNormAct2d(C, relu=True).cuda()
    m_ref = torch.nn.Sequential(
        torch.nn.BatchNorm2d(C), torch.nn.ReLU()).cuda()

    t_ours = bench(lambda t: m_ours(t), x)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        pass
    def ref_fn(t):
        with torch.autocast("cuda", dtype=torch.bfloat16):
            return m_ref(t)
    t_ref = bench(ref_fn, x)
    print("N%d C%4d H%3d: ours %7.3f ms  torch(BN+ReLU,autocast) %7.3f ms"
          % (N, C, H, t_ours, t_ref), flush=True)
