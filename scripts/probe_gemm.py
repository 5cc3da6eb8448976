"""Perf probe: hand-written MFMA GEMM+bias+GELU vs hipBLASLt matmul +
bias_gelu kernel, on the BERT FFN/projection shapes."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa
import time

import torch

import sparkdl.ops as ops

shapes = [
    (32768, 3072, 768),   # FFN in (LinearGelu)
    (32768, 768, 3072),   # FFN out
    (32768, 2304, 768),   # QKV
    (32768, 768, 768),    # attn out
    (4096, 4096, 4096),   # square reference point
]


def t_ms(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


for M, N, K in shapes:
    A = torch.randn(M, K, device="cuda").bfloat16()
    W = torch.randn(N, K, device="cuda").bfloat16()
    bias = torch.randn(N, device="cuda")
    bias_bf = bias.bfloat16()
    ext = ops.ext()

    ours = t_ms(lambda: ext.gemm_bias_act(A, W, bias, 1, False))
    lib = t_ms(lambda: ext.bias_gelu_fwd(A @ W.t(), bias))
    lib_lin = t_ms(lambda: torch.nn.functional.gelu(
        torch.nn.functional.linear(A, W, bias_bf)))
    tf = 2 * M * N * K / 1e12
    print("M%6d N%5d K%5d: ours %7.3f ms (%6.0f TF) | "
          "matmul+fused-epilogue %7.3f ms (%6.0f TF) | "
          "torch linear+gelu %7.3f ms (%6.0f TF)"
          % (M, N, K, ours, tf / ours * 1000, lib, tf / lib * 1000,
             lib_lin, tf / lib_lin * 1000), flush=True)
