"""Library-GEMM baseline on the probe shapes (A/B vs the hand-written
256x256 kernels). torch.matmul bf16 on ROCm routes to hipBLASLt /
rocBLAS (with the committed TunableOp tunings when enabled).

Run on the GPU box:
    PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=0 \
    PYTORCH_TUNABLEOP_FILENAME=tunableop_gfx950.csv \
    python scripts/hipblaslt_bench.py
"""
import torch

SHAPES = [
    (4096, 4096, 4096),
    (8192, 8192, 8192),
    (32768, 3072, 768),
    (32768, 768, 3072),
    (32768, 768, 768),
    (32768, 2304, 768),
]


def main():
    torch.manual_seed(0)
    for (m, n, k) in SHAPES:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        for _ in range(3):
            c = a @ w.t()
        torch.cuda.synchronize()
        iters = 10
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record()
        for _ in range(iters):
            c = a @ w.t()
        e1.record()
        torch.cuda.synchronize()
        ms = e0.elapsed_time(e1) / iters
        tf = 2.0 * m * n * k / (ms / 1e3) / 1e12
        print("hipblaslt %dx%dx%d: %.3f ms, %.0f TFLOP/s" % (m, n, k, ms, tf))


if __name__ == "__main__":
    main()
