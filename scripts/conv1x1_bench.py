"""Per-site A/B of the ResNet-50 1x1 conv GEMMs: streaming kernel vs
256x256 tile kernel vs MIOpen conv (channels_last bf16)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa
import torch
import sparkdl.ops as ops

# (M, K, N) per distinct ResNet-50 1x1 site at bs512 (NHWC rows)
SITES = [
    (512 * 56 * 56, 64, 64),     # L1 conv1 (first block)
    (512 * 56 * 56, 64, 256),    # L1 conv3 / downsample
    (512 * 56 * 56, 256, 64),    # L1 conv1 (later blocks)
    (512 * 28 * 28, 256, 128),   # L2 conv1 (first)
    (512 * 28 * 28, 128, 512),   # L2 conv3
    (512 * 28 * 28, 512, 128),   # L2 conv1 (later)
    (512 * 14 * 14, 512, 256),   # L3 conv1 (first)
    (512 * 14 * 14, 256, 1024),  # L3 conv3
    (512 * 14 * 14, 1024, 256),  # L3 conv1 (later)
    (512 * 7 * 7, 1024, 512),    # L4 conv1 (first)
    (512 * 7 * 7, 512, 2048),    # L4 conv3
    (512 * 7 * 7, 2048, 512),    # L4 conv1 (later)
]


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / iters * 1000  # us


def main():
    C = ops.ext()
    torch.backends.cudnn.benchmark = True
    for (M, K, N) in SITES:
        a = torch.randn(M, K, device="cuda").bfloat16()
        w = torch.randn(N, K, device="cuda").bfloat16()
        res = {}
        if K in (64, 128, 256) and N % 64 == 0:
            res["stream"] = timeit(lambda: C.gemm_stream(a, w))
        res["g16"] = timeit(
            lambda: C.gemm_bias_act(a, w, None, 0, False))
        # MIOpen conv view: x [B, K, H, W] channels_last
        HW = {56 * 56 * 512: 56, 28 * 28 * 512: 28, 14 * 14 * 512: 14,
              7 * 7 * 512: 7}[M]
        x4 = a.view(512, HW, HW, K).permute(0, 3, 1, 2)
        w4 = w.view(N, K, 1, 1)
        res["miopen"] = timeit(lambda: torch.nn.functional.conv2d(
            x4, w4))
        res["blaslt"] = timeit(lambda: a @ w.t())
        gf = 2.0 * M * K * N / 1e9
        line = "M=%8d K=%4d N=%4d  " % (M, K, N)
        for k, v in res.items():
            line += "%s=%7.1fus(%4.0fTF)  " % (k, v, gf / (v / 1e6) / 1e3)
        print(line, flush=True)


if __name__ == "__main__":
    main()
