"""Per-step wall-clock and all-reduce bandwidth timers
(tracing/profiling subsystem).

For kernel-level evidence use rocprofv3 (see scripts/profile_bench.sh);
these timers cover the runner-level metrics: step time distribution and
achieved all-reduce bus bandwidth vs the xGMI roofline.
"""

import time

import torch


class StepTimer:
    """Wall-clock step timer with device sync; prints percentiles."""

    def __init__(self, sync_cuda=True):
        self.sync_cuda = sync_cuda and torch.cuda.is_available()
        self.times = []
        self._t0 = None

    def __enter__(self):
        if self.sync_cuda:
            torch.cuda.synchronize()
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self.sync_cuda:
            torch.cuda.synchronize()
        self.times.append(time.perf_counter() - self._t0)

    def summary(self):
        if not self.times:
            return {}
        ts = sorted(self.times)
        n = len(ts)
        return {
            "steps": n,
            "mean_ms": sum(ts) / n * 1000,
            "p50_ms": ts[n // 2] * 1000,
            "p95_ms": ts[min(n - 1, int(n * 0.95))] * 1000,
            "max_ms": ts[-1] * 1000,
        }


# Single-node 8x MI355X: each GPU reaches its 7 peers over dedicated
# xGMI links at ~153 GB/s each.
XGMI_LINK_GBPS = 153.0
XGMI_LINKS_PER_GPU = 7


class CommTimer:
    """Measures all-reduce algorithmic and bus bandwidth.

    bus_bw = algo_bw * 2*(n-1)/n for ring all-reduce; reported against
    the per-GPU xGMI roofline (7 x 153 GB/s).
    """

    def __init__(self):
        self.records = []

    def allreduce(self, tensor, iters=10, warmup=3):
        import torch.distributed as dist
        n = dist.get_world_size()
        dev_sync = tensor.is_cuda
        for _ in range(warmup):
            dist.all_reduce(tensor)
        if dev_sync:
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            dist.all_reduce(tensor)
        if dev_sync:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        bytes_ = tensor.numel() * tensor.element_size()
        algo_bw = bytes_ / dt / 1e9
        bus_bw = algo_bw * 2 * (n - 1) / n
        rec = {
            "bytes": bytes_, "seconds": dt, "world": n,
            "algo_GBps": algo_bw, "bus_GBps": bus_bw,
            "xgmi_roofline_GBps": XGMI_LINK_GBPS * XGMI_LINKS_PER_GPU,
            "roofline_frac": bus_bw / (XGMI_LINK_GBPS *
                                       XGMI_LINKS_PER_GPU),
        }
        self.records.append(rec)
        return rec
