"""DistributedOptimizer — bucketed all-reduce overlapped with backward.

The MI355X-native re-implementation of the Horovod ``DistributedOptimizer``
semantics the reference's runner exists to host (reference
runner_base.py:25-35).  Design:

  - Parameters are packed, in reverse registration order (approximate
    backward order), into flat per-dtype gradient buckets.  ``p.grad`` is
    pre-set to a view into the flat bucket, so autograd accumulates
    directly into the communication buffer — no pack/unpack copies on the
    hot path.
  - When the last gradient of a bucket lands
    (``register_post_accumulate_grad_hook``), the bucket's flat buffer is
    all-reduced asynchronously.  With the RCCL backend the collective runs
    on RCCL's own HIP stream, overlapping the remaining backward compute.
  - ``step()`` waits for outstanding collectives, averages, and runs the
    wrapped optimizer.

Bucket sizing for xGMI: each MI355X reaches its 7 peers over dedicated
point-to-point links (~153 GB/s each), so ring collectives are per-link
bound and message sizes must be large enough to amortize per-collective
launch latency across the ring hops.  Default bucket cap is 64 MiB
(SPARKDL_BUCKET_MB overrides) — larger than NCCL/NVSwitch-tuned defaults
(e.g. DDP's 25 MiB) because the per-link-bound ring favours fewer, larger
transfers while backward still provides ample overlap window.
"""

import contextlib
import os

import torch
import torch.distributed as dist

from sparkdl.parallel import comm

_DEFAULT_BUCKET_MB = float(os.environ.get("SPARKDL_BUCKET_MB", "64"))


class _Bucket:
    __slots__ = ("params", "flat", "numel", "offsets", "work", "launched")

    def __init__(self, params, offsets, numel, device, dtype):
        self.params = params
        self.offsets = offsets
        self.numel = numel
        self.flat = torch.zeros(numel, device=device, dtype=dtype)
        self.work = None
        self.launched = False


class DistributedOptimizer(torch.optim.Optimizer):
    """Wrap a torch optimizer with gradient averaging across all ranks.

    Horovod-compatible surface: ``DistributedOptimizer(opt,
    named_parameters=None, backward_passes_per_step=1)``; also provides
    ``synchronize()`` and ``skip_synchronize()``.
    """

    def __init__(self, optimizer, named_parameters=None,
                 backward_passes_per_step=1, bucket_cap_mb=None,
                 compression=None, op=None):
        # compression/op: accepted for Horovod drop-in compatibility;
        # reduction is always sum-then-average in the bucket dtype
        # (bf16 buckets already halve the wire traffic).
        # Not calling super().__init__: we delegate everything to the
        # wrapped optimizer and only intercept step/zero_grad.
        self.optimizer = optimizer
        self.backward_passes_per_step = max(1, int(backward_passes_per_step))
        self._bucket_bytes = int(
            (bucket_cap_mb or _DEFAULT_BUCKET_MB) * 1024 * 1024)
        self._buckets = []
        self._param_bucket = {}
        self._ready = {}
        self._hooks = []
        self._pass = 0
        self._skip_sync = False
        self._require_sync = True
        if comm.size() > 1:
            self._build_buckets()
            self._register_hooks()

    # -- Optimizer protocol delegation ----------------------------------
    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @param_groups.setter
    def param_groups(self, value):
        self.optimizer.param_groups = value

    @property
    def state(self):
        return self.optimizer.state

    @property
    def defaults(self):
        return self.optimizer.defaults

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, state_dict):
        self.optimizer.load_state_dict(state_dict)

    def add_param_group(self, group):
        if comm.size() > 1:
            raise RuntimeError(
                "add_param_group after DistributedOptimizer construction "
                "is not supported; build the optimizer with all groups.")
        self.optimizer.add_param_group(group)

    def __repr__(self):
        return "DistributedOptimizer(%r)" % (self.optimizer,)

    # -- Bucketing ------------------------------------------------------
    def _all_params(self):
        seen = set()
        out = []
        for group in self.optimizer.param_groups:
            for p in group["params"]:
                if p.requires_grad and id(p) not in seen:
                    seen.add(id(p))
                    out.append(p)
        return out

    def _build_buckets(self):
        # Reverse order approximates gradient-ready order during backward
        # (last layers produce grads first), maximizing comm overlap.
        params = list(reversed(self._all_params()))
        groups = {}
        for p in params:
            groups.setdefault((p.device, p.dtype), []).append(p)
        for (device, dtype), ps in groups.items():
            cap = max(1, self._bucket_bytes // ps[0].element_size())
            cur, offsets, numel = [], [], 0
            for p in ps:
                if cur and numel + p.numel() > cap:
                    self._buckets.append(
                        _Bucket(cur, offsets, numel, device, dtype))
                    cur, offsets, numel = [], [], 0
                cur.append(p)
                offsets.append(numel)
                numel += p.numel()
            if cur:
                self._buckets.append(
                    _Bucket(cur, offsets, numel, device, dtype))
        # Alias p.grad to flat views so autograd accumulates in place.
        for bi, b in enumerate(self._buckets):
            for p, off in zip(b.params, b.offsets):
                p.grad = b.flat[off:off + p.numel()].view_as(p)
                self._param_bucket[id(p)] = bi
            self._ready[bi] = 0

    def _register_hooks(self):
        for b in self._buckets:
            for p in b.params:
                h = p.register_post_accumulate_grad_hook(self._grad_ready)
                self._hooks.append(h)

    def _grad_ready(self, p):
        bi = self._param_bucket[id(p)]
        b = self._buckets[bi]
        self._ready[bi] += 1
        # Each param fires once per backward; with gradient accumulation
        # (backward_passes_per_step=k) launch only on the k-th backward.
        if self._ready[bi] == len(b.params) * self.backward_passes_per_step:
            self._ready[bi] = 0
            self._launch(b)

    def _launch(self, bucket):
        if bucket.launched:
            return
        bucket.launched = True
        bucket.work = dist.all_reduce(
            bucket.flat, op=dist.ReduceOp.SUM, async_op=True)

    # -- Synchronization ------------------------------------------------
    def synchronize(self):
        """Wait for all gradient all-reduces and average."""
        if comm.size() <= 1:
            return
        ws = comm.size()
        for b in self._buckets:
            # Buckets whose params never fired (unused in graph) still
            # hold zeros — reducing them keeps ranks consistent.
            self._launch(b)
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            b.launched = False
            b.flat.div_(ws)
        for bi in self._ready:
            self._ready[bi] = 0
        self._pass += 1

    @contextlib.contextmanager
    def skip_synchronize(self):
        """Horovod-compatible: run step() without gradient averaging."""
        self._skip_sync = True
        try:
            yield
        finally:
            self._skip_sync = False

    def step(self, closure=None):
        if comm.size() > 1 and not self._skip_sync:
            self.synchronize()
        return self.optimizer.step(closure)

    def zero_grad(self, set_to_none=False):
        if comm.size() > 1:
            # Grads are views of the flat buffers — zero in place, never
            # detach (set_to_none would break the aliasing).
            for b in self._buckets:
                b.flat.zero_()
            for bi in self._ready:
                self._ready[bi] = 0
        else:
            self.optimizer.zero_grad(set_to_none=set_to_none)


def broadcast_parameters(params, root_rank=0):
    """Broadcast model parameters (module, state_dict, or iterable of
    (name, tensor) / tensors) from root_rank to all ranks — the Horovod
    initial-state sync."""
    if comm.size() <= 1:
        return
    tensors = _extract_tensors(params)
    with torch.no_grad():
        for t in tensors:
            dist.broadcast(t, src=root_rank)


def broadcast_optimizer_state(optimizer, root_rank=0):
    """Broadcast optimizer state from root_rank (Horovod-compatible)."""
    if comm.size() <= 1:
        return
    if isinstance(optimizer, DistributedOptimizer):
        optimizer = optimizer.optimizer
    obj = [optimizer.state_dict() if comm.rank() == root_rank else None]
    dist.broadcast_object_list(obj, src=root_rank)
    if comm.rank() != root_rank:
        optimizer.load_state_dict(obj[0])


def _extract_tensors(params):
    import torch.nn as nn
    if isinstance(params, nn.Module):
        return [t for t in params.state_dict().values()
                if isinstance(t, torch.Tensor)]
    if isinstance(params, dict):
        return [t for t in params.values() if isinstance(t, torch.Tensor)]
    out = []
    for item in params:
        if isinstance(item, tuple):
            item = item[1]
        if isinstance(item, torch.Tensor):
            out.append(item)
    return out
