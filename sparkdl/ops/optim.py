"""Fused optimizers: single-launch multi-tensor AdamW / SGD
(SURVEY.md §2.2 N3).

The whole model's optimizer step is ONE HIP kernel launch: parameters are
chunked into a device-resident work table (TensorChunk structs + a
block→chunk map) built once and reused until any tensor pointer changes.
This works with arbitrary parameter/grad layouts — in particular with
:class:`sparkdl.parallel.DistributedOptimizer`'s flat gradient buckets,
whose views are just more pointers in the table.

fp32 parameters (the autocast master-weight regime for the bf16 benches);
CPU tensors run a plain PyTorch reference path (also the numerics oracle
for the GPU tests).
"""

import struct

import torch

import sparkdl.ops as _ops

_CHUNK = 16384  # must match kOptChunk in csrc/kernels.h


class _MultiTensorTable:
    """Device-resident chunk table for one param group set."""

    def __init__(self, entries, device):
        # entries: list of (p_fp32_or_master, g, m, v, p_bf16_or_None).
        blob = bytearray()
        bmap = []
        for ti, (p, g, m, v, pl) in enumerate(entries):
            n = p.numel()
            blob += struct.pack(
                "<QQQQqQ", p.data_ptr(), g.data_ptr(), m.data_ptr(),
                v.data_ptr() if v is not None else 0, n,
                pl.data_ptr() if pl is not None else 0)
            for start in range(0, n, _CHUNK):
                bmap.append((ti, start))
        self.chunks = torch.frombuffer(
            blob, dtype=torch.uint8).to(device)
        self.bmap = torch.tensor(
            bmap, dtype=torch.int32).flatten().to(device)
        self.nblocks = len(bmap)
        self.key = tuple(
            (p.data_ptr(), g.data_ptr()) for p, g, _, _, _ in entries)


def _entries_key(entries):
    return tuple((p.data_ptr(), g.data_ptr()) for p, g, _, _, _ in entries)


class _FusedOptimizerMixin:
    def zero_grad(self, set_to_none=False):
        """Default to zeroing in place: stable grad pointers keep the
        device chunk table valid and make the step hipGraph-capturable.
        When the chunk tables are current, all grads are zeroed in ONE
        kernel launch instead of a per-tensor fill storm."""
        if not set_to_none and getattr(self, "_tables", None):
            current = []
            for group in self.param_groups:
                for p in group["params"]:
                    if p.grad is not None:
                        current.append((p.data_ptr(), p.grad.data_ptr()))
            table_key = tuple(k for t in self._tables.values()
                              for k in t.key)
            if tuple(current) == table_key:
                for t in self._tables.values():
                    _ops.ext().zero_grads_(t.chunks, t.bmap, t.nblocks)
                return
        super().zero_grad(set_to_none=set_to_none)


class FusedAdamW(_FusedOptimizerMixin, torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._tables = {}
        self._step_state = {}

    def _gather(self, group):
        entries = []
        for p in group["params"]:
            if p.grad is None:
                continue
            if p.grad.is_sparse:
                raise RuntimeError("FusedAdamW does not support sparse grads")
            state = self.state[p]
            lowp = p.dtype == torch.bfloat16
            if not state:
                # bf16 params train against an fp32 master copy; moments
                # and the update run in fp32 (SURVEY.md §7 hard part 4).
                if lowp:
                    state["master"] = p.detach().float()
                f = state.get("master", p)
                state["exp_avg"] = torch.zeros_like(f)
                state["exp_avg_sq"] = torch.zeros_like(f)
            master = state.get("master")
            entries.append((master if lowp else p, p.grad,
                            state["exp_avg"], state["exp_avg_sq"],
                            p if lowp else None))
        return entries

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            entries = self._gather(group)
            if not entries:
                continue
            group.setdefault("step", 0)
            group["step"] += 1
            beta1, beta2 = group["betas"]
            if entries[0][0].is_cuda:
                dev = entries[0][0].device
                gi = self.param_groups.index(group)
                tbl = self._tables.get(gi)
                if tbl is None or tbl.key != _entries_key(entries):
                    tbl = self._tables[gi] = _MultiTensorTable(entries, dev)
                if gi not in self._step_state:
                    # Device-side step counter + bias-correction scratch:
                    # advanced by a prep kernel so hipGraph replays of the
                    # whole step see fresh bias corrections.
                    self._step_state[gi] = (
                        torch.tensor([group["step"] - 1],
                                     dtype=torch.int64, device=dev),
                        torch.empty(2, dtype=torch.float32, device=dev))
                step_gpu, coeffs = self._step_state[gi]
                _ops.ext().fused_adamw_(
                    tbl.chunks, tbl.bmap, tbl.nblocks, group["lr"],
                    beta1, beta2, group["eps"], group["weight_decay"],
                    step_gpu, coeffs)
            else:
                self._ref_step(entries, group)
        return loss

    def _ref_step(self, entries, group):
        beta1, beta2 = group["betas"]
        step = group["step"]
        bc1 = 1 - beta1 ** step
        bc2 = 1 - beta2 ** step
        for p, g, m, v, pl in entries:
            g = g.float() if g.dtype != torch.float32 else g
            m.mul_(beta1).add_(g, alpha=1 - beta1)
            v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
            p.mul_(1 - group["lr"] * group["weight_decay"])
            denom = (v / bc2).sqrt_().add_(group["eps"])
            p.addcdiv_(m, denom, value=-group["lr"] / bc1)
            if pl is not None:
                pl.detach().copy_(p)


class FusedSGD(_FusedOptimizerMixin, torch.optim.Optimizer):
    def __init__(self, params, lr=1e-2, momentum=0.0, weight_decay=0.0,
                 nesterov=False):
        defaults = dict(lr=lr, momentum=momentum,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)
        self._tables = {}

    def _gather(self, group):
        entries = []
        for p in group["params"]:
            if p.grad is None:
                continue
            state = self.state[p]
            if not state:
                state["momentum_buffer"] = torch.zeros_like(p)
            entries.append((p, p.grad, state["momentum_buffer"], None,
                            None))
        return entries

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            entries = self._gather(group)
            if not entries:
                continue
            group.setdefault("step", 0)
            group["step"] += 1
            first = group["step"] == 1 and group["momentum"] != 0.0
            if entries[0][0].is_cuda:
                if entries[0][0].dtype != torch.float32:
                    raise RuntimeError("FusedSGD GPU path requires fp32 params")
                gi = self.param_groups.index(group)
                tbl = self._tables.get(gi)
                if tbl is None or tbl.key != _entries_key(entries):
                    tbl = self._tables[gi] = _MultiTensorTable(
                        entries, entries[0][0].device)
                _ops.ext().fused_sgd_(
                    tbl.chunks, tbl.bmap, tbl.nblocks, group["lr"],
                    group["momentum"], group["weight_decay"],
                    group["nesterov"], first)
            else:
                self._ref_step(entries, group, first)
        return loss

    def _ref_step(self, entries, group, first):
        mom = group["momentum"]
        for p, g, buf, _, _ in entries:
            d = g.add(p, alpha=group["weight_decay"]) \
                if group["weight_decay"] else g
            if mom != 0.0:
                if first:
                    buf.copy_(d)
                else:
                    buf.mul_(mom).add_(d)
                d = d.add(buf, alpha=mom) if group["nesterov"] else buf
            p.add_(d, alpha=-group["lr"])
