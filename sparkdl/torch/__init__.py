"""sparkdl.torch — Horovod-idiom API for user training code.

The L4 contract of the reference (reference runner_base.py:85-95,
README.md:64-93) is user ``main`` functions written against Horovod's
idioms: ``hvd.init()``, ``hvd.rank()/size()/local_rank()``,
``hvd.DistributedOptimizer``, parameter broadcast.  This module provides
those idioms natively for PyTorch-ROCm; typical usage inside ``main``::

    import sparkdl.torch as hvd

    def main():
        hvd.init()                       # rendezvous (RCCL on GPU, gloo on CPU)
        model = ...                      # device is already pinned to local_rank
        opt = hvd.DistributedOptimizer(torch.optim.AdamW(model.parameters()))
        hvd.broadcast_parameters(model, root_rank=0)
        ...

Collectives route through torch.distributed: backend "nccl" IS RCCL on
ROCm, so all-reduce runs over the node's xGMI fabric.
"""

import torch

from sparkdl.parallel import comm
from sparkdl.parallel.distributed_optimizer import (  # noqa: F401
    DistributedOptimizer, broadcast_parameters, broadcast_optimizer_state,
)
from sparkdl.torch.callbacks import LogCallback  # noqa: F401


def init(timeout_s=300):
    """Initialize the process group from the launcher's environment.

    Pins this rank to GPU ``local_rank()`` on GPU boxes.  Safe to call
    when running single-process (np=-1): becomes a no-op with
    rank 0 / size 1 when no rendezvous env is present.
    """
    import os
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        comm.init_process_group(timeout_s=timeout_s)
    elif torch.cuda.is_available():
        torch.cuda.set_device(comm.local_rank() % torch.cuda.device_count())


def shutdown():
    comm.shutdown()


def rank():
    return comm.rank()


def size():
    return comm.size()


def local_rank():
    return comm.local_rank()


def local_size():
    return comm.local_size()


def is_initialized():
    return comm.is_initialized()


def allreduce(tensor, average=True, name=None):
    """Out-of-place average (or sum) all-reduce."""
    out = tensor.clone()
    comm.allreduce_(out, average=average)
    return out


def allreduce_(tensor, average=True, name=None):
    return comm.allreduce_(tensor, average=average)


def broadcast(tensor, root_rank=0, name=None):
    out = tensor.clone()
    comm.broadcast_(out, root_rank=root_rank)
    return out


def broadcast_(tensor, root_rank=0, name=None):
    return comm.broadcast_(tensor, root_rank=root_rank)


def allgather(tensor, name=None):
    """Gather tensors from all ranks, concatenated along dim 0."""
    if comm.size() == 1:
        return tensor.clone()
    import torch.distributed as dist
    outs = [torch.empty_like(tensor) for _ in range(comm.size())]
    dist.all_gather(outs, tensor)
    return torch.cat(outs, dim=0)


def barrier():
    comm.barrier()


def broadcast_object(obj, root_rank=0, name=None):
    """Broadcast an arbitrary picklable object from root_rank."""
    if comm.size() == 1:
        return obj
    import torch.distributed as dist
    lst = [obj if comm.rank() == root_rank else None]
    dist.broadcast_object_list(lst, src=root_rank)
    return lst[0]


def allgather_object(obj, name=None):
    """Gather one picklable object per rank; returns a list of size()."""
    if comm.size() == 1:
        return [obj]
    import torch.distributed as dist
    out = [None] * comm.size()
    dist.all_gather_object(out, obj)
    return out


def metric_average(value, name=None):
    """Average a python scalar across ranks (common Horovod idiom)."""
    t = torch.tensor([float(value)])
    comm.allreduce_(t, average=True)
    return float(t[0])


__all__ = [
    'init', 'shutdown', 'rank', 'size', 'local_rank', 'local_size',
    'is_initialized', 'allreduce', 'allreduce_', 'broadcast', 'broadcast_',
    'allgather', 'barrier', 'broadcast_object', 'allgather_object',
    'metric_average', 'DistributedOptimizer', 'broadcast_parameters',
    'broadcast_optimizer_state', 'LogCallback',
]
