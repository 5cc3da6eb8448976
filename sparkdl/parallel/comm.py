"""Process-group bootstrap and collective wrappers.

On GPU ranks ``torch.distributed`` with backend ``"nccl"`` IS RCCL on
ROCm — collectives route over the node's xGMI point-to-point fabric
(7 links x ~153 GB/s per GPU).  CPU ranks (HorovodRunner np<-1 on a
GPU-less driver) use gloo, which keeps the whole data-parallel path
testable without hardware.
"""

import datetime
import os

import torch
import torch.distributed as dist


def _env_int(name, default):
    return int(os.environ.get(name, default))


def init_process_group(timeout_s=300):
    """Idempotent rendezvous from the launcher's environment.

    Reads RANK / WORLD_SIZE / LOCAL_RANK / MASTER_ADDR / MASTER_PORT (set
    by sparkdl.engine.rendezvous, or by torchrun) and creates the global
    process group.  On GPU boxes the device is pinned to LOCAL_RANK first
    so RCCL ring construction sees one device per rank.
    """
    if dist.is_initialized():
        return
    # The launcher sets SPARKDL_USE_GPU=0 when it decided the gang must
    # run on CPU (e.g. np<-1 asked for more ranks than GPUs exist —
    # pinning two RCCL ranks to one device can hang). Honor that here.
    use_gpu = (torch.cuda.is_available()
               and os.environ.get("SPARKDL_USE_GPU", "1") != "0")
    lr = _env_int("LOCAL_RANK", 0)
    if use_gpu:
        torch.cuda.set_device(lr % torch.cuda.device_count())
    backend = "nccl" if use_gpu else "gloo"
    kwargs = {}
    if use_gpu:
        kwargs["device_id"] = torch.device(
            "cuda", lr % torch.cuda.device_count())
    dist.init_process_group(
        backend=backend,
        timeout=datetime.timedelta(seconds=timeout_s),
        **kwargs)


def is_initialized():
    return dist.is_available() and dist.is_initialized()


def rank():
    return dist.get_rank() if is_initialized() else 0


def size():
    return dist.get_world_size() if is_initialized() else 1


def local_rank():
    return _env_int("LOCAL_RANK", 0)


def local_size():
    return _env_int("LOCAL_WORLD_SIZE", size())


def allreduce_(tensor, average=True, async_op=False):
    """In-place sum (or mean) all-reduce. Returns the Work handle when
    async_op, else the tensor."""
    if not is_initialized() or size() == 1:
        return None if async_op else tensor
    work = dist.all_reduce(tensor, op=dist.ReduceOp.SUM, async_op=async_op)
    if async_op:
        # Caller divides after wait() when averaging.
        return work
    if average:
        tensor.div_(size())
    return tensor


def broadcast_(tensor, root_rank=0):
    if is_initialized() and size() > 1:
        dist.broadcast(tensor, src=root_rank)
    return tensor


def barrier():
    if is_initialized() and size() > 1:
        dist.barrier()


def shutdown():
    """Tear down the process group. A final barrier keeps fast ranks
    from destroying their communicator while peers still have
    collectives in flight (RCCL hang risk on teardown); any error here
    must never mask the job's real result."""
    if not is_initialized():
        return
    try:
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dist.barrier()
    except Exception:
        pass
    try:
        dist.destroy_process_group()
    except Exception:
        pass
