"""Checkpoint/resume helpers.

DL checkpointing is user-land in the reference contract; these helpers
make the common data-parallel pattern one call: rank 0 writes model +
optimizer + RNG state atomically; every rank can restore, with
parameters re-broadcast so late joiners are consistent.
"""

import os
import tempfile

import torch


def save_checkpoint(path, model, optimizer=None, step=0, extra=None):
    """Rank-0-only atomic checkpoint write."""
    from sparkdl.parallel import comm
    if comm.rank() != 0:
        comm.barrier()
        return
    state = {
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer else None,
        "step": step,
        "rng": torch.get_rng_state(),
        "cuda_rng": (torch.cuda.get_rng_state_all()
                     if torch.cuda.is_available() else None),
        "extra": extra,
    }
    d = os.path.dirname(os.path.abspath(path)) or "."
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    os.close(fd)
    torch.save(state, tmp)
    os.replace(tmp, path)
    comm.barrier()


def load_checkpoint(path, model, optimizer=None, map_location="cpu"):
    """Restore model/optimizer on every rank; parameters are broadcast
    from rank 0 afterwards so all ranks are bit-identical."""
    from sparkdl.parallel import comm
    from sparkdl.parallel.distributed_optimizer import (
        broadcast_parameters)
    state = torch.load(path, map_location=map_location,
                       weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer") is not None:
        optimizer.load_state_dict(state["optimizer"])
    if comm.size() > 1:
        broadcast_parameters(model, root_rank=0)
    return state.get("step", 0), state.get("extra")
