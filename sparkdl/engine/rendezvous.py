"""Local rendezvous for the gang launcher.

Replaces the reference's documented Spark-barrier rendezvous (reference
README.md:49-56): every rank gets RANK/WORLD_SIZE/LOCAL_RANK plus a
MASTER_ADDR/MASTER_PORT on 127.0.0.1 so that ``torch.distributed``
(RCCL on GPU, gloo on CPU) can bootstrap its communicator.  The RCCL
unique-id exchange itself happens over torch.distributed's TCP store at
this address.
"""

import os
import socket


MASTER_ADDR = "127.0.0.1"


def free_port():
    """Reserve an ephemeral TCP port on loopback and return it."""
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    s.bind((MASTER_ADDR, 0))
    port = s.getsockname()[1]
    s.close()
    return port


def rank_env(rank, world_size, master_port, log_addr, payload_path,
             use_gpu, driver_log_verbosity):
    """Build the environment for one worker rank.

    One process per GPU: LOCAL_RANK selects the device
    (``sparkdl.torch.init()`` calls ``torch.cuda.set_device``), all GPUs
    stay visible so RCCL can route point-to-point over xGMI.
    """
    env = dict(os.environ)
    env.update({
        "RANK": str(rank),
        "WORLD_SIZE": str(world_size),
        "LOCAL_RANK": str(rank),
        "LOCAL_WORLD_SIZE": str(world_size),
        "MASTER_ADDR": MASTER_ADDR,
        "MASTER_PORT": str(master_port),
        "SPARKDL_LOG_ADDR": log_addr,
        "SPARKDL_PAYLOAD": payload_path,
        "SPARKDL_USE_GPU": "1" if use_gpu else "0",
        "SPARKDL_DRIVER_LOG_VERBOSITY": driver_log_verbosity,
        # dmabuf IPC is required for RCCL / cross-process CUDA tensors on
        # this host driver stack.
        "HSA_ENABLE_IPC_MODE_LEGACY":
            env.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"),
    })
    return env
