"""Worker entry point: ``python -m sparkdl.engine.worker``.

Per-rank process body for the gang launcher (SURVEY.md §3.2's "per task:
unpickle, set up rank/size, invoke main(**kwargs)"; reference
README.md:70,92-93): loads the cloudpickled ``(main, kwargs)`` payload,
runs it, and — on rank 0 — ships the return value back to the driver over
the log socket.

``sparkdl.torch.init()`` inside user ``main`` picks up RANK / WORLD_SIZE /
LOCAL_RANK / MASTER_* from the environment set by the launcher.
"""

import os
import sys
import traceback

import cloudpickle

from sparkdl.engine import logsink


def main():
    rank = int(os.environ["RANK"])
    payload_path = os.environ["SPARKDL_PAYLOAD"]
    with open(payload_path, "rb") as f:
        fn, kwargs = cloudpickle.load(f)

    try:
        result = fn(**kwargs)
    except Exception:
        traceback.print_exc()
        sys.stdout.flush()
        sys.stderr.flush()
        sys.exit(1)
    finally:
        # If user main initialized torch.distributed, tear it down cleanly
        # (final barrier + destroy) so RCCL communicators don't leak
        # across the gang teardown.
        try:
            from sparkdl.parallel import comm
            comm.shutdown()
        except Exception:
            pass

    if rank == 0:
        logsink.send_return_value(cloudpickle.dumps(result))
    sys.stdout.flush()
    sys.exit(0)


if __name__ == "__main__":
    main()
