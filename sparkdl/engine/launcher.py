"""Gang launcher: spawn, babysit, and harvest one process per rank.

Native replacement for the documented Databricks launch path (reference
runner_base.py:48-95, README.md:43-61):

  - resolves ``np`` to a world size (np>0 → one rank per GPU; np<-1 →
    ``-np`` driver-local subprocesses; np==0 → all available slots,
    deprecated),
  - cloudpickles ``(main, kwargs)`` to a payload file (reference
    README.md:70),
  - spawns ``python -m sparkdl.engine.worker`` per rank with rendezvous
    env, streams rank stdout/stderr according to ``driver_log_verbosity``
    (reference runner_base.py:62-72) and always mirrors them into a
    per-run log directory,
  - watchdog: if any rank exits nonzero the whole gang is torn down and
    the failing rank's log tail is raised to the caller (gang semantics,
    SURVEY.md §5.3),
  - returns rank-0's cloudpickled return value (reference README.md:92-93).
"""

import logging
import os
import subprocess
import sys
import tempfile
import threading
import time

import cloudpickle

from sparkdl.engine import logsink, rendezvous

logger = logging.getLogger("HorovodRunner")

# How long to wait, after rank 0 exits cleanly, for its return-value frame.
_RETURN_GRACE_S = 10.0
_POLL_S = 0.1
_LOG_TAIL_LINES = 50


def _gpu_count():
    try:
        import torch
        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return 0


def resolve_world_size(np_):
    """Map the reference ``np`` contract onto this node.

    Returns (world_size, use_gpu).
    """
    ngpus = _gpu_count()
    if np_ > 0:
        if ngpus == 0:
            raise RuntimeError(
                "HorovodRunner(np=%d) requires %d GPUs but none are "
                "visible; the job fails (reference README.md:53). Use "
                "np<0 for CPU-local runs." % (np_, np_))
        if np_ > ngpus:
            raise RuntimeError(
                "HorovodRunner(np=%d) exceeds the %d visible GPUs on this "
                "node; the job fails (reference README.md:53)."
                % (np_, ngpus))
        return np_, True
    if np_ == 0:
        logger.warning(
            "np=0 (use all task slots) is deprecated (reference "
            "README.md:57-61); using all %s." %
            ("%d GPUs" % ngpus if ngpus else "CPU cores (capped at 8)"))
        if ngpus:
            return ngpus, True
        return min(os.cpu_count() or 1, 8), False
    # np < -1 (np == -1 is handled in-process by HorovodRunner.run).
    # Local mode is a debug mode (reference README.md:43-48); if more
    # subprocesses are requested than GPUs exist, pinning two RCCL ranks
    # to one device is a known hang risk — fall back to CPU/gloo ranks
    # instead of oversubscribing devices.
    world = -np_
    if ngpus and world > ngpus:
        logger.warning(
            "np=%d requests %d local ranks but only %d GPUs are "
            "visible; running the gang on CPU (gloo) to avoid "
            "oversubscribing devices. Use np>=-%d for GPU ranks."
            % (np_, world, ngpus, ngpus))
        return world, False
    return world, ngpus > 0


def launch_gang(main, kwargs, *, np, driver_log_verbosity, timeout=None):
    world_size, use_gpu = resolve_world_size(np)
    timeout = timeout or float(os.environ.get("SPARKDL_TIMEOUT", "0")) or None

    run_dir = tempfile.mkdtemp(prefix="sparkdl_run_")
    payload_path = os.path.join(run_dir, "payload.pkl")
    with open(payload_path, "wb") as f:
        cloudpickle.dump((main, kwargs), f)

    log_server = logsink.LogServer().start()
    master_port = rendezvous.free_port()

    procs = []
    log_files = []
    drains = []
    # Local mode (np<0) always streams rank output to the driver
    # (reference README.md:44-47: "stdout and stderr messages go to the
    # notebook cell output... useful for debugging"); the verbosity gate
    # applies to the np>0 worker mode (reference runner_base.py:62-72).
    stream_all = driver_log_verbosity == "all" or np < 0
    try:
        for rank in range(world_size):
            env = rendezvous.rank_env(
                rank, world_size, master_port, log_server.addr,
                payload_path, use_gpu, driver_log_verbosity)
            log_path = os.path.join(run_dir, "rank%d.log" % rank)
            log_files.append(log_path)
            p = subprocess.Popen(
                [sys.executable, "-u", "-m", "sparkdl.engine.worker"],
                env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                start_new_session=True)
            procs.append(p)
            t = threading.Thread(
                target=_drain, args=(p.stdout, rank, log_path, stream_all),
                daemon=True)
            t.start()
            drains.append(t)
        logger.info("Launched %d ranks (%s); logs in %s",
                    world_size, "GPU" if use_gpu else "CPU", run_dir)

        _watch(procs, log_files, timeout)
    finally:
        for p in procs:
            if p.poll() is None:
                p.terminate()
        for p in procs:
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait()
        for t in drains:
            t.join(timeout=2.0)
        log_server.close()

    deadline = time.time() + _RETURN_GRACE_S
    while log_server.return_value_bytes is None and time.time() < deadline:
        time.sleep(0.01)
    if log_server.return_value_bytes is None:
        raise RuntimeError(
            "Rank 0 exited without sending a return value; logs in %s"
            % run_dir)
    return cloudpickle.loads(log_server.return_value_bytes)


def _drain(pipe, rank, log_path, stream_all):
    """Tee a rank's combined stdout/stderr to its log file (always) and to
    the driver's stdout when driver_log_verbosity == 'all'."""
    with open(log_path, "w") as lf:
        for raw in iter(pipe.readline, b""):
            line = raw.decode("utf-8", "replace")
            lf.write(line)
            lf.flush()
            if stream_all:
                sys.stdout.write("[rank %d] %s" % (rank, line))
                sys.stdout.flush()
    pipe.close()


def _watch(procs, log_files, timeout):
    """Babysit the gang: fail fast if any rank dies, kill all on timeout."""
    start = time.time()
    while True:
        codes = [p.poll() for p in procs]
        for rank, code in enumerate(codes):
            if code is not None and code != 0:
                for p in procs:
                    if p.poll() is None:
                        p.terminate()
                raise RuntimeError(
                    "Rank %d failed with exit code %d.\n--- rank %d log "
                    "tail ---\n%s" % (rank, code, rank,
                                      _tail(log_files[rank])))
        if all(code == 0 for code in codes):
            return
        if timeout and time.time() - start > timeout:
            for p in procs:
                if p.poll() is None:
                    p.terminate()
            raise RuntimeError(
                "HorovodRunner job timed out after %.0f s" % timeout)
        time.sleep(_POLL_S)


def _tail(path, n=_LOG_TAIL_LINES):
    try:
        with open(path) as f:
            return "".join(f.readlines()[-n:])
    except OSError:
        return "<no log captured>"
