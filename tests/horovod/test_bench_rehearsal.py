"""CPU rehearsal of the driver's multi-rank bench invocation: the
exact torchrun command shape the round-end SCALE sweep uses, with 8
gloo ranks on CPU (VERDICT round-1 item 3: make the 8-GPU sweep
un-failable — the distributed path must be deadlock-free by
construction and exercised without hardware)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


@pytest.mark.timeout(600)
def test_bench_torchrun_8rank_cpu_rehearsal():
    import torch
    if torch.cuda.is_available():
        pytest.skip("CPU rehearsal (GPU boxes run the real sweep)")
    from sparkdl.engine.rendezvous import free_port
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""
    env["HIP_VISIBLE_DEVICES"] = ""
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "8",
           "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()),
           os.path.join(REPO, "bench.py"),
           "--gpus", "8", "--steps", "1", "--warmup", "0",
           "--batch", "2", "--model", "resnet50"]
    out = subprocess.run(cmd, cwd=REPO, env=env, timeout=540,
                         capture_output=True, text=True)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines()
            if ln.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 0  # CPU rehearsal
    assert rec["config"]["parallelism"] == "dp8"
    assert rec["value"] > 0
