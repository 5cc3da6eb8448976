"""Driver-observable bench smokes for BASELINE configs 4 (BERT) and 5
(GBT): `pytest -m gpu` exercises the same bench.py entry the driver's
round-end BENCH run uses, validating the JSON contract on hardware
(VERDICT round-1 item 10)."""

import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(autouse=True)
def _require_gpu():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")


def _run_bench(extra, env=None, timeout=420):
    e = dict(os.environ)
    if env:
        e.update(env)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + extra,
        cwd=REPO, env=e, timeout=timeout, capture_output=True, text=True)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    return json.loads(line)


@pytest.mark.timeout(600)
def test_bench_bert_smoke():
    rec = _run_bench(["--model", "bert", "--steps", "2", "--warmup", "1"])
    assert rec["metric"] == "sequences/sec"
    assert rec["config"]["model"] == "bert-base"
    assert rec["config"]["seq_len"] == 512
    assert rec["dtype"] == "bf16" and rec["value"] > 0


@pytest.mark.timeout(600)
def test_bench_gbt_smoke():
    rec = _run_bench(["--model", "gbt", "--steps", "3", "--warmup", "1"],
                     env={"SPARKDL_GBT_ROWS": "400000"})
    assert rec["metric"] == "boost_rounds/sec"
    assert rec["config"]["features"] == 64
    assert rec["value"] > 0
