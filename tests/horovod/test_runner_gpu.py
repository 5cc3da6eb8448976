"""GPU gang tests: HorovodRunner np>0 (one process per GPU, RCCL)."""

import pytest
import torch

from sparkdl import HorovodRunner

pytestmark = pytest.mark.gpu


def _gpu_main():
    import torch as _t
    import sparkdl.torch as hvd
    hvd.init()
    assert _t.cuda.is_available()
    x = _t.ones(8, device="cuda") * (hvd.rank() + 1)
    s = hvd.allreduce(x, average=False)
    _t.cuda.synchronize()
    return {"rank": hvd.rank(), "size": hvd.size(),
            "sum0": float(s[0]), "device": _t.cuda.current_device()}


@pytest.fixture(autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")


def test_np_positive_single_gpu():
    hr = HorovodRunner(np=1)
    out = hr.run(_gpu_main)
    assert out["rank"] == 0 and out["size"] == 1
    assert out["sum0"] == 1.0


def test_np_exceeding_gpus_fails():
    n = torch.cuda.device_count()
    hr = HorovodRunner(np=n + 1)
    with pytest.raises(RuntimeError):
        hr.run(_gpu_main)


def test_mlp_train_np1_gpu():
    from sparkdl.models.mlp import train_step_fn
    hr = HorovodRunner(np=1)
    losses = hr.run(train_step_fn, seed=0, steps=4, batch=64,
                    device="cuda")
    assert losses[-1] < losses[0]
