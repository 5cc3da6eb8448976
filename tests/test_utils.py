"""Tests for log transport details, callbacks, timers, checkpointing."""

import os
import unittest

import torch

from sparkdl import HorovodRunner
from sparkdl.horovod import log_to_driver, MAX_LOG_MESSAGE_LEN


class LogTransportTestCase(unittest.TestCase):
    def test_truncation_local(self, capsys=None):
        # outside a run: prints locally, truncated at 4000 chars
        # (reference sparkdl/horovod/__init__.py:21-24)
        import io
        import contextlib
        buf = io.StringIO()
        with contextlib.redirect_stdout(buf):
            log_to_driver("x" * 5000)
        out = buf.getvalue().rstrip("\n")
        self.assertEqual(len(out), MAX_LOG_MESSAGE_LEN)

    def test_type_check(self):
        with self.assertRaises(TypeError):
            log_to_driver(12345)

    def test_log_callback(self):
        import io
        import contextlib
        from sparkdl.torch import LogCallback
        cb = LogCallback(per_batch_log=True)
        buf = io.StringIO()
        with contextlib.redirect_stdout(buf):
            cb.on_epoch_begin(0)
            cb.on_batch_end(3, {"loss": 1.25})
            cb.on_epoch_end(0, {"loss": 1.0})
        out = buf.getvalue()
        self.assertIn("Epoch 0 begin", out)
        self.assertIn("Batch 3 end loss=1.25", out)
        self.assertIn("loss=1", out)

    def test_keras_callback_gated_on_tf(self):
        try:
            import tensorflow  # noqa: F401
            has_tf = True
        except ImportError:
            has_tf = False
        if not has_tf:
            with self.assertRaises(ImportError):
                import sparkdl.horovod.tensorflow.keras  # noqa: F401


class TimerTestCase(unittest.TestCase):
    def test_step_timer(self):
        from sparkdl.utils import StepTimer
        t = StepTimer(sync_cuda=False)
        for _ in range(5):
            with t:
                sum(range(1000))
        s = t.summary()
        self.assertEqual(s["steps"], 5)
        self.assertGreater(s["mean_ms"], 0)


def _comm_timer_main():
    import torch as _t
    import sparkdl.torch as hvd
    from sparkdl.utils.profiling import CommTimer
    hvd.init()
    rec = CommTimer().allreduce(_t.randn(1024), iters=3, warmup=1)
    return rec["world"], rec["bus_GBps"] > 0


class CommTimerTestCase(unittest.TestCase):
    def test_comm_timer_gloo(self):
        hr = HorovodRunner(np=-2)
        world, ok = hr.run(_comm_timer_main)
        self.assertEqual(world, 2)
        self.assertTrue(ok)


def _ckpt_main(path):
    import torch as _t
    import sparkdl.torch as hvd
    from sparkdl.utils import save_checkpoint, load_checkpoint
    hvd.init()
    model = _t.nn.Linear(4, 2)
    opt = _t.optim.SGD(model.parameters(), lr=0.1)
    _t.manual_seed(42 + hvd.rank())  # desync on purpose
    with _t.no_grad():
        model.weight.add_(hvd.rank())
    save_checkpoint(path, model, opt, step=7)
    model2 = _t.nn.Linear(4, 2)
    step, _ = load_checkpoint(path, model2)
    # all ranks end with rank0's weights
    return step, float(model2.weight.sum())


class CheckpointTestCase(unittest.TestCase):
    def test_checkpoint_roundtrip_distributed(self):
        import tempfile
        with tempfile.TemporaryDirectory() as d:
            path = os.path.join(d, "ck.pt")
            hr = HorovodRunner(np=-2)
            step, wsum = hr.run(_ckpt_main, path=path)
            self.assertEqual(step, 7)

    def test_checkpoint_single(self):
        import tempfile
        from sparkdl.utils import save_checkpoint, load_checkpoint
        model = torch.nn.Linear(3, 3)
        with tempfile.TemporaryDirectory() as d:
            path = os.path.join(d, "ck.pt")
            save_checkpoint(path, model, step=3)
            m2 = torch.nn.Linear(3, 3)
            step, _ = load_checkpoint(path, m2)
            self.assertEqual(step, 3)
            self.assertTrue(torch.equal(m2.weight, model.weight))


if __name__ == "__main__":
    unittest.main()
