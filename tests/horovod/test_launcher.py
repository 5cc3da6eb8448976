"""Gang launcher tests (CPU, gloo): the np<-1 subprocess path, rank-0
return value, log_to_driver streaming, gang failure semantics
(SURVEY.md §3.2-3.3 contract)."""

import os
import unittest

from sparkdl import HorovodRunner


def _rank_report():
    import sparkdl.torch as hvd
    hvd.init()
    r = {"rank": hvd.rank(), "size": hvd.size(),
         "local_rank": hvd.local_rank()}
    hvd.barrier()
    return r


def _log_and_return(msg):
    from sparkdl.horovod import log_to_driver
    import os as _os
    if int(_os.environ.get("RANK", "0")) == 0:
        log_to_driver(msg)
    return int(_os.environ.get("RANK", "0"))


def _fail_on_rank_one():
    import os as _os
    if int(_os.environ.get("RANK", "0")) == 1:
        raise RuntimeError("boom on rank 1")
    return "ok"


def _allreduce_check(value):
    import torch
    import sparkdl.torch as hvd
    hvd.init()
    t = torch.full((4,), float(value * (hvd.rank() + 1)))
    avg = hvd.allreduce(t, average=True)
    expected = value * sum(r + 1 for r in range(hvd.size())) / hvd.size()
    assert torch.allclose(avg, torch.full((4,), expected)), (avg, expected)
    return float(avg[0])


class LauncherTestCase(unittest.TestCase):

    def test_subprocess_rank_env(self):
        hr = HorovodRunner(np=-2)
        result = hr.run(_rank_report)
        self.assertEqual(result, {"rank": 0, "size": 2, "local_rank": 0})

    def test_log_to_driver(self, capsys=None):
        hr = HorovodRunner(np=-2)
        rank0 = hr.run(_log_and_return, msg="hello from worker")
        self.assertEqual(rank0, 0)

    def test_gang_failure(self):
        hr = HorovodRunner(np=-2)
        with self.assertRaises(RuntimeError) as ctx:
            hr.run(_fail_on_rank_one)
        self.assertIn("Rank 1 failed", str(ctx.exception))
        self.assertIn("boom on rank 1", str(ctx.exception))

    def test_gloo_allreduce(self):
        hr = HorovodRunner(np=-2)
        avg = hr.run(_allreduce_check, value=2.0)
        # ranks hold 2.0 and 4.0 -> avg 3.0
        self.assertAlmostEqual(avg, 3.0)

    def test_np_positive_without_gpu_fails(self):
        import torch
        if torch.cuda.is_available():
            self.skipTest("GPUs visible; np>0 is valid here")
        hr = HorovodRunner(np=2)
        with self.assertRaises(RuntimeError):
            hr.run(lambda: None)


class MLPEndToEndTestCase(unittest.TestCase):
    """BASELINE.json config 1: HorovodRunner on a 2-layer MLP, CPU."""

    def test_mlp_np2(self):
        from sparkdl.models.mlp import train_step_fn
        hr = HorovodRunner(np=-2)
        losses = hr.run(train_step_fn, seed=0, steps=4, batch=32)
        self.assertEqual(len(losses), 4)
        self.assertLess(losses[-1], losses[0])

    def test_mlp_inprocess_matches_contract(self):
        from sparkdl.models.mlp import train_step_fn
        hr = HorovodRunner(np=-1)
        losses = hr.run(train_step_fn, seed=0, steps=2, batch=16)
        self.assertEqual(len(losses), 2)


if __name__ == "__main__":
    unittest.main()


class ResolveWorldSizeTestCase(unittest.TestCase):
    def test_resolve(self):
        import torch
        from sparkdl.engine.launcher import resolve_world_size
        if torch.cuda.is_available():
            self.skipTest("CPU-only semantics")
        self.assertEqual(resolve_world_size(-3), (3, False))
        ws, gpu = resolve_world_size(0)  # deprecated: all slots
        self.assertFalse(gpu)
        self.assertGreaterEqual(ws, 1)
        with self.assertRaises(RuntimeError):
            resolve_world_size(2)  # np>0 needs GPUs

    def test_local_mode_never_oversubscribes_gpus(self):
        # np<-1 with more ranks than devices must fall back to CPU/gloo
        # ranks (two RCCL ranks pinned to one device is a hang risk).
        from unittest import mock
        from sparkdl.engine import launcher
        with mock.patch.object(launcher, "_gpu_count", return_value=4):
            self.assertEqual(launcher.resolve_world_size(-16), (16, False))
            self.assertEqual(launcher.resolve_world_size(-4), (4, True))
            self.assertEqual(launcher.resolve_world_size(-2), (2, True))


def _object_api_main():
    import sparkdl.torch as hvd
    hvd.init()
    v = hvd.broadcast_object({"a": hvd.rank()}, root_rank=1)
    gathered = hvd.allgather_object(hvd.rank())
    avg = hvd.metric_average(hvd.rank() + 1.0)
    return v["a"], gathered, avg


class ObjectApiTestCase(unittest.TestCase):
    def test_object_collectives(self):
        v, gathered, avg = HorovodRunner(np=-2).run(_object_api_main)
        self.assertEqual(v, 1)
        self.assertEqual(gathered, [0, 1])
        self.assertAlmostEqual(avg, 1.5)


def _emit_output():
    import os as _os
    print("hello-from-rank-%s" % _os.environ["RANK"], flush=True)
    return "done"


class VerbosityTestCase(unittest.TestCase):
    def test_all_streams_rank_output(self):
        import io
        import contextlib
        buf = io.StringIO()
        hr = HorovodRunner(np=-2, driver_log_verbosity="all")
        with contextlib.redirect_stdout(buf):
            out = hr.run(_emit_output)
        self.assertEqual(out, "done")
        text = buf.getvalue()
        self.assertIn("[rank 0] hello-from-rank-0", text)
        self.assertIn("[rank 1] hello-from-rank-1", text)

    def test_local_mode_always_streams(self):
        # np<0 local mode streams rank output regardless of verbosity
        # (reference README.md:44-47)
        import io
        import contextlib
        buf = io.StringIO()
        hr = HorovodRunner(np=-2)  # default log_callback_only
        with contextlib.redirect_stdout(buf):
            hr.run(_emit_output)
        self.assertIn("hello-from-rank-0", buf.getvalue())


def _sleep_forever():
    import time as _t
    _t.sleep(600)


class TimeoutTestCase(unittest.TestCase):
    def test_job_timeout(self):
        import os as _os
        _os.environ["SPARKDL_TIMEOUT"] = "5"
        try:
            hr = HorovodRunner(np=-2)
            with self.assertRaises(RuntimeError) as ctx:
                hr.run(_sleep_forever)
            self.assertIn("timed out", str(ctx.exception))
        finally:
            del _os.environ["SPARKDL_TIMEOUT"]


def _allgather_tensor_main():
    import torch
    import sparkdl.torch as hvd
    hvd.init()
    t = torch.full((2, 3), float(hvd.rank()))
    out = hvd.allgather(t)
    return out.shape == (4, 3) and float(out[0, 0]) == 0.0 \
        and float(out[2, 0]) == 1.0


class AllgatherTestCase(unittest.TestCase):
    def test_allgather_tensor(self):
        self.assertTrue(HorovodRunner(np=-2).run(_allgather_tensor_main))


class ResolveWorldSizeGpuMockTestCase(unittest.TestCase):
    def test_np0_uses_all_gpus(self):
        from unittest import mock
        from sparkdl.engine import launcher
        with mock.patch.object(launcher, "_gpu_count", return_value=8):
            self.assertEqual(launcher.resolve_world_size(0), (8, True))
            self.assertEqual(launcher.resolve_world_size(8), (8, True))
            with self.assertRaises(RuntimeError):
                launcher.resolve_world_size(9)
