"""Unit tests for the pyspark.ml-shaped Params system (sparkdl.ml)."""

import unittest

from sparkdl.ml import Param, Params, TypeConverters
from sparkdl.xgboost import XgboostRegressor


class ParamsTestCase(unittest.TestCase):
    def test_param_identity_and_repr(self):
        p = Param(Params._dummy(), "alpha", "doc", TypeConverters.toFloat)
        self.assertEqual(repr(p), "Param(alpha)")
        q = Param(Params._dummy(), "alpha", "other doc")
        self.assertEqual(p, q)  # name-keyed equality
        self.assertEqual(hash(p), hash(q))

    def test_type_converters(self):
        self.assertEqual(TypeConverters.toInt("3"), 3)
        self.assertEqual(TypeConverters.toFloat("2.5"), 2.5)
        self.assertIs(TypeConverters.toBoolean(True), True)
        with self.assertRaises(TypeError):
            TypeConverters.toBoolean("yes")
        with self.assertRaises(TypeError):
            TypeConverters.toInt(True)

    def test_set_get_default_precedence(self):
        est = XgboostRegressor()
        self.assertEqual(est.getOrDefault("num_workers"), 1)  # default
        self.assertFalse(est.isSet("num_workers"))
        est._set(num_workers=4)
        self.assertTrue(est.isSet("num_workers"))
        self.assertEqual(est.getOrDefault("num_workers"), 4)

    def test_unknown_param_raises(self):
        est = XgboostRegressor()
        with self.assertRaises(AttributeError):
            est.getParam("nope")

    def test_copy_isolated(self):
        est = XgboostRegressor(missing=0.0)
        dup = est.copy()
        dup._set(missing=float("nan"))
        self.assertEqual(est.getOrDefault("missing"), 0.0)

    def test_params_enumeration(self):
        est = XgboostRegressor()
        names = {p.name for p in est.params}
        self.assertIn("missing", names)
        self.assertIn("featuresCol", names)
        self.assertTrue(est.hasParam("use_gpu"))

    def test_typed_param_conversion_on_set(self):
        est = XgboostRegressor(num_workers="2")
        self.assertEqual(est.getOrDefault("num_workers"), 2)


class LogSinkFramingTestCase(unittest.TestCase):
    def test_frames_roundtrip(self):
        import os
        from sparkdl.engine import logsink
        import io
        import contextlib
        import time

        server = logsink.LogServer().start()
        os.environ[logsink.LOG_ADDR_ENV] = server.addr
        try:
            logsink.reset_client()
            buf = io.StringIO()
            with contextlib.redirect_stdout(buf):
                logsink.forward_to_driver("frame one")
                logsink.send_return_value(b"\x01\x02payload")
                deadline = time.time() + 5
                while (server.return_value_bytes is None
                       and time.time() < deadline):
                    time.sleep(0.01)
            self.assertEqual(server.return_value_bytes, b"\x01\x02payload")
        finally:
            del os.environ[logsink.LOG_ADDR_ENV]
            logsink.reset_client()
            server.close()

    def test_forward_without_server_prints(self):
        import io
        import contextlib
        from sparkdl.engine import logsink
        logsink.reset_client()
        buf = io.StringIO()
        with contextlib.redirect_stdout(buf):
            logsink.forward_to_driver("local fallback")
        self.assertIn("local fallback", buf.getvalue())


if __name__ == "__main__":
    unittest.main()


def test_explain_params():
    from sparkdl.xgboost import XgboostRegressor
    est = XgboostRegressor(num_workers=3)
    line = est.explainParam("num_workers")
    assert "num_workers" in line and "current: 3" in line
    text = est.explainParams()
    assert "missing" in text and "baseMarginCol" in text
