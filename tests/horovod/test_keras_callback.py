"""Exercise the Keras LogCallback implementation against a stub
tensorflow module (TF itself is an optional dependency)."""

import sys
import types
import unittest


class KerasCallbackStubTestCase(unittest.TestCase):
    def test_callback_with_stub_tf(self):
        if "tensorflow" in sys.modules and not isinstance(
                sys.modules["tensorflow"], types.ModuleType):
            self.skipTest("real tensorflow importable")
        try:
            import tensorflow  # noqa: F401
            self.skipTest("real tensorflow importable")
        except ImportError:
            pass

        tf = types.ModuleType("tensorflow")
        keras = types.ModuleType("tensorflow.keras")
        callbacks = types.ModuleType("tensorflow.keras.callbacks")

        class Callback:
            pass

        callbacks.Callback = Callback
        keras.callbacks = callbacks
        tf.keras = keras
        sys.modules["tensorflow"] = tf
        sys.modules["tensorflow.keras"] = keras
        sys.modules["tensorflow.keras.callbacks"] = callbacks
        try:
            sys.modules.pop("sparkdl.horovod.tensorflow.keras", None)
            import io
            import contextlib
            from sparkdl.horovod.tensorflow.keras import LogCallback
            cb = LogCallback(per_batch_log=True)
            buf = io.StringIO()
            with contextlib.redirect_stdout(buf):
                cb.on_epoch_begin(1)
                cb.on_batch_end(2, {"loss": 0.5})
                cb.on_epoch_end(1, {"loss": 0.25})
            out = buf.getvalue()
            self.assertIn("Epoch 1 begin", out)
            self.assertIn("Batch 2 end loss=0.5", out)
            self.assertIn("Epoch 1 end", out)
        finally:
            for m in ("tensorflow", "tensorflow.keras",
                      "tensorflow.keras.callbacks",
                      "sparkdl.horovod.tensorflow.keras"):
                sys.modules.pop(m, None)


if __name__ == "__main__":
    unittest.main()
