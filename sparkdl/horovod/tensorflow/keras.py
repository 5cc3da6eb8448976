"""Keras LogCallback (reference sparkdl/horovod/tensorflow/keras.py:16-34).

TensorFlow is an optional dependency; importing this module without TF
installed raises ImportError, matching the reference's hard
``from tensorflow import keras`` (reference keras.py:9).  The PyTorch
equivalent — the first-class path in this framework — is
:class:`sparkdl.torch.LogCallback`.
"""

import time

from tensorflow import keras

from sparkdl.horovod import log_to_driver

__all__ = ["LogCallback"]


class LogCallback(keras.callbacks.Callback):
    """
    A simple HorovodRunner log callback that streams event logs to the
    driver / notebook cell output.
    """

    def __init__(self, per_batch_log=False):
        """
        :param per_batch_log: emit a log line after every batch as well as every epoch (off by default).
        """
        super().__init__()
        self.per_batch_log = per_batch_log
        self._epoch_start = None

    @staticmethod
    def _fmt(logs):
        if not logs:
            return ""
        return " " + " ".join(
            "%s=%.6g" % (k, v) if isinstance(v, float) else "%s=%s" % (k, v)
            for k, v in sorted(logs.items()))

    def on_epoch_begin(self, epoch, logs=None):
        self._epoch_start = time.time()
        log_to_driver("Epoch %d begin%s" % (epoch, self._fmt(logs)))

    def on_batch_end(self, batch, logs=None):
        if self.per_batch_log:
            log_to_driver("Batch %d end%s" % (batch, self._fmt(logs)))

    def on_epoch_end(self, epoch, logs=None):
        dur = (time.time() - self._epoch_start
               if self._epoch_start is not None else 0.0)
        log_to_driver(
            "Epoch %d end (%.1fs)%s" % (epoch, dur, self._fmt(logs)))
