"""LogCallback for PyTorch training loops.

PyTorch-native equivalent of the reference's Keras ``LogCallback``
contract (reference sparkdl/horovod/tensorflow/keras.py:16-34): streams
epoch/batch event logs from the first worker to the driver output via
:func:`sparkdl.horovod.log_to_driver` (4000-char truncation applies).
"""

import time

from sparkdl.horovod import log_to_driver


class LogCallback:
    """
    A simple HorovodRunner log callback that streams event logs to the
    driver output.  Call the hooks from your training loop::

        cb = LogCallback(per_batch_log=False)
        for epoch in range(E):
            cb.on_epoch_begin(epoch)
            for batch, data in enumerate(loader):
                ...
                cb.on_batch_end(batch, {"loss": loss.item()})
            cb.on_epoch_end(epoch, {"loss": epoch_loss})

    Only rank 0 should drive the callback (matching "use log callback in
    the first worker process", reference runner_base.py:70-72).
    """

    def __init__(self, per_batch_log=False):
        """
        :param per_batch_log: emit a log line after every batch as well as every epoch (off by default).
        """
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
