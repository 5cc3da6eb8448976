"""Horovod-compatible namespace: worker→driver logging.

Implements the contract the reference only documents
(reference sparkdl/horovod/__init__.py:20-28): ``log_to_driver(message)``
sends a string to the driver, which prints it to stdout; messages longer
than 4000 characters are truncated.
"""

MAX_LOG_MESSAGE_LEN = 4000  # reference sparkdl/horovod/__init__.py:23-24


def log_to_driver(message):
    """
    Send a log message (string type) to driver side, and driver will print
    log to stdout.  If message length is greater than 4000, it will be
    truncated.

    Inside a :class:`sparkdl.HorovodRunner` worker this forwards the
    message over the run's log socket; outside a run (or in ``np == -1``
    in-process mode, where the worker *is* the driver) it prints directly.
    """
    if not isinstance(message, str):
        raise TypeError("log_to_driver expects a str, got %s" % type(message))
    message = message[:MAX_LOG_MESSAGE_LEN]
    from sparkdl.engine.logsink import forward_to_driver
    forward_to_driver(message)


__all__ = ['log_to_driver']
