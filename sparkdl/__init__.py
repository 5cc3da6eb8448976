"""sparkdl — MI355X-native distributed deep learning framework.

A from-scratch re-implementation of the capabilities documented by
databricks/spark-deep-learning (the reference is an API shell; see
reference README.md:10-11).  The public API is a drop-in for the
reference: ``sparkdl.HorovodRunner`` keeps the exact signature locked by
the reference's oracle test (reference
tests/horovod/runner_base_test.py:26-37), but ``run`` actually launches
one rank per MI355X GPU (np>0) or driver-local subprocesses (np<-1),
with gradient all-reduce over RCCL/xGMI and hand-written CDNA4 HIP
kernels on the training hot path.

Reference parity map:
  - ``HorovodRunner``        → reference sparkdl/horovod/runner_base.py:23-103
  - ``sparkdl.horovod.log_to_driver`` → reference sparkdl/horovod/__init__.py:20-28
  - ``sparkdl.xgboost``      → reference sparkdl/xgboost/xgboost.py
"""

from sparkdl.horovod.runner_base import HorovodRunner

__all__ = ['HorovodRunner']

__version__ = '2.2.0-db1'
