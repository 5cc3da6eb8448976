"""sparkdl.engine — local rendezvous, gang launcher, and log transport.

This is the MI355X-native replacement for the Spark-barrier launch path
the reference documents (reference README.md:43-61, runner_base.py:48-95):
a driver-local gang launcher that spawns one process per rank (one rank
per GPU for np>0), ships ``main`` via cloudpickle, streams worker logs to
the driver, and returns rank-0's return value.
"""
