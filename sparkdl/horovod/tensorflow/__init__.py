"""Namespace package mirroring the reference layout
(reference sparkdl/horovod/tensorflow/__init__.py)."""
