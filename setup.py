#!/usr/bin/env python3
"""Packaging for sparkdl (MI355X-native).

Mirrors the reference packaging surface (reference setup.py:1-45: name
'sparkdl', tests excluded) and adds the in-tree HIP extension build for
gfx950.  Build kernels with:

    python setup.py build_ext --inplace

(hipcc cross-compiles gfx950 without a GPU; PYTORCH_ROCM_ARCH=gfx950.)
"""

import os
from setuptools import setup, find_packages

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ext_modules = []
cmdclass = {}
try:
    from sparkdl.ops.build import make_extensions, make_build_ext
    ext_modules = make_extensions()
    cmdclass = {"build_ext": make_build_ext()}
except Exception:
    pass

setup(
    name='sparkdl',
    version='2.2.0-db1',
    packages=find_packages(exclude=['tests', 'tests.*']),
    description='MI355X-native distributed deep learning framework '
                '(HorovodRunner-compatible API)',
    license='Apache 2.0',
    ext_modules=ext_modules,
    cmdclass=cmdclass,
)
