"""GPU tests: HIP histogram kernel vs numpy reference; GPU-trained
booster matches quality of the CPU build."""

import numpy as np
import pytest
import torch

from sparkdl.xgboost import gbt

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")


def test_histogram_matches_numpy():
    rng = np.random.RandomState(0)
    n, F, n_nodes = 5000, 13, 4
    B = rng.randint(0, 256, size=(n, F)).astype(np.uint8)
    g = rng.randn(n)
    h = rng.rand(n) + 0.5
    node = rng.randint(0, n_nodes, size=n).astype(np.int32)

    ref = gbt.CpuHistogramBuilder(B).build(g, h, node, n_nodes)
    got = gbt.GpuHistogramBuilder(B).build(g, h, node, n_nodes)
    assert np.allclose(got, ref, atol=1e-3, rtol=1e-5), \
        np.abs(got - ref).max()


def test_gpu_training_quality():
    rng = np.random.RandomState(1)
    X = rng.rand(4000, 10)
    y = 2 * X[:, 0] + X[:, 1] * X[:, 2] + 0.05 * rng.randn(4000)
    p = {"n_estimators": 30, "max_depth": 5}
    b_cpu = gbt.train(X, y, p)
    b_gpu = gbt.train(X, y, p, use_gpu=True)
    mse_cpu = float(np.mean((b_cpu.predict(X) - y) ** 2))
    mse_gpu = float(np.mean((b_gpu.predict(X) - y) ** 2))
    assert mse_gpu < 1.5 * mse_cpu + 1e-6, (mse_cpu, mse_gpu)


def test_gpu_histogram_speed():
    import time
    rng = np.random.RandomState(2)
    n, F = 2_000_000, 32
    B = rng.randint(0, 255, size=(n, F)).astype(np.uint8)
    g = rng.randn(n)
    h = np.ones(n)
    node = np.zeros(n, dtype=np.int32)

    cpu = gbt.CpuHistogramBuilder(B)
    t0 = time.perf_counter()
    cpu.build(g, h, node, 1)
    t_cpu = time.perf_counter() - t0

    gpu = gbt.GpuHistogramBuilder(B)
    gpu.build(g, h, node, 1)  # warm
    t0 = time.perf_counter()
    gpu.build(g, h, node, 1)
    t_gpu = time.perf_counter() - t0
    print("hist cpu %.3fs gpu %.3fs" % (t_cpu, t_gpu))
    assert t_gpu < t_cpu, (t_cpu, t_gpu)
