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


def test_gpu_resident_tree_build_quality_and_pred():
    """The device-resident tree build (histograms, split search and row
    partition on GPU) matches CPU quality, and its per-row predictions
    equal the tree's own traversal."""
    rng = np.random.RandomState(3)
    X = rng.rand(20000, 16)
    X[rng.rand(*X.shape) < 0.05] = np.nan  # exercise missing routing
    y = (np.nan_to_num(X[:, 0]) * 2 + np.nan_to_num(X[:, 1])
         + 0.05 * rng.randn(20000))
    binner = gbt.Binner().fit(X)
    B = binner.transform(X)
    g = (np.zeros(len(y)) + 0.5) - y
    h = np.ones(len(y))
    pred_out = []
    t_gpu = gbt._build_tree(B, g, h, gbt.GpuHistogramBuilder(B), 6,
                            1.0, 0.0, 1.0, 0.3, None, pred_out=pred_out)
    assert pred_out, "GPU path must hand back per-row predictions"
    assert np.allclose(pred_out[0], t_gpu.predict_binned(B), atol=1e-5)
    t_cpu = gbt._build_tree(B, g, h, gbt.CpuHistogramBuilder(B), 6,
                            1.0, 0.0, 1.0, 0.3, None)
    r_gpu = float(np.mean((t_gpu.predict_binned(B) + g) ** 2))
    r_cpu = float(np.mean((t_cpu.predict_binned(B) + g) ** 2))
    assert r_gpu < 1.2 * r_cpu + 1e-9, (r_cpu, r_gpu)


def test_gpu_boost_rounds_faster_than_cpu():
    """Round-level speed: device-resident GPU rounds must beat the CPU
    engine (VERDICT round-1 item 6)."""
    import time
    rng = np.random.RandomState(4)
    n, f = 500_000, 32
    X = rng.rand(n, f)
    y = X[:, 0] - X[:, 1] + 0.1 * rng.randn(n)
    p = {"n_estimators": 3, "max_depth": 6}
    t0 = time.perf_counter()
    gbt.train(X, y, p)
    t_cpu = time.perf_counter() - t0
    gbt.train(X[:4096], y[:4096], p, use_gpu=True)  # warm kernels
    t0 = time.perf_counter()
    gbt.train(X, y, p, use_gpu=True)
    t_gpu = time.perf_counter() - t0
    print("gbt 3 rounds: cpu %.2fs gpu %.2fs" % (t_cpu, t_gpu))
    assert t_gpu < t_cpu, (t_cpu, t_gpu)
