"""GPU refcheck for the hand-written MFMA GEMM + bias(+GELU) kernel
(asymmetric random operands — catches operand/output transposes)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    import sparkdl.ops as ops
    from sparkdl.ops import functional as F_


@pytest.fixture(autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")


@pytest.mark.parametrize("M,N,K", [(256, 256, 64), (512, 256, 128),
                                   (384, 512, 192), (2048, 768, 768),
                                   # ragged edges: clamped loads +
                                   # predicated stores
                                   (200, 100, 64), (300, 2304, 768),
                                   (1000, 64, 128)])
@pytest.mark.parametrize("act", [0, 1])
def test_gemm_bias_act_refcheck(M, N, K, act):
    torch.manual_seed(10 + M + act)
    A = (torch.randn(M, K, device="cuda") / K ** 0.25).bfloat16()
    W = (torch.randn(N, K, device="cuda") / K ** 0.25).bfloat16()
    bias = torch.randn(N, device="cuda")

    out = ops.ext().gemm_bias_act(A, W, bias, act, act == 1)
    ref = A.float() @ W.float().t() + bias
    if act == 1:
        y, z = out
        assert torch.allclose(z.float(), ref, atol=5e-2, rtol=5e-2), \
            (z.float() - ref).abs().max()
        ref = torch.nn.functional.gelu(ref)
        assert torch.allclose(y.float(), ref, atol=5e-2, rtol=5e-2), \
            (y.float() - ref).abs().max()
    else:
        (y,) = out
        assert torch.allclose(y.float(), ref, atol=5e-2, rtol=5e-2), \
            (y.float() - ref).abs().max()


def test_linear_gelu_fused_autograd():
    torch.manual_seed(20)
    M, N, K = 256, 128, 64
    x = (torch.randn(M, K, device="cuda") / 2).bfloat16().requires_grad_(True)
    w = (torch.randn(N, K, device="cuda") / 8).bfloat16().requires_grad_(True)
    b = torch.randn(N, device="cuda", requires_grad=True)
    y = F_.linear_gelu_fused(x, w, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = torch.nn.functional.gelu(xr @ wr.t() + br)
    yr.backward(dy.float())

    assert torch.allclose(y.float(), yr.detach(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(x.grad.float(), xr.grad, atol=1e-1, rtol=1e-1), \
        (x.grad.float() - xr.grad).abs().max()
    assert torch.allclose(w.grad.float(), wr.grad, atol=2e-1, rtol=1e-1), \
        (w.grad.float() - wr.grad).abs().max()
    assert torch.allclose(b.grad, br.grad, atol=2e-1, rtol=5e-2)


def test_transpose_bf16():
    torch.manual_seed(3)
    for (R, C) in [(64, 64), (768, 3072), (100, 200), (2304, 768)]:
        X = torch.randn(R, C, device="cuda").bfloat16()
        Y = ops.ext().transpose_bf16(X)
        assert Y.shape == (C, R)
        assert torch.equal(Y, X.t().contiguous())


def test_linear_fused_autograd():
    torch.manual_seed(21)
    M, N, K = 512, 256, 128
    x = (torch.randn(M, K, device="cuda") / 2).bfloat16().requires_grad_(True)
    w = (torch.randn(N, K, device="cuda") / 8).bfloat16().requires_grad_(True)
    b = torch.randn(N, device="cuda", requires_grad=True)
    y = F_.linear_fused(x, w, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = xr @ wr.t() + br
    yr.backward(dy.float())

    assert torch.allclose(y.float(), yr.detach(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(x.grad.float(), xr.grad, atol=1e-1, rtol=1e-1)
    assert torch.allclose(w.grad.float(), wr.grad, atol=2e-1, rtol=1e-1)
    assert torch.allclose(b.grad, br.grad, atol=2e-1, rtol=5e-2)


def test_linear_module_matches_reference():
    torch.manual_seed(22)
    lin = ops.Linear(128, 192).cuda()
    lin.weight.data = lin.weight.data.bfloat16()
    x = torch.randn(64, 128, device="cuda").bfloat16()
    y = lin(x)
    yr = x.float() @ lin.weight.float().t() + lin.bias
    assert torch.allclose(y.float(), yr, atol=5e-2, rtol=5e-2)


def test_wgrad_splitk_kernel_refcheck():
    """In-house split-M wgrad (tr_b16 fragment path) vs the library
    GEMM, fp32 output."""
    torch.manual_seed(55)
    for (M, N, K) in [(256, 256, 128), (4096, 512, 256), (8192, 256, 384)]:
        dz = (torch.randn(M, N, device="cuda") / 4).bfloat16()
        x = (torch.randn(M, K, device="cuda") / 4).bfloat16()
        got = ops.ext().wgrad_splitk(dz, x)
        want = dz.float().t() @ x.float()
        assert torch.allclose(got, want, atol=5e-1, rtol=5e-2), \
            (M, N, K, (got - want).abs().max())


def test_wgrad_env_routing(monkeypatch):
    """SPARKDL_FUSED_WGRAD=1 routes _wgrad through the in-house kernel
    with matching numerics."""
    monkeypatch.setenv("SPARKDL_FUSED_WGRAD", "1")
    dz = (torch.randn(512, 256, device="cuda") / 4).bfloat16()
    x = (torch.randn(512, 128, device="cuda") / 4).bfloat16()
    got = F_._wgrad(dz, x)
    want = dz.t() @ x
    assert got.dtype == dz.dtype
    assert torch.allclose(got.float(), want.float(), atol=5e-1, rtol=5e-2)
