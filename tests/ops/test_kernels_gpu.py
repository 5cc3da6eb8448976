"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference
of the same op (SURVEY.md §4 test-pyramid plan)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    import sparkdl.ops as ops
    from sparkdl.ops import functional as F_


def _bf16_close(a, b, atol=2e-2, rtol=2e-2):
    return torch.allclose(a.float(), b.float(), atol=atol, rtol=rtol)


@pytest.fixture(autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")


@pytest.mark.parametrize("rows,cols", [(128, 768), (512, 1024), (64, 769),
                                       (8192, 768)])
def test_layernorm_fwd(rows, cols):
    torch.manual_seed(0)
    x = torch.randn(rows, cols, device="cuda").bfloat16()
    g = torch.randn(cols, device="cuda")
    b = torch.randn(cols, device="cuda")
    y, mean, rstd = ops.ext().layernorm_fwd(x, g, b, 1e-5)
    ref = F_.layer_norm_ref(x, g, b, 1e-5)
    assert _bf16_close(y, ref), (y - ref).abs().max()
    # statistics in fp32
    ref_mean = x.float().mean(-1)
    assert torch.allclose(mean, ref_mean, atol=1e-3)


@pytest.mark.parametrize("rows,cols", [(128, 768), (512, 1024), (64, 769)])
def test_layernorm_bwd(rows, cols):
    torch.manual_seed(1)
    x = torch.randn(rows, cols, device="cuda").bfloat16()
    g = torch.randn(cols, device="cuda", requires_grad=True)
    b = torch.randn(cols, device="cuda", requires_grad=True)
    dy = torch.randn(rows, cols, device="cuda").bfloat16()

    # reference in fp32
    xr = x.float().detach().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(xr, (cols,), g, b, 1e-5)
    yr.backward(dy.float())

    yk, mean, rstd = ops.ext().layernorm_fwd(x, g.detach(), b.detach(), 1e-5)
    dx, dgamma, dbeta = ops.ext().layernorm_bwd(
        x, dy, g.detach(), mean, rstd)
    assert _bf16_close(dx, xr.grad.bfloat16(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(dgamma, g.grad, atol=0.5, rtol=1e-2), \
        (dgamma - g.grad).abs().max()
    assert torch.allclose(dbeta, b.grad, atol=0.5, rtol=1e-2)


@pytest.mark.parametrize("rows,cols", [(256, 3072), (128, 769)])
def test_bias_gelu_fwd_bwd(rows, cols):
    torch.manual_seed(2)
    x = torch.randn(rows, cols, device="cuda").bfloat16()
    b = torch.randn(cols, device="cuda")
    y = ops.ext().bias_gelu_fwd(x, b)
    ref = F_.bias_gelu_ref(x, b)
    assert _bf16_close(y, ref)

    dy = torch.randn(rows, cols, device="cuda").bfloat16()
    xr = x.float().detach().requires_grad_(True)
    br = b.detach().requires_grad_(True)
    torch.nn.functional.gelu(xr + br).backward(dy.float())
    dx, dbias = ops.ext().bias_gelu_bwd(x, b, dy)
    assert _bf16_close(dx, xr.grad.bfloat16(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(dbias, br.grad, atol=0.5, rtol=1e-2)


@pytest.mark.parametrize("relu,res", [(False, False), (True, False),
                                      (True, True)])
@pytest.mark.parametrize("C,HW", [(64, 56), (256, 14)])
def test_batch_norm_act_matches_reference(relu, res, C, HW):
    torch.manual_seed(7)
    N = 8
    x = torch.randn(N, C, HW, HW, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    residual = None
    if res:
        residual = torch.randn_like(x.detach()).requires_grad_(True)
    gamma = torch.rand(C, device="cuda") + 0.5
    beta = torch.randn(C, device="cuda")
    gamma.requires_grad_(True)
    beta.requires_grad_(True)
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")

    y = ops.batch_norm_act(x, gamma, beta, rm, rv, training=True,
                           relu=relu, residual=residual)
    dy = torch.randn_like(y.detach())
    y.backward(dy)

    # fp32 reference
    xr = x.detach().float().requires_grad_(True)
    gr = gamma.detach().clone().requires_grad_(True)
    br = beta.detach().clone().requires_grad_(True)
    rm_r = torch.zeros(C, device="cuda")
    rv_r = torch.ones(C, device="cuda")
    yr = torch.nn.functional.batch_norm(xr, rm_r, rv_r, gr, br, True,
                                        0.1, 1e-5)
    if res:
        rr = residual.detach().float().requires_grad_(True)
        yr = yr + rr
    if relu:
        yr = torch.relu(yr)
    yr.backward(dy.float())

    # Exclude ReLU-boundary elements: bf16 rounding flips the mask where
    # bn(x)+res ~ 0, making both implementations differ by a full dy.
    with torch.no_grad():
        pre = torch.nn.functional.batch_norm(
            x.detach().float(), torch.zeros_like(rm), torch.ones_like(rv),
            gamma.detach(), beta.detach(), True, 0.0, 1e-5)
        if res:
            pre = pre + residual.detach().float()
    interior = (pre.abs() > 3e-2) if relu else torch.ones_like(pre,
                                                               dtype=bool)
    assert torch.allclose(y.float()[interior], yr.detach()[interior],
                          atol=5e-2, rtol=5e-2)
    assert torch.allclose(rm, rm_r, atol=1e-2, rtol=1e-2)
    assert torch.allclose(rv, rv_r, atol=1e-2, rtol=1e-2)
    assert torch.allclose(x.grad.float()[interior], xr.grad[interior],
                          atol=5e-2, rtol=5e-2), \
        (x.grad.float()[interior] - xr.grad[interior]).abs().max()
    nhw = N * HW * HW
    assert torch.allclose(gamma.grad, gr.grad, atol=2e-2 * nhw ** 0.5,
                          rtol=2e-2), (gamma.grad - gr.grad).abs().max()
    assert torch.allclose(beta.grad, br.grad, atol=2e-2 * nhw ** 0.5,
                          rtol=2e-2)
    if res:
        assert torch.allclose(residual.grad.float()[interior],
                              rr.grad[interior], atol=5e-2, rtol=5e-2)


def test_layer_norm_autograd_roundtrip():
    torch.manual_seed(3)
    x = torch.randn(32, 768, device="cuda").bfloat16().requires_grad_(True)
    ln = ops.LayerNorm(768).cuda()
    y = ln(x)
    y.sum().backward()
    assert x.grad is not None and ln.weight.grad is not None


@pytest.mark.parametrize("shapes", [
    [(1000,), (257, 33), (64,)],
    [(25_600_000,)],  # ResNet-50-sized flat bucket
])
def test_fused_adamw_matches_reference(shapes):
    torch.manual_seed(4)
    params_k = [torch.randn(*s, device="cuda") for s in shapes]
    params_r = [p.clone() for p in params_k]
    for p in params_k + params_r:
        p.requires_grad_(True)
    grads = [torch.randn_like(p) for p in params_k]

    opt_k = ops.FusedAdamW(params_k, lr=1e-2, weight_decay=0.01)
    opt_r = torch.optim.AdamW(params_r, lr=1e-2, weight_decay=0.01,
                              eps=1e-8, betas=(0.9, 0.999))
    for step in range(3):
        for p, g in zip(params_k, grads):
            p.grad = (g * (step + 1)).clone()
        for p, g in zip(params_r, grads):
            p.grad = (g * (step + 1)).clone()
        opt_k.step()
        opt_r.step()
    for pk, pr in zip(params_k, params_r):
        assert torch.allclose(pk, pr, atol=1e-5, rtol=1e-5), \
            (pk - pr).abs().max()


def test_fused_adamw_bf16_master():
    torch.manual_seed(8)
    shapes = [(777,), (128, 96)]
    params_bf = [torch.randn(*s, device="cuda").bfloat16()
                 .requires_grad_(True) for s in shapes]
    params_ref = [p.detach().float().requires_grad_(True)
                  for p in params_bf]
    grads = [torch.randn_like(p) for p in params_bf]  # bf16 grads

    opt_k = ops.FusedAdamW(params_bf, lr=1e-2, weight_decay=0.01)
    opt_r = torch.optim.AdamW(params_ref, lr=1e-2, weight_decay=0.01,
                              eps=1e-8)
    for step in range(3):
        for p, g in zip(params_bf, grads):
            p.grad = (g * (step + 1)).clone()
        for p, g in zip(params_ref, grads):
            p.grad = (g * (step + 1)).float()
        opt_k.step()
        opt_r.step()
    for p, pr in zip(params_bf, params_ref):
        master = opt_k.state[p]["master"]
        assert torch.allclose(master, pr, atol=1e-5, rtol=1e-5), \
            (master - pr).abs().max()
        assert torch.equal(p.detach(), master.bfloat16())


def test_fused_zero_grads_bf16():
    params = [torch.randn(300, device="cuda").bfloat16()
              .requires_grad_(True)]
    opt = ops.FusedAdamW(params, lr=1e-3)
    params[0].grad = torch.randn_like(params[0])
    opt.step()
    params[0].grad.add_(1.0)
    opt.zero_grad()
    assert torch.all(params[0].grad == 0)


def test_fused_zero_grads():
    torch.manual_seed(6)
    params = [torch.randn(100, device="cuda", requires_grad=True),
              torch.randn(64, 64, device="cuda", requires_grad=True)]
    opt = ops.FusedAdamW(params, lr=1e-3)
    for p in params:
        p.grad = torch.randn_like(p)
    opt.step()  # builds the chunk table
    for p in params:
        p.grad.add_(1.0)
    opt.zero_grad()
    for p in params:
        assert p.grad is not None
        assert torch.all(p.grad == 0), p.grad.abs().max()


def test_fused_sgd_matches_reference():
    torch.manual_seed(5)
    shapes = [(1234,), (128, 256)]
    params_k = [torch.randn(*s, device="cuda") for s in shapes]
    params_r = [p.clone() for p in params_k]
    grads = [torch.randn_like(p) for p in params_k]
    opt_k = ops.FusedSGD(params_k, lr=0.1, momentum=0.9, weight_decay=1e-4)
    opt_r = torch.optim.SGD(params_r, lr=0.1, momentum=0.9,
                            weight_decay=1e-4)
    for step in range(3):
        for p, g in zip(params_k, grads):
            p.grad = (g * (step + 1)).clone()
        for p, g in zip(params_r, grads):
            p.grad = (g * (step + 1)).clone()
        opt_k.step()
        opt_r.step()
    for pk, pr in zip(params_k, params_r):
        assert torch.allclose(pk, pr, atol=1e-5, rtol=1e-5), \
            (pk - pr).abs().max()


def test_batch_norm_act_large_channels_falls_back():
    """C=4096 exceeds the kernel's LDS fold staging (C<=2048): the op
    must route to the torch reference path, not corrupt LDS."""
    torch.manual_seed(9)
    C = 4096
    x = torch.randn(2, C, 4, 4, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    gamma = torch.rand(C, device="cuda") + 0.5
    beta = torch.randn(C, device="cuda")
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")
    y = ops.batch_norm_act(x, gamma, beta, rm, rv, training=True,
                           relu=True)
    yr = torch.relu(torch.nn.functional.batch_norm(
        x.float(), torch.zeros(C, device="cuda"),
        torch.ones(C, device="cuda"), gamma, beta, True, 0.1, 1e-5))
    assert _bf16_close(y, yr.bfloat16(), atol=3e-2, rtol=3e-2)


def test_batch_norm_act_ragged_channels_falls_back():
    """C not divisible by 8 cannot use the 8-wide bf16 packs."""
    C = 36
    x = torch.randn(2, C, 8, 8, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last)
    gamma = torch.ones(C, device="cuda")
    beta = torch.zeros(C, device="cuda")
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")
    y = ops.batch_norm_act(x, gamma, beta, rm, rv, training=True)
    assert y.shape == x.shape


def test_conv1x1_matches_conv2d():
    """Conv1x1 (GEMM path) vs nn.functional.conv2d fp32 reference,
    forward and input gradient."""
    import torch.nn.functional as NF
    torch.manual_seed(44)
    N, C, H, W, Co = 4, 64, 14, 14, 256
    x = torch.randn(N, C, H, W, device="cuda").bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_(True)
    conv = ops.Conv1x1(C, Co).cuda()
    w0 = conv.weight.detach().clone()
    y = conv(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    wr = w0.float().view(Co, C, 1, 1).requires_grad_(True)
    yr = NF.conv2d(xr, wr)
    yr.backward(dy.float())
    assert torch.allclose(y.float(), yr.detach(), atol=5e-2, rtol=5e-2), \
        (y.float() - yr.detach()).abs().max()
    assert torch.allclose(x.grad.float(), xr.grad, atol=1e-1, rtol=1e-1)
    assert torch.allclose(conv.weight.grad.float(),
                          wr.grad.view(Co, C), atol=2e-1, rtol=1e-1)
