"""GPU numerics for the hand-written flash attention (fwd + bwd)
against a plain fp32 PyTorch reference (SURVEY.md §4 kernel-test bar)."""

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


def _ref_attention(q, k, v):
    """fp32 reference: softmax(QK^T/sqrt(d)) V."""
    d = q.shape[-1]
    s = q @ k.transpose(-1, -2) / d ** 0.5
    return torch.softmax(s, dim=-1) @ v


@pytest.mark.parametrize("BH,S", [(4, 64), (2, 128), (3, 256), (8, 512)])
def test_attn_fwd_matches_reference(BH, S):
    torch.manual_seed(100 + S)
    q = torch.randn(BH, S, 64, device="cuda").bfloat16()
    k = torch.randn(BH, S, 64, device="cuda").bfloat16()
    v = torch.randn(BH, S, 64, device="cuda").bfloat16()
    o, lse = ops.ext().attn_fwd(q, k, v, 0.0, None)
    ref = _ref_attention(q.float(), k.float(), v.float())
    assert torch.allclose(o.float(), ref, atol=3e-2, rtol=3e-2), \
        (o.float() - ref).abs().max()
    # LSE check: logsumexp of the scaled scores
    s = q.float() @ k.float().transpose(-1, -2) / 8.0
    lse_ref = torch.logsumexp(s, dim=-1)
    assert torch.allclose(lse, lse_ref, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("BH,S", [(2, 128), (4, 256)])
def test_attn_bwd_matches_reference(BH, S):
    torch.manual_seed(200 + S)
    q = (torch.randn(BH, S, 64, device="cuda") / 2).bfloat16()
    k = (torch.randn(BH, S, 64, device="cuda") / 2).bfloat16()
    v = (torch.randn(BH, S, 64, device="cuda") / 2).bfloat16()
    qg = q.clone().requires_grad_(True)
    kg = k.clone().requires_grad_(True)
    vg = v.clone().requires_grad_(True)
    o = F_._FlashAttnFn.apply(qg, kg, vg, 0.0)
    do = torch.randn_like(o)
    o.backward(do)

    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    vr = v.float().requires_grad_(True)
    orf = _ref_attention(qr, kr, vr)
    orf.backward(do.float())

    assert torch.allclose(o.float(), orf.detach(), atol=3e-2, rtol=3e-2)
    for got, want, name in ((qg.grad, qr.grad, "dq"),
                            (kg.grad, kr.grad, "dk"),
                            (vg.grad, vr.grad, "dv")):
        err = (got.float() - want).abs().max()
        assert torch.allclose(got.float(), want, atol=8e-2, rtol=8e-2), \
            (name, err)


def test_attn_dropout_deterministic_and_unbiased():
    torch.manual_seed(7)
    BH, S = 4, 256
    q = torch.randn(BH, S, 64, device="cuda").bfloat16()
    k = torch.randn(BH, S, 64, device="cuda").bfloat16()
    v = torch.randn(BH, S, 64, device="cuda").bfloat16()
    seed = torch.tensor([12345], dtype=torch.int64, device="cuda")
    o1, _ = ops.ext().attn_fwd(q, k, v, 0.3, seed)
    o2, _ = ops.ext().attn_fwd(q, k, v, 0.3, seed)
    assert torch.equal(o1, o2)  # same seed -> identical mask
    seed2 = torch.tensor([54321], dtype=torch.int64, device="cuda")
    o3, _ = ops.ext().attn_fwd(q, k, v, 0.3, seed2)
    assert not torch.equal(o1, o3)  # different seed -> different mask
    # dropout is mean-preserving: averaging over many seeds approaches
    # the p=0 output
    o0, _ = ops.ext().attn_fwd(q, k, v, 0.0, None)
    acc = torch.zeros_like(o0, dtype=torch.float32)
    n = 24
    for i in range(n):
        s = torch.tensor([1000 + i], dtype=torch.int64, device="cuda")
        oi, _ = ops.ext().attn_fwd(q, k, v, 0.3, s)
        acc += oi.float()
    mean_err = (acc / n - o0.float()).abs().mean()
    base = o0.float().abs().mean()
    assert mean_err < 0.15 * base, (mean_err, base)


def test_attn_dropout_bwd_consistent():
    """With dropout the kernel pair must agree with an autograd
    reference built from the SAME mask. Recover the mask by probing the
    fwd kernel with one-hot V columns is expensive; instead check
    self-consistency: grads at p>0 from two identical calls match, and
    the finite-difference direction for V matches (dV is linear in
    the mask)."""
    BH, S = 2, 128
    torch.manual_seed(9)
    q = (torch.randn(BH, S, 64, device="cuda") / 2).bfloat16()
    k = (torch.randn(BH, S, 64, device="cuda") / 2).bfloat16()
    v = (torch.randn(BH, S, 64, device="cuda") / 2).bfloat16()
    seed = torch.tensor([777], dtype=torch.int64, device="cuda")
    C = ops.ext()
    o, lse = C.attn_fwd(q, k, v, 0.25, seed)
    do = torch.randn_like(o)
    drow = (do.float() * o.float()).sum(-1)
    dq1, dk1, dv1 = C.attn_bwd(q, k, v, do, lse, drow, 0.25, seed)
    dq2, dk2, dv2 = C.attn_bwd(q, k, v, do, lse, drow, 0.25, seed)
    assert torch.equal(dq1, dq2) and torch.equal(dk1, dk2) \
        and torch.equal(dv1, dv2)
    # dV linearity: O(v + e*dv_dir) - O(v) == e * (Pm @ dv_dir); check
    # the directional derivative of sum(O*do) against <dV, dv_dir>
    dv_dir = torch.randn_like(v).float() * 0.1
    vp = (v.float() + dv_dir).bfloat16()
    op, _ = C.attn_fwd(q, k, vp, 0.25, seed)
    lhs = ((op.float() - o.float()) * do.float()).sum()
    rhs = (dv1.float() * (vp.float() - v.float())).sum()
    assert torch.allclose(lhs, rhs, rtol=0.1, atol=2.0), (lhs, rhs)


def test_flash_attention_module_path():
    """BertSelfAttention routes through the hand-written kernels for
    bf16/D=64 and still trains (grads flow)."""
    from sparkdl.models.bert import BertSelfAttention, BertConfig
    cfg = BertConfig(hidden=128, heads=2, dropout=0.1)
    attn = BertSelfAttention(cfg).cuda()
    from sparkdl.ops.modules import convert_bf16_training
    convert_bf16_training(attn)
    x = torch.randn(2, 128, 128, device="cuda").bfloat16()
    y = attn(x)
    assert y.shape == x.shape
    y.float().pow(2).mean().backward()
    assert attn.qkv.weight.grad is not None


def test_attn_packed_matches_unpacked():
    torch.manual_seed(31)
    B, S, H = 2, 256, 3
    qkv = torch.randn(B, S, 3, H, 64, device="cuda").bfloat16()
    o_p, lse_p = ops.ext().attn_fwd_packed(qkv, 0.0, None)
    q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)  # [B,H,S,64]
    q3 = q.reshape(B * H, S, 64).contiguous()
    k3 = k.reshape(B * H, S, 64).contiguous()
    v3 = v.reshape(B * H, S, 64).contiguous()
    o_u, lse_u = ops.ext().attn_fwd(q3, k3, v3, 0.0, None)
    o_u_bsH = o_u.view(B, H, S, 64).transpose(1, 2).reshape(B, S, H * 64)
    assert torch.equal(o_p, o_u_bsH)
    assert torch.equal(lse_p, lse_u)


def test_attn_packed_autograd_matches_reference():
    torch.manual_seed(33)
    B, S, H = 2, 128, 2
    qkv = (torch.randn(B, S, 3, H, 64, device="cuda") / 2).bfloat16()
    qkv_g = qkv.clone().requires_grad_(True)
    o = F_.flash_attention_packed(qkv_g, 0.0)
    do = torch.randn_like(o)
    o.backward(do)

    qkv_r = qkv.float().requires_grad_(True)
    q, k, v = qkv_r.permute(2, 0, 3, 1, 4).unbind(0)
    s = q @ k.transpose(-1, -2) / 8.0
    o_r = (torch.softmax(s, -1) @ v).transpose(1, 2).reshape(B, S, H * 64)
    o_r.backward(do.float())

    assert torch.allclose(o.float(), o_r.detach(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(qkv_g.grad.float(), qkv_r.grad, atol=1e-1,
                          rtol=1e-1), \
        (qkv_g.grad.float() - qkv_r.grad).abs().max()
