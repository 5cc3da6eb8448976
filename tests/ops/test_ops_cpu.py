"""CPU tests for the ops package: reference paths, optimizer reference
math vs torch, module plumbing.  (GPU kernel numerics live in
test_kernels_gpu.py, marked gpu.)"""

import torch

import sparkdl.ops as ops
from sparkdl.ops import functional as F_


def test_layer_norm_cpu_path():
    x = torch.randn(8, 64)
    g, b = torch.ones(64), torch.zeros(64)
    y = F_.layer_norm(x, g, b)
    ref = torch.nn.functional.layer_norm(x, (64,), g, b)
    assert torch.allclose(y, ref, atol=1e-6)


def test_bias_gelu_cpu_path():
    x = torch.randn(8, 64)
    b = torch.randn(64)
    y = F_.bias_gelu(x, b)
    ref = torch.nn.functional.gelu(x + b)
    assert torch.allclose(y, ref, atol=1e-6)


def test_fused_adamw_cpu_matches_torch():
    torch.manual_seed(0)
    p1 = [torch.randn(33, 7, requires_grad=True),
          torch.randn(100, requires_grad=True)]
    p2 = [p.detach().clone().requires_grad_(True) for p in p1]
    o1 = ops.FusedAdamW(p1, lr=1e-2, weight_decay=0.01)
    o2 = torch.optim.AdamW(p2, lr=1e-2, weight_decay=0.01)
    for step in range(4):
        for a, b in zip(p1, p2):
            g = torch.randn_like(a)
            a.grad, b.grad = g.clone(), g.clone()
        o1.step()
        o2.step()
    for a, b in zip(p1, p2):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_fused_sgd_cpu_matches_torch():
    torch.manual_seed(1)
    p1 = [torch.randn(17, 3, requires_grad=True)]
    p2 = [p.detach().clone().requires_grad_(True) for p in p1]
    o1 = ops.FusedSGD(p1, lr=0.1, momentum=0.9, weight_decay=1e-4,
                      nesterov=True)
    o2 = torch.optim.SGD(p2, lr=0.1, momentum=0.9, weight_decay=1e-4,
                         nesterov=True)
    for _ in range(4):
        for a, b in zip(p1, p2):
            g = torch.randn_like(a)
            a.grad, b.grad = g.clone(), g.clone()
        o1.step()
        o2.step()
    for a, b in zip(p1, p2):
        assert torch.allclose(a, b, atol=1e-6)


def test_linear_gelu_module():
    m = ops.LinearGelu(32, 64)
    y = m(torch.randn(4, 32))
    assert y.shape == (4, 64)
    y.sum().backward()
    assert m.weight.grad is not None and m.bias.grad is not None


def test_resnet50_forward_cpu():
    from sparkdl.models.resnet import ResNet50
    m = ResNet50(num_classes=10)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)


def test_batch_norm_act_cpu_path():
    m = ops.BatchNormAct2d(8, relu=True)
    x = torch.randn(2, 8, 4, 4)
    res = torch.randn(2, 8, 4, 4)
    y = m(x, residual=res)
    assert y.shape == x.shape and (y >= 0).all()
    y.sum().backward()
    assert m.weight.grad is not None


def test_fused_adamw_cpu_bf16_master():
    torch.manual_seed(9)
    pbf = [torch.randn(50).bfloat16().requires_grad_(True)]
    pref = [p.detach().float().requires_grad_(True) for p in pbf]
    o1 = ops.FusedAdamW(pbf, lr=1e-2)
    o2 = torch.optim.AdamW(pref, lr=1e-2, weight_decay=1e-2)
    for _ in range(3):
        g = torch.randn(50).bfloat16()
        pbf[0].grad = g.clone()
        pref[0].grad = g.float()
        o1.step()
        o2.step()
    master = o1.state[pbf[0]]["master"]
    assert torch.allclose(master, pref[0], atol=1e-6)
    assert torch.equal(pbf[0].detach(), master.bfloat16())


def test_convert_bf16_training():
    from sparkdl.models.bert import BertBase, BertConfig
    m = BertBase(BertConfig(vocab_size=100, hidden=32, layers=1, heads=2,
                            ffn=64, max_seq=16))
    ops.convert_bf16_training(m)
    assert m.embeddings.word.weight.dtype == torch.bfloat16
    assert m.encoder[0].ffn_in.weight.dtype == torch.bfloat16
    assert m.encoder[0].ffn_in.bias.dtype == torch.float32
    assert m.encoder[0].ln1.weight.dtype == torch.float32


def test_bert_forward_and_mlm_cpu():
    from sparkdl.models.bert import BertBase, BertConfig
    torch.manual_seed(0)
    cfg = BertConfig(vocab_size=120, hidden=32, layers=2, heads=2,
                     ffn=64, max_seq=16, dropout=0.0)
    m = BertBase(cfg)
    ids = torch.randint(0, 120, (3, 16))
    logits = m(ids)
    assert logits.shape == (3, 16, 120)
    positions = torch.tensor([0, 5, 17, 40])
    mlm = m.forward_mlm(ids, positions)
    assert mlm.shape == (4, 120)
    # masked-position head must agree with the full head
    full = logits.reshape(-1, 120)[positions]
    assert torch.allclose(mlm, full, atol=1e-5)
    mlm.sum().backward()
    assert m.embeddings.word.weight.grad is not None


def test_linear_module_cpu_fallback():
    import torch
    from sparkdl.ops import Linear
    torch.manual_seed(5)
    lin = Linear(32, 48)
    x = torch.randn(7, 32)
    y = lin(x)
    ref = x @ lin.weight.t() + lin.bias
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    assert lin.weight.grad is not None and lin.bias.grad is not None


def test_linear_no_bias_cpu():
    import torch
    from sparkdl.ops import Linear
    lin = Linear(16, 8, bias=False)
    assert lin.bias is None
    y = lin(torch.randn(3, 16))
    assert y.shape == (3, 8)


def test_conv1x1_cpu_fallback_matches_conv2d():
    import torch
    from sparkdl.ops import Conv1x1
    torch.manual_seed(6)
    conv = Conv1x1(8, 12)
    x = torch.randn(2, 8, 5, 5)
    y = conv(x)
    ref = torch.nn.functional.conv2d(
        x, conv.weight.view(12, 8, 1, 1))
    assert torch.allclose(y, ref, atol=1e-5)


def test_conv1x1_stride2_cpu():
    import torch
    from sparkdl.ops import Conv1x1
    conv = Conv1x1(4, 6, stride=2)
    y = conv(torch.randn(1, 4, 8, 8))
    assert y.shape == (1, 6, 4, 4)


def test_bert_attention_cpu_fallback():
    """On CPU the attention routes through torch SDPA and still trains."""
    import torch
    from sparkdl.models.bert import BertSelfAttention, BertConfig
    attn = BertSelfAttention(BertConfig(hidden=64, heads=2, dropout=0.0))
    x = torch.randn(2, 16, 64)
    y = attn(x)
    assert y.shape == x.shape
    y.pow(2).mean().backward()
    assert attn.qkv.weight.grad is not None


def test_wgrad_splitk_matches_direct():
    """The batched split-K wgrad equals the direct GEMM (fp32 partial
    reduction) for divisible and non-divisible row counts."""
    import torch
    from sparkdl.ops.functional import _wgrad_splitk
    torch.manual_seed(11)
    for M in (1024, 200_704, 100_000):  # 100k has no divisor <=256 path?
        dy = torch.randn(M, 24)
        x = torch.randn(M, 16)
        got = _wgrad_splitk(dy, x, chunk_rows=4096)
        want = dy.t() @ x
        assert torch.allclose(got, want, rtol=1e-4, atol=1e-2), \
            (M, (got - want).abs().max())
