"""Autograd functions over the sparkdl._C HIP kernels, plus the plain
PyTorch fp32 reference implementations the GPU numerics tests compare
against ("HIP-kernel unit tests vs PyTorch-ROCm reference
outputs")."""

import os

import torch

import sparkdl.ops as _ops


# ---------------------------------------------------------------------------
# Reference implementations (fp32, plain PyTorch) — used on CPU and as the
# numerics oracle for the HIP kernels.
# ---------------------------------------------------------------------------

def layer_norm_ref(x, gamma, beta, eps=1e-5):
    xf = x.float()
    out = torch.nn.functional.layer_norm(
        xf, (x.shape[-1],), gamma.float(), beta.float(), eps)
    return out.to(x.dtype)


def bias_gelu_ref(x, bias):
    return torch.nn.functional.gelu(
        x.float() + bias.float(), approximate="none").to(x.dtype)


# ---------------------------------------------------------------------------
# Autograd bindings
# ---------------------------------------------------------------------------

class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        C = _ops.ext()
        x = x.contiguous()
        y, mean, rstd = C.layernorm_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        x, gamma, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = C.layernorm_bwd(
            x, dy.contiguous(), gamma, mean, rstd)
        return dx, dgamma, dbeta, None


class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        C = _ops.ext()
        x = x.contiguous()
        ctx.save_for_backward(x, bias)
        return C.bias_gelu_fwd(x, bias)

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        x, bias = ctx.saved_tensors
        dx, dbias = C.bias_gelu_bwd(x, bias, dy.contiguous())
        return dx, dbias


def _to_nhwc(t):
    """[N,C,H,W] channels_last tensor -> [N,H,W,C] contiguous view."""
    return t.permute(0, 2, 3, 1).contiguous(memory_format=torch.contiguous_format)


def _from_nhwc(t):
    """[N,H,W,C] contiguous -> [N,C,H,W] tensor in channels_last layout."""
    return t.permute(0, 3, 1, 2)


class _BNReLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, gamma, beta, running_mean, running_var,
                momentum, eps, training, relu):
        C = _ops.ext()
        xp = _to_nhwc(x)
        rp = _to_nhwc(residual) if residual is not None else None
        y, mean, rstd, mask = C.bn_fwd(
            xp, rp, gamma, beta, running_mean, running_var, momentum,
            eps, training, relu)
        # mask (1 bit/elem) replaces y in backward — y itself need not
        # be kept alive by autograd
        ctx.save_for_backward(xp, mask, gamma, mean, rstd)
        ctx.bn_flags = (training, relu, residual is not None)
        return _from_nhwc(y)

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        xp, mask, gamma, mean, rstd = ctx.saved_tensors
        training, relu, has_res = ctx.bn_flags
        needs_dres = has_res and ctx.needs_input_grad[1]
        outs = C.bn_bwd(xp, mask, _to_nhwc(dy), gamma, mean, rstd,
                        training, relu, needs_dres)
        dx = _from_nhwc(outs[0])
        dres = _from_nhwc(outs[3]) if needs_dres else None
        return (dx, dres, outs[1], outs[2], None, None, None, None, None,
                None)


def batch_norm_act(x, gamma, beta, running_mean, running_var,
                   momentum=0.1, eps=1e-5, training=True, relu=False,
                   residual=None):
    """Fused BatchNorm(+residual add)(+ReLU).

    bf16 channels_last GPU tensors hit the CDNA4 kernels; everything else
    runs the reference torch path (same math, also the numerics oracle)."""
    # The NHWC kernels stage per-channel folds in LDS sized for
    # C <= 2048 (csrc/batchnorm.hip `lds[2*2048]`) and read bf16 in
    # 8-wide packs; larger or ragged channel counts take the reference
    # path rather than corrupting LDS.
    C = x.shape[1]
    if (x.is_cuda and x.dtype == torch.bfloat16
            and C <= 2048 and C % 8 == 0):
        return _BNReLUFn.apply(x, residual, gamma, beta, running_mean,
                               running_var, momentum, eps, training, relu)
    # reference path: fp32 compute (stats/affine are fp32), cast back
    y = torch.nn.functional.batch_norm(
        x.float(), running_mean, running_var, gamma, beta,
        training, momentum, eps).to(x.dtype)
    if residual is not None:
        y = y + residual
    return torch.relu(y) if relu else y


def layer_norm(x, gamma, beta, eps=1e-5):
    """LayerNorm over the last dim. HIP kernel for bf16-on-GPU, reference
    path otherwise (CPU tests)."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _LayerNormFn.apply(x, gamma, beta, eps)
    return torch.nn.functional.layer_norm(
        x, (x.shape[-1],), gamma.to(x.dtype), beta.to(x.dtype), eps)


def bias_gelu(x, bias):
    """Fused bias+GELU (erf). HIP kernel for bf16-on-GPU, reference path
    otherwise."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        return _BiasGeluFn.apply(x, bias)
    return torch.nn.functional.gelu(x + bias.to(x.dtype), approximate="none")


# ---------------------------------------------------------------------------
# Fused MFMA GEMM + bias + GELU
# ---------------------------------------------------------------------------

_zero_bias_cache = {}


def _zero_bias(n, device):
    key = (n, device)
    if key not in _zero_bias_cache:
        _zero_bias_cache[key] = torch.zeros(
            n, dtype=torch.float32, device=device)
    return _zero_bias_cache[key]


class _LinearGeluFusedFn(torch.autograd.Function):
    """y = gelu(x @ W^T + b) with the GEMM + epilogue in one hand-written
    MFMA kernel; backward reuses the bias_gelu_bwd kernel on the saved
    pre-activation and rocBLAS for dgrad/wgrad."""

    @staticmethod
    def forward(ctx, x2d, weight, bias):
        C = _ops.ext()
        y, z = C.gemm_bias_act(x2d, weight, bias, 1, True)
        ctx.save_for_backward(x2d, weight, z)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        x2d, weight, z = ctx.saved_tensors
        dy = dy.contiguous()
        # dz = dy * gelu'(z); dbias = column-sum(dz)
        dz, dbias = C.bias_gelu_bwd(z, _zero_bias(z.shape[-1], z.device),
                                    dy)
        # dgrad through the same MFMA kernel: dX[M,K] = dZ[M,N] @ W[N,K]
        # == gemm(A=dZ, B=W^T[K,N]) — W^T is weight-sized, transposed on
        # the fly by a tiled bf16 kernel.
        if (weight.shape[0] % 64 == 0
                and os.environ.get("SPARKDL_FUSED_DGRAD", "1") != "0"):
            # dgrad contracts over N
            wt = C.transpose_bf16(weight)
            dx = C.gemm_bias_act(dz, wt, None, 0, False)[0]
        else:
            dx = dz @ weight        # [M,N] @ [N,K]
        dw = _wgrad(dz, x2d)
        return dx, dw, dbias


def linear_gelu_fused(x, weight, bias):
    """Apply the fused MFMA GEMM+bias+GELU path; caller guarantees bf16
    CUDA tensors with K % 64 == 0 (M/N edges are handled in-kernel)."""
    shape = x.shape
    x2d = x.reshape(-1, shape[-1]).contiguous()
    y = _LinearGeluFusedFn.apply(x2d, weight.contiguous(), bias)
    return y.reshape(*shape[:-1], weight.shape[0])


def linear_gelu_fused_ok(x, weight):
    # the 256x256 kernel clamps/predicates M and N edges; only the
    # K-loop step (64) is a hard requirement
    return (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and x.shape[-1] % 64 == 0)


# ---------------------------------------------------------------------------
# Flash attention (SURVEY.md N6 attention GEMMs) — hand-written fwd+bwd,
# D=64 heads, replaces torch SDPA (aotriton) on the BERT hot path.
# ---------------------------------------------------------------------------

_attn_seed = {}


def _attn_seed_tensor(device):
    """Per-device persistent dropout seed, advanced by a DEVICE op each
    forward so the pattern changes per step even under hipGraph capture
    (the increment is captured and replayed). Offset by the launcher's
    RANK: data-parallel ranks seed torch identically for weight init,
    which must not make their dropout masks identical too."""
    if device not in _attn_seed:
        rank = int(os.environ.get("RANK", "0"))
        _attn_seed[device] = torch.randint(
            0, 2 ** 31, (1,), dtype=torch.int64,
            device=device) + rank * 2654435761 % (2 ** 31)
    return _attn_seed[device]


class _FlashAttnFn(torch.autograd.Function):
    """O = softmax(Q K^T / sqrt(64)) V with optional attention-prob
    dropout. q/k/v: [BH, S, 64] bf16 contiguous, S % 64 == 0."""

    @staticmethod
    def forward(ctx, q, k, v, dropout_p):
        C = _ops.ext()
        seed = None
        if dropout_p > 0:
            seed = _attn_seed_tensor(q.device)
            seed.add_(1)
            # the backward must replay this step's seed even if later
            # steps advanced the counter
            seed = seed.clone()
        o, lse = C.attn_fwd(q, k, v, float(dropout_p), seed)
        ctx.save_for_backward(q, k, v, o, lse,
                              seed if seed is not None else
                              torch.empty(0))
        ctx.dropout_p = float(dropout_p)
        return o

    @staticmethod
    def backward(ctx, do):
        C = _ops.ext()
        q, k, v, o, lse, seed = ctx.saved_tensors
        if seed.numel() == 0:
            seed = None
        do = do.contiguous()
        # D_i = rowsum(dO o O), fp32 (flash backward precompute)
        drow = (do.float() * o.float()).sum(-1)
        dq, dk, dv = C.attn_bwd(q, k, v, do, lse, drow, ctx.dropout_p,
                                seed)
        return dq, dk, dv, None


def flash_attention(q, k, v, dropout_p=0.0):
    """[B, h, S, d] attention through the hand-written kernels; the
    caller checks flash_attention_ok first."""
    B, h, S, d = q.shape
    q3 = q.reshape(B * h, S, d).contiguous()
    k3 = k.reshape(B * h, S, d).contiguous()
    v3 = v.reshape(B * h, S, d).contiguous()
    o = _FlashAttnFn.apply(q3, k3, v3, dropout_p)
    return o.view(B, h, S, d)


def flash_attention_ok(q):
    return (q.is_cuda and q.dtype == torch.bfloat16 and q.dim() == 4
            and q.shape[-1] == 64 and q.shape[-2] % 64 == 0)


class _FlashAttnPackedFn(torch.autograd.Function):
    """Zero-copy BERT attention: reads the packed qkv buffer
    [B,S,3,H,64] (the qkv Linear's output, viewed) with strided kernel
    rows, writes O directly as [B,S,H*64], and the backward emits the
    packed dqkv the qkv Linear's backward consumes — no permute/
    contiguous copies anywhere around the attention."""

    @staticmethod
    def forward(ctx, qkv, dropout_p):
        C = _ops.ext()
        seed = None
        if dropout_p > 0:
            seed = _attn_seed_tensor(qkv.device)
            seed.add_(1)
            seed = seed.clone()
        o, lse = C.attn_fwd_packed(qkv, float(dropout_p), seed)
        ctx.save_for_backward(qkv, o, lse,
                              seed if seed is not None else
                              torch.empty(0))
        ctx.dropout_p = float(dropout_p)
        return o

    @staticmethod
    def backward(ctx, do):
        C = _ops.ext()
        qkv, o, lse, seed = ctx.saved_tensors
        if seed.numel() == 0:
            seed = None
        do = do.contiguous()
        B, S, HD = o.shape
        H = HD // 64
        # D_i = rowsum(dO o O) per (b,h,s): [B,S,H] -> [B*H, S]
        drow = (do.float() * o.float()).view(B, S, H, 64).sum(-1) \
            .transpose(1, 2).reshape(B * H, S).contiguous()
        dqkv = C.attn_bwd_packed(qkv, do, lse, drow, ctx.dropout_p, seed)
        return dqkv, None


def flash_attention_packed(qkv, dropout_p=0.0):
    """qkv: [B, S, 3, H, 64] bf16 contiguous -> O [B, S, H*64]."""
    return _FlashAttnPackedFn.apply(qkv.contiguous(), dropout_p)


def _conv_gemm(a2d, w):
    """Best-kernel cascade for bias-free GEMMs (1x1 conv fwd/dgrad):
    streaming tall-skinny kernel when the shape fits, else the 256x256
    tile kernel, else the library GEMM."""
    C = _ops.ext()
    M, K = a2d.shape
    N = w.shape[0]
    if (K in (64, 128, 256)) and N % 64 == 0:
        return C.gemm_stream(a2d, w)
    if K % 64 == 0:
        return C.gemm_bias_act(a2d, w, None, 0, False)[0]
    return a2d @ w.t()


def _wgrad(dz, x2d):
    """Weight gradient dW = dZ^T @ X. Default: library GEMM (Tensile's
    split-K TN kernels measure ~650 TF on the BERT shapes vs the
    in-house tr_b16 kernel's 506-534 — probe ladder in
    profiles/wgrad_*.log). SPARKDL_FUSED_WGRAD=1 opts into the
    in-house kernel where its shape constraints hold."""
    if (os.environ.get("SPARKDL_FUSED_WGRAD", "0") == "1"
            and dz.shape[1] % 256 == 0 and x2d.shape[1] % 128 == 0
            and dz.shape[0] % 64 == 0):
        C = _ops.ext()
        return C.wgrad_splitk(dz, x2d).to(dz.dtype)
    return dz.t() @ x2d


def _wgrad_splitk(dy, x2d, chunk_rows=32768):
    """dW[N,K] = dy^T @ x for tall-skinny operands. A single TN GEMM
    with a tiny N x K output gives Tensile a near-empty grid (measured
    ~2 ms for the ResNet L1 wgrad); splitting the huge contraction into
    row chunks via a batched GEMM fills the chip, then the partials
    reduce in fp32."""
    M = dy.shape[0]
    if M <= 2 * chunk_rows:
        return dy.t() @ x2d
    # smallest divisor split keeping chunks near the target (avoids
    # padded copies of the big operands; NHW row counts are highly
    # composite)
    S = 0
    for s in range((M + chunk_rows - 1) // chunk_rows, 257):
        if M % s == 0:
            S = s
            break
    if S == 0:
        return dy.t() @ x2d
    cr = M // S
    dyb = dy.view(S, cr, -1).transpose(1, 2)
    xb = x2d.view(S, cr, -1)
    return torch.bmm(dyb, xb).sum(0, dtype=torch.float32).to(dy.dtype)


class _Conv1x1Fn(torch.autograd.Function):
    """y = x @ W^T over NHWC rows (1x1 conv). Forward and dgrad run on
    the in-house kernels (streaming kernel at the ResNet shapes);
    wgrad on the library GEMM."""

    @staticmethod
    def forward(ctx, x2d, weight):
        ctx.save_for_backward(x2d, weight)
        return _conv_gemm(x2d, weight)

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        x2d, weight = ctx.saved_tensors
        dy = dy.contiguous()
        wt = C.transpose_bf16(weight)
        dx = _conv_gemm(dy, wt)
        dw = _wgrad_splitk(dy, x2d)
        return dx, dw


def conv1x1_gemm(x_nhwc, weight):
    """x_nhwc: [..., Cin] bf16; weight [Cout, Cin] bf16."""
    shape = x_nhwc.shape
    x2d = x_nhwc.reshape(-1, shape[-1]).contiguous()
    y = _Conv1x1Fn.apply(x2d, weight.contiguous())
    return y.reshape(*shape[:-1], weight.shape[0])


class _LinearFusedFn(torch.autograd.Function):
    """y = x @ W^T + b with the bias fused into the MFMA GEMM epilogue
    (act=0). dgrad runs on the same kernel via W^T; wgrad/dbias on
    library GEMM / torch reductions."""

    @staticmethod
    def forward(ctx, x2d, weight, bias):
        C = _ops.ext()
        y = C.gemm_bias_act(x2d, weight, bias, 0, False)[0]
        ctx.save_for_backward(x2d, weight)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _ops.ext()
        x2d, weight = ctx.saved_tensors
        dy = dy.contiguous()
        if (weight.shape[0] % 64 == 0
                and os.environ.get("SPARKDL_FUSED_DGRAD", "1") != "0"):
            wt = C.transpose_bf16(weight)
            dx = C.gemm_bias_act(dy, wt, None, 0, False)[0]
        else:
            dx = dy @ weight
        dw = _wgrad(dy, x2d)
        # dtype-arg sum fuses the bf16->fp32 conversion into the
        # reduction (no intermediate fp32 copy of dy)
        dbias = dy.sum(0, dtype=torch.float32) if ctx.has_bias else None
        return dx, dw, dbias


def linear_fused(x, weight, bias):
    """Fused-bias MFMA linear (act=0); same eligibility rule as
    linear_gelu_fused_ok."""
    shape = x.shape
    x2d = x.reshape(-1, shape[-1]).contiguous()
    y = _LinearFusedFn.apply(x2d, weight.contiguous(), bias)
    return y.reshape(*shape[:-1], weight.shape[0])
