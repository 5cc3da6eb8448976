"""Autograd functions over the sparkdl._C HIP kernels, plus the plain
PyTorch fp32 reference implementations the GPU numerics tests compare
against ("HIP-kernel unit tests vs PyTorch-ROCm reference
outputs")."""

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
