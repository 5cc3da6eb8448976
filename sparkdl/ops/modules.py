"""nn.Module wrappers over the fused ops.

``LayerNorm`` keeps fp32 affine parameters (fp32 accumulation everywhere,
SURVEY.md §7 hard part 4) while activations flow in bf16.
``LinearGelu``/``Linear`` run on the hand-written 256x256 16-wave MFMA
GEMM with the bias(+GELU) epilogue fused in-kernel (default since round
2; ``SPARKDL_FUSED_GEMM=0`` restores the library path).  ``Conv1x1``
exposes the streaming tall-skinny GEMM for 1x1 convolutions (opt-in —
see its docstring for the measured routing decision).
"""

import torch
import torch.nn as nn

from sparkdl.ops import functional as F_


class LayerNorm(nn.Module):
    def __init__(self, hidden, eps=1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden, dtype=torch.float32))
        self.bias = nn.Parameter(torch.zeros(hidden, dtype=torch.float32))
        self.eps = eps

    def forward(self, x):
        return F_.layer_norm(x, self.weight, self.bias, self.eps)

    def extra_repr(self):
        return "%d, eps=%g" % (self.weight.numel(), self.eps)


class BatchNormAct2d(nn.Module):
    """Fused BatchNorm2d with optional ReLU and residual add:
    ``y = relu?(bn(x) [+ residual])`` in ONE kernel chain (bf16 NHWC on
    GPU; reference torch path elsewhere).  fp32 affine params and running
    stats, matching torch.nn.BatchNorm2d semantics (unbiased running
    var, momentum EMA)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(
            torch.ones(num_features, dtype=torch.float32))
        self.bias = nn.Parameter(
            torch.zeros(num_features, dtype=torch.float32))
        self.register_buffer(
            "running_mean", torch.zeros(num_features, dtype=torch.float32))
        self.register_buffer(
            "running_var", torch.ones(num_features, dtype=torch.float32))

    def forward(self, x, residual=None):
        return F_.batch_norm_act(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            momentum=self.momentum, eps=self.eps, training=self.training,
            relu=self.relu, residual=residual)

    def extra_repr(self):
        return "%d, relu=%s" % (self.num_features, self.relu)


def convert_bf16_training(model):
    """Pure-bf16 training regime: GEMM/embedding weights become bf16
    (grads flow in bf16, halving all-reduce traffic; no per-step
    autocast weight casts), while norm affines and fused-epilogue biases
    stay fp32 and FusedAdamW keeps fp32 master weights + moments."""
    for mod in model.modules():
        if isinstance(mod, (nn.Linear, nn.Embedding)):
            mod.to(torch.bfloat16)
        elif isinstance(mod, (Linear, LinearGelu)):
            mod.weight.data = mod.weight.data.bfloat16()  # bias stays fp32
    return model


class LinearGelu(nn.Module):
    """y = gelu(x @ W^T + b) with the bias+GELU fused into one kernel."""

    def __init__(self, in_features, out_features):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        # fp32 bias: consumed directly by the fused epilogue kernel.
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=torch.float32))
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        import os
        # Default ON since round 2: the 256x256 16-wave MFMA kernel with
        # the fused epilogue beats hipBLASLt + separate bias_gelu on the
        # BERT shapes. SPARKDL_FUSED_GEMM=0 restores the library path.
        if (os.environ.get("SPARKDL_FUSED_GEMM", "1") != "0"
                and F_.linear_gelu_fused_ok(x, self.weight)):
            return F_.linear_gelu_fused(x, self.weight, self.bias)
        h = torch.nn.functional.linear(x, self.weight.to(x.dtype))
        return F_.bias_gelu(h, self.bias)


class Linear(nn.Module):
    """Drop-in nn.Linear with the bias add fused into the MFMA GEMM
    epilogue (act=0). fp32 bias, consumed directly by the kernel; the
    torch path (CPU / non-bf16 / ragged K) matches numerics."""

    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(
            torch.zeros(out_features, dtype=torch.float32)) if bias \
            else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        import os
        if (os.environ.get("SPARKDL_FUSED_GEMM", "1") != "0"
                and F_.linear_gelu_fused_ok(x, self.weight)):
            return F_.linear_fused(x, self.weight, self.bias)
        y = torch.nn.functional.linear(x, self.weight.to(x.dtype))
        if self.bias is not None:
            y = y + self.bias.to(y.dtype)
        return y

    def extra_repr(self):
        return "%d, %d, bias=%s" % (self.in_features, self.out_features,
                                    self.bias is not None)


class Conv1x1(nn.Module):
    """1x1 convolution as a row-major MFMA GEMM over NHWC rows
    (SURVEY.md §2.2 N5 — the ResNet bottleneck reduce/expand convs).

    For stride-1 bf16 channels_last inputs with Cin % 64 == 0 the op is
    exactly ``y[nhw, co] = x[nhw, :] @ W[co, :]^T`` on the in-house
    256x256 16-wave GEMM kernel (dgrad included); anything else falls
    back to the library conv. SPARKDL_CONV1X1=0 forces the fallback.
    """

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.cin, self.cout, self.stride = cin, cout, stride
        self.weight = nn.Parameter(torch.empty(cout, cin))
        # kaiming fan_out for a 1x1 conv is just cout
        nn.init.kaiming_normal_(self.weight, mode="fan_out",
                                nonlinearity="relu")

    def forward(self, x):
        import os
        mode = os.environ.get("SPARKDL_CONV1X1", "0")
        # Per-site A/B vs MIOpen (profiles/conv1x1_sites.log): the
        # in-house GEMM forward wins at the K>=512 sites but the full
        # fwd+dgrad+wgrad package still trails MIOpen's tuned igemm
        # end-to-end (same-box: miopen 8701 img/s, K>=512-gated 8451,
        # all-in-house 7988), so the DEFAULT stays on MIOpen. "1"
        # routes the K>=512 sites in-house, "all" routes every 1x1.
        use = (self.stride == 1 and x.is_cuda
               and x.dtype == torch.bfloat16 and self.cin % 64 == 0
               and mode != "0"
               and (mode == "all" or self.cin >= 512))
        if use:
            xp = x.permute(0, 2, 3, 1)  # channels_last -> contiguous view
            y = F_.conv1x1_gemm(xp, self.weight.to(x.dtype))
            return y.permute(0, 3, 1, 2)
        w4 = self.weight.view(self.cout, self.cin, 1, 1)
        return torch.nn.functional.conv2d(x, w4.to(x.dtype),
                                          stride=self.stride)

    def extra_repr(self):
        return "%d, %d, stride=%d" % (self.cin, self.cout, self.stride)
