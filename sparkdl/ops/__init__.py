"""sparkdl.ops — hand-written CDNA4 (gfx950) kernels and their wrappers.

Hot-path ops: fused multi-tensor AdamW/SGD,
LayerNorm, fused bias+GELU.  The HIP extension (sparkdl._C) is built
in-tree for gfx950; on a GPU these ops REQUIRE it — a missing extension
raises instead of silently falling back to eager PyTorch.  On CPU
tensors the pure-PyTorch reference path runs (that reference is also
what the GPU numerics tests compare against).
"""

_EXT = None
_EXT_ERR = None


def ext():
    """Return the loaded sparkdl._C extension; raise loudly on GPU boxes
    where it is missing (no silent eager fallback)."""
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from sparkdl import _C
            _EXT = _C
        except ImportError as e:  # pragma: no cover
            _EXT_ERR = e
    if _EXT is None:
        raise RuntimeError(
            "sparkdl._C HIP extension is not built. Run "
            "`python setup.py build_ext --inplace` (gfx950). "
            "Original error: %s" % _EXT_ERR)
    return _EXT


def has_ext():
    try:
        ext()
        return True
    except RuntimeError:
        return False


from sparkdl.ops.functional import (  # noqa: F401,E402
    layer_norm, bias_gelu, batch_norm_act, layer_norm_ref, bias_gelu_ref,
)
from sparkdl.ops.modules import (  # noqa: F401,E402
    LayerNorm, Linear, LinearGelu, Conv1x1, BatchNormAct2d,
    convert_bf16_training,
)
from sparkdl.ops.optim import FusedAdamW, FusedSGD  # noqa: F401,E402
