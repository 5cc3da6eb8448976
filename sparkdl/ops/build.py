"""In-tree build of the sparkdl._C HIP extension for gfx950.

Used by setup.py (``python setup.py build_ext --inplace``) and by
``__graft_entry__.build()``.  hipcc cross-compiles gfx950 without a GPU;
the resulting ``sparkdl/_C*.so`` travels with the repo snapshot.
"""

import os

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")

_HIP_FLAGS = [
    "--offload-arch=gfx950",
    "-O3",
    "-std=c++17",
    "-fno-gpu-rdc",
]
_CXX_FLAGS = ["-O3", "-std=c++17"]

# Debug/sanitizer build: SPARKDL_BUILD_ASAN=1 compiles
# the extension with address sanitizer where ROCm supports it.
if os.environ.get("SPARKDL_BUILD_ASAN") == "1":
    _HIP_FLAGS += ["-fsanitize=address", "-shared-libsan", "-g"]
    _CXX_FLAGS += ["-fsanitize=address", "-g"]


def _sources():
    out = [os.path.join(CSRC, "bindings.cpp")]
    for f in sorted(os.listdir(CSRC)):
        if f.endswith(".hip"):
            out.append(os.path.join(CSRC, f))
    return out


def make_extensions():
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import CUDAExtension
    return [
        CUDAExtension(
            name="sparkdl._C",
            sources=_sources(),
            extra_compile_args={"cxx": _CXX_FLAGS, "nvcc": _HIP_FLAGS},
        )
    ]


def make_build_ext():
    from torch.utils.cpp_extension import BuildExtension
    return BuildExtension.with_options(use_ninja=True)


def build_inplace():
    """Compile sparkdl._C into the source tree (the graft build check)."""
    import subprocess
    import sys
    # CSRC = <repo>/sparkdl/ops/csrc — three levels below the repo root.
    repo_root = os.path.dirname(os.path.dirname(os.path.dirname(CSRC)))
    env = dict(os.environ, PYTORCH_ROCM_ARCH="gfx950")
    subprocess.check_call(
        [sys.executable, "setup.py", "build_ext", "--inplace"],
        cwd=repo_root, env=env)
