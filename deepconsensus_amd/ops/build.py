"""In-tree build of the HIP/CDNA4 extensions (gfx950 only).

Builds with torch.utils.cpp_extension driving hipcc under
PYTORCH_ROCM_ARCH=gfx950, into deepconsensus_amd/ops/_build so the .so
travels with the repo snapshot to GPU boxes (no JIT cache dependence).
"""
from __future__ import annotations

import os
import shutil

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
_BUILD_DIR = os.path.join(_OPS_DIR, "_build")
_SRCS = [
    os.path.join(_OPS_DIR, "hip", "dc_kernels.hip"),
    os.path.join(_OPS_DIR, "hip", "banded_attn.hip"),
    os.path.join(_OPS_DIR, "hip", "banded_attn_mfma.hip"),
    os.path.join(_OPS_DIR, "hip", "embed_gather.hip"),
    os.path.join(_OPS_DIR, "hip", "fused_ffn.hip"),
    os.path.join(_OPS_DIR, "hip", "alignment_dp.hip"),
    os.path.join(_OPS_DIR, "hip", "fused_linear.hip"),
    os.path.join(_OPS_DIR, "hip", "fused_ffn_glds.hip"),
    os.path.join(_OPS_DIR, "hip", "fused_ffn_v3.hip"),
    os.path.join(_OPS_DIR, "hip", "embed_grad.hip"),
    os.path.join(_OPS_DIR, "hip", "alignment_metric.hip"),
    os.path.join(_OPS_DIR, "hip", "ffn_ablate.hip"),
    os.path.join(_OPS_DIR, "hip", "fused_ffn_v4.hip"),
    os.path.join(_OPS_DIR, "hip", "banded_attn_train.hip"),
    os.path.join(_OPS_DIR, "hip", "banded_attn_bwd_mfma.hip"),
    os.path.join(_OPS_DIR, "hip", "fused_condense.hip"),
    os.path.join(_OPS_DIR, "hip", "ffn_train.hip"),
    os.path.join(_OPS_DIR, "hip", "resid_dropout.hip"),
]
EXT_NAME = "dc_hip_kernels"


_SPACING_SRC = os.path.join(
    os.path.dirname(_OPS_DIR), "preprocess", "_spacing_cpp.cpp"
)


def build_spacing(verbose: bool = False):
    """Compiles (if needed) and loads the CPU _spacing extension."""
    os.makedirs(_BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    return load(
        name="_spacing",
        sources=[_SPACING_SRC],
        build_directory=_BUILD_DIR,
        extra_cflags=["-O3"],
        verbose=verbose,
    )


def build(verbose: bool = False):
    """Compiles (if needed) and loads the dc_hip_kernels extension."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(_BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    build_spacing(verbose=verbose)
    module = load(
        name=EXT_NAME,
        sources=_SRCS,
        build_directory=_BUILD_DIR,
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
    )
    return module


def build_asan(verbose: bool = False):
    """Device-AddressSanitizer build (SURVEY section 5.2 sanitizer pass).

    Compiles the same kernels for gfx950:xnack+ with -fsanitize=address
    into _build_asan/dc_hip_kernels_asan.so. Run on a GPU box with
    HSA_XNACK=1 and the host ASAN runtime preloaded:
      LD_PRELOAD=$(hipcc -print-file-name=libclang_rt.asan-x86_64.so)
      ASAN_OPTIONS=detect_leaks=0 HSA_XNACK=1 python ...
    """
    os.environ["PYTORCH_ROCM_ARCH"] = "gfx950:xnack+"
    build_dir = os.path.join(_OPS_DIR, "_build_asan")
    os.makedirs(build_dir, exist_ok=True)
    from torch.utils.cpp_extension import load

    flags = ["-O2", "-fsanitize=address", "-shared-libsan"]
    module = load(
        name=EXT_NAME + "_asan",
        sources=_SRCS,
        build_directory=build_dir,
        extra_cuda_cflags=flags,
        # The link step runs plain c++ (no -shared-libsan there); the
        # __asan_* refs stay undefined in the .so and resolve from the
        # LD_PRELOADed clang runtime at load time.
        extra_ldflags=[],
        verbose=verbose,
    )
    return module


def clean():
    if os.path.isdir(_BUILD_DIR):
        shutil.rmtree(_BUILD_DIR)


if __name__ == "__main__":
    build(verbose=True)
    print("built", EXT_NAME, "in", _BUILD_DIR)
