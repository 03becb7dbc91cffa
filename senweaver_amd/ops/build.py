"""In-tree build of the gfx950 HIP extension.

Compiles every kernel with hipcc for --offload-arch=gfx950 via
torch.utils.cpp_extension and leaves the .so inside the repo
(senweaver_amd/ops/_build/) so it travels to GPU boxes with the source
snapshot.  No JIT cache outside the tree, no multi-arch fatbins.
"""

from __future__ import annotations

import os

_HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(_HERE, "csrc")
BUILD_DIR = os.path.join(_HERE, "_build")
EXT_NAME = "senweaver_amd_hip"

SOURCES = [
    os.path.join(CSRC, f)
    for f in ("bindings.hip", "elemwise.hip", "sampling.hip", "gemm.hip",
              "gemm_pipe.hip", "gemm_asm.hip", "attention.hip", "attention_v2.hip",
              "attention_v4.hip",
              "decode_attention.hip", "moe.hip", "fp8.hip")
]


def build(verbose: bool = False):
    """Compile (if stale) and load the extension module."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    return load(
        name=EXT_NAME,
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        with_cuda=True,  # drives hipcc on a ROCm torch build
    )


def load_prebuilt():
    """Load the already-built .so without invoking the compiler (GPU box path).

    Refuses a STALE .so (any csrc source or header newer than the binary):
    returning None makes the caller fall through to build(), whose ninja
    dependency check recompiles.  Without this check an edited kernel that
    failed to compile would silently keep running the old binary.
    """
    so = os.path.join(BUILD_DIR, EXT_NAME + ".so")
    if not os.path.exists(so):
        return None
    so_mtime = os.path.getmtime(so)
    for f in os.listdir(CSRC):
        if f.endswith((".hip", ".h", ".cuh")):
            if os.path.getmtime(os.path.join(CSRC, f)) > so_mtime:
                return None  # stale -> rebuild path
    import importlib.util

    spec = importlib.util.spec_from_file_location(EXT_NAME, so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("built", EXT_NAME, "->", BUILD_DIR)
