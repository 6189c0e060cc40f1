"""HIP extension build & load machinery (gfx950).

The extension is built IN-TREE (r2d2_amd/ops/hip/_build/r2d2_hip.so) so the
.so travels to the GPU box with the repo snapshot.  ``build()`` cross-compiles
on the CPU container via torch.utils.cpp_extension (hipcc, ninja,
PYTORCH_ROCM_ARCH=gfx950).  ``load()`` imports the prebuilt .so without
recompiling; on a GPU box with no .so it raises — ops must never silently
fall back to eager on a GPU (the CPU/test path selects eager explicitly).
"""

import importlib.util
import os
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))
SRC_DIR = os.path.join(_HERE, "hip")
BUILD_DIR = os.path.join(SRC_DIR, "_build")
EXT_NAME = "r2d2_hip"

SOURCES = [
    os.path.join(SRC_DIR, "bindings.cpp"),
    os.path.join(SRC_DIR, "loss_kernels.hip"),
    os.path.join(SRC_DIR, "sumtree.hip"),
    os.path.join(SRC_DIR, "replay_gather.hip"),
    os.path.join(SRC_DIR, "lstm_kernels.hip"),
    os.path.join(SRC_DIR, "gemm_kernels.hip"),
    os.path.join(SRC_DIR, "conv_kernels.hip"),
    os.path.join(SRC_DIR, "impala_kernels.hip"),
    os.path.join(SRC_DIR, "optim_kernels.hip"),
]

_module = None


def _so_path():
    return os.path.join(BUILD_DIR, f"{EXT_NAME}.so")


def build(verbose: bool = False):
    """Compile the extension for gfx950 (works without a GPU)."""
    global _module
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils import cpp_extension
    sources = [s for s in SOURCES if os.path.exists(s)]
    _module = cpp_extension.load(
        name=EXT_NAME,
        sources=sources,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950"],
        verbose=verbose,
    )
    return _module


def load(required: bool = False):
    """Import the prebuilt extension; None if absent and not required."""
    global _module
    if _module is not None:
        return _module
    so = _so_path()
    if not os.path.exists(so):
        if required:
            raise RuntimeError(
                f"HIP extension not built ({so} missing). Run "
                f"__graft_entry__.build() (hipcc cross-compiles without a GPU).")
        return None
    import torch  # noqa: F401  (extension links against torch libs)
    spec = importlib.util.spec_from_file_location(EXT_NAME, so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.modules[EXT_NAME] = mod
    _module = mod
    return _module
