"""In-tree build of the gfx950 HIP extension.

The .so is built into ``distributed_rl_amd/ops/_build/`` (tracked by the
snapshot that travels to GPU boxes, git-ignored) so GPU runs never JIT into
an off-tree cache. hipcc cross-compiles fine on GPU-less hosts.
"""

from __future__ import annotations

import os
import sys

EXT_NAME = "_drl_hip"
_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
BUILD_DIR = os.path.join(_THIS_DIR, "_build")
SOURCES = [
    os.path.join(_THIS_DIR, "hip", "drl_kernels.hip"),
    os.path.join(_THIS_DIR, "hip", "conv_mfma.hip"),
]


def so_path() -> str:
    return os.path.join(BUILD_DIR, f"{EXT_NAME}.so")


def build(verbose: bool = False):
    """Compile the extension for gfx950 (idempotent; ninja skips clean builds)."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils import cpp_extension

    mod = cpp_extension.load(
        name=EXT_NAME,
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
        with_cuda=True,
        is_python_module=True,
    )
    return mod


def load_prebuilt():
    """dlopen the in-tree .so without invoking the build system."""
    import importlib.machinery
    import importlib.util

    path = so_path()
    if not os.path.exists(path):
        return None
    loader = importlib.machinery.ExtensionFileLoader(EXT_NAME, path)
    spec = importlib.util.spec_from_loader(EXT_NAME, loader)
    mod = importlib.util.module_from_spec(spec)
    loader.exec_module(mod)
    sys.modules[EXT_NAME] = mod
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("built", so_path())
