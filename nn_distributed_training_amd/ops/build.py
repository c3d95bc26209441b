"""In-tree build of the CDNA4 HIP extension.

The shared object is built into ``ops/_build`` INSIDE the package (not a
home-directory JIT cache) so it travels with any snapshot of the repo to
a GPU box. gfx950-only by design: ``PYTORCH_ROCM_ARCH=gfx950`` is forced,
there are no other offload targets and no CUDA fallback.
"""

from __future__ import annotations

import os

_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
_BUILD_DIR = os.path.join(_THIS_DIR, "_build")
_SRC = [os.path.join(_THIS_DIR, "hip", "ext.hip")]
_EXT_NAME = "ndta_hip_ext"


def build(verbose: bool = False):
    """Compile (if stale) and return the extension module."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(_BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    extra = ["-O3", "--offload-arch=gfx950"]
    if os.environ.get("NDTA_BUILD_DEFINES"):
        extra += os.environ["NDTA_BUILD_DEFINES"].split()
    return load(
        name=_EXT_NAME,
        sources=_SRC,
        build_directory=_BUILD_DIR,
        extra_cuda_cflags=extra,
        extra_cflags=["-O3"],
        verbose=verbose,
    )
