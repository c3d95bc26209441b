"""In-tree build of the CDNA4 HIP extension.

The shared object is built into ``ops/_build`` INSIDE the package (not a
home-directory JIT cache) so it travels with any snapshot of the repo to
a GPU box. gfx950-only by design: ``PYTORCH_ROCM_ARCH=gfx950`` is forced,
there are no other offload targets and no CUDA fallback.
"""

from __future__ import annotations

import glob
import hashlib
import os
import shutil

_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
_BUILD_DIR = os.path.join(_THIS_DIR, "_build")
_SRC = [os.path.join(_THIS_DIR, "hip", "ext.hip")]
_EXT_NAME = "ndta_hip_ext"
_HASH_STAMP = os.path.join(_BUILD_DIR, "source_hash.txt")


def _flags():
    extra = ["-O3", "--offload-arch=gfx950"]
    if os.environ.get("NDTA_BUILD_DEFINES"):
        extra += os.environ["NDTA_BUILD_DEFINES"].split()
    return extra


def source_hash() -> str:
    """Content hash of every kernel source + build flags.

    A snapshot of this repo can carry a prebuilt ``_build/*.so`` (it
    travels to GPU boxes so tests don't pay a cold compile every call);
    this hash is the proof that the binary corresponds to the committed
    sources: ninja's mtime check is unreliable across snapshot copies,
    so :func:`build` wipes ``_build`` and recompiles whenever the stored
    stamp doesn't match the current sources.
    """
    h = hashlib.sha256()
    srcs = sorted(
        p
        for p in glob.glob(os.path.join(_THIS_DIR, "hip", "*"))
        if p.endswith((".hip", ".h", ".hpp"))
        and not p.endswith("_hip.hip")  # hipify passthrough artifact
    )
    for p in srcs:
        h.update(os.path.basename(p).encode())
        with open(p, "rb") as f:
            h.update(f.read())
    h.update(" ".join(_flags()).encode())
    return h.hexdigest()


def build(verbose: bool = False):
    """Compile (if stale) and return the extension module.

    Staleness is judged by source CONTENT hash, not mtimes: a prebuilt
    binary that doesn't match the current sources is discarded and the
    extension is recompiled from scratch (provable source->binary
    correspondence).
    """
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    want = source_hash()
    have = None
    if os.path.exists(_HASH_STAMP):
        with open(_HASH_STAMP) as f:
            have = f.read().strip()
    if have != want and os.path.isdir(_BUILD_DIR):
        shutil.rmtree(_BUILD_DIR)
    os.makedirs(_BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    mod = load(
        name=_EXT_NAME,
        sources=_SRC,
        build_directory=_BUILD_DIR,
        extra_cuda_cflags=_flags(),
        extra_cflags=["-O3"],
        verbose=verbose,
    )
    with open(_HASH_STAMP, "w") as f:
        f.write(want)
    return mod
