"""Offline floorplan -> signed-distance-field preprocessing.

Capability parity with the reference's ``floorplans/cubi_preproc.py``
(SDFTransform: edge detection + euclidean distance transform -> signed
distance field, plus a train/test split file). The reference uses
skimage's Canny; this environment has scipy only, so edges come from a
gradient-magnitude detector (equivalent input to the distance
transform for binary floorplans).

Usage:
    python -m nn_distributed_training_amd.data.sdf_preproc \
        --out ./sdf_data --count 8
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import scipy.ndimage as ndi

from .floorplan import synthetic_floorplan


def edge_map(img: np.ndarray, thresh: float = 0.25) -> np.ndarray:
    """Binary wall-boundary map via gradient magnitude."""
    gx = ndi.sobel(img.astype(float), axis=0)
    gy = ndi.sobel(img.astype(float), axis=1)
    mag = np.hypot(gx, gy)
    if mag.max() > 0:
        mag = mag / mag.max()
    return mag > thresh


def sdf_transform(img: np.ndarray) -> np.ndarray:
    """Signed distance field: positive outside walls, negative inside,
    zero on wall boundaries (parity with SDFTransform,
    cubi_preproc.py:11-34)."""
    edges = edge_map(img)
    dist = ndi.distance_transform_edt(~edges)
    sign = np.where(img >= 0.5, -1.0, 1.0)
    return sign * dist


def preprocess(out_dir: str, count: int = 8, size: int = 256,
               train_frac: float = 0.8, seed: int = 0):
    """Generate floorplans, save (img, sdf) pairs and a train/test
    split file (parity with cubi_preprocess, cubi_preproc.py:37-92)."""
    os.makedirs(out_dir, exist_ok=True)
    rng = np.random.default_rng(seed)
    names = []
    for k in range(count):
        img = synthetic_floorplan(nx=size, ny=size, seed=seed + k)
        sdf = sdf_transform(img)
        name = f"floorplan_{k:03d}"
        np.savez(
            os.path.join(out_dir, name + ".npz"), img=img, sdf=sdf
        )
        names.append(name)
    perm = rng.permutation(count)
    ntr = int(round(train_frac * count))
    split = {
        "train": [names[i] for i in perm[:ntr]],
        "test": [names[i] for i in perm[ntr:]],
    }
    with open(os.path.join(out_dir, "split.txt"), "w") as f:
        for part in ("train", "test"):
            for n in split[part]:
                f.write(f"{part} {n}\n")
    return split


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="./sdf_data")
    p.add_argument("--count", type=int, default=8)
    p.add_argument("--size", type=int, default=256)
    args = p.parse_args()
    split = preprocess(args.out, args.count, args.size)
    print(f"wrote {args.count} SDFs to {args.out}; "
          f"{len(split['train'])} train / {len(split['test'])} test")
