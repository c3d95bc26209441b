"""MNIST data loading with an offline synthetic fallback.

The reference downloads MNIST via torchvision
(experiments/dist_mnist_ex.py:98-105). This environment has no network and
no torchvision, so the default source is a *synthetic* class-conditional
image distribution of the same shape and scale (10 classes, 1x28x28,
normalized with the MNIST mean/std). Each class has a fixed smooth
prototype image (seeded); samples are the prototype plus pixel noise, so
the classification task is learnable and convergence curves are
meaningful. ``data_source: torchvision`` can be requested in the YAML and
is used when the package + files are available.
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset, Subset, TensorDataset, random_split

MNIST_MEAN, MNIST_STD = 0.1307, 0.3081


class SyntheticMNIST(TensorDataset):
    """Tensor dataset of (image [1,28,28], label) with MNIST statistics."""

    def __init__(self, num_samples: int, seed: int, noise: float = 0.35,
                 proto_seed: int = 1234):
        # class prototypes are fixed by proto_seed so train/val (different
        # sample seeds) share one underlying task
        protos = self._prototypes(
            torch.Generator().manual_seed(proto_seed)
        )
        g = torch.Generator().manual_seed(seed)
        labels = torch.randint(0, 10, (num_samples,), generator=g)
        imgs = protos[labels]
        imgs = imgs + noise * torch.randn(imgs.shape, generator=g)
        imgs = imgs.clamp_(0.0, 1.0)
        imgs = (imgs - MNIST_MEAN) / MNIST_STD
        imgs = imgs.unsqueeze(1).to(torch.get_default_dtype())
        super().__init__(imgs, labels)
        self.targets = labels  # same attribute torchvision exposes

    @staticmethod
    def _prototypes(g: torch.Generator) -> torch.Tensor:
        """10 fixed smooth class prototypes in [0, 1], shape [10, 28, 28]."""
        raw = torch.rand(10, 1, 7, 7, generator=g)
        up = torch.nn.functional.interpolate(
            raw, size=(28, 28), mode="bilinear", align_corners=False
        ).squeeze(1)
        # stretch contrast so classes are well separated
        up = (up - up.amin(dim=(1, 2), keepdim=True)) / (
            up.amax(dim=(1, 2), keepdim=True)
            - up.amin(dim=(1, 2), keepdim=True)
        )
        return up


def _read_idx(path):
    """Parse an IDX file (the official MNIST distribution format;
    supports plain and .gz). First-party so REAL MNIST files work
    without torchvision (not installed in this image)."""
    import gzip
    import os
    import struct

    opener = gzip.open if path.endswith(".gz") else open
    if not os.path.exists(path) and os.path.exists(path + ".gz"):
        path, opener = path + ".gz", gzip.open
    with opener(path, "rb") as f:
        magic = struct.unpack(">I", f.read(4))[0]
        dtype_code = (magic >> 8) & 0xFF
        ndim = magic & 0xFF
        if dtype_code != 0x08:
            raise ValueError(f"unsupported IDX dtype 0x{dtype_code:x}")
        dims = [struct.unpack(">I", f.read(4))[0] for _ in range(ndim)]
        data = f.read()
    arr = torch.frombuffer(
        bytearray(data), dtype=torch.uint8
    ).reshape(dims)
    return arr


def load_mnist_idx(data_dir: str):
    """(train_set, val_set) from the standard MNIST IDX files in
    ``data_dir`` (train-images-idx3-ubyte[.gz] etc. — the layout the
    official distribution and torchvision's raw/ directory both use),
    normalized exactly like the reference's torchvision pipeline."""
    import os

    def find(stem):
        for cand in (
            stem,
            os.path.join("MNIST", "raw", stem),
            os.path.join("raw", stem),
        ):
            p = os.path.join(data_dir, cand)
            if os.path.exists(p) or os.path.exists(p + ".gz"):
                return p
        raise FileNotFoundError(f"{stem} not under {data_dir}")

    sets = []
    for img_stem, lbl_stem in (
        ("train-images-idx3-ubyte", "train-labels-idx1-ubyte"),
        ("t10k-images-idx3-ubyte", "t10k-labels-idx1-ubyte"),
    ):
        imgs = _read_idx(find(img_stem)).to(torch.get_default_dtype())
        labels = _read_idx(find(lbl_stem)).to(torch.long)
        imgs = imgs / 255.0
        imgs = (imgs - MNIST_MEAN) / MNIST_STD
        ds = TensorDataset(imgs.unsqueeze(1), labels)
        ds.targets = labels
        sets.append(ds)
    return sets[0], sets[1]


def load_mnist(
    data_dir: str,
    source: str = "synthetic",
    train_samples: int = 60000,
    val_samples: int = 10000,
    seed: int = 0,
):
    """Return (train_set, val_set). Both expose ``.targets``.

    Sources: ``synthetic`` (default — no files needed),
    ``idx_files`` (real MNIST from the official IDX files, first-party
    parser), ``torchvision`` (tries torchvision, then the IDX files,
    then synthetic).
    """
    if source == "idx_files":
        return load_mnist_idx(data_dir)
    if source == "torchvision":
        try:
            from torchvision import datasets, transforms

            tfm = transforms.Compose(
                [
                    transforms.ToTensor(),
                    transforms.Normalize((MNIST_MEAN,), (MNIST_STD,)),
                ]
            )
            train = datasets.MNIST(
                data_dir, train=True, download=True, transform=tfm
            )
            val = datasets.MNIST(data_dir, train=False, transform=tfm)
            return train, val
        except Exception as e:  # pragma: no cover - depends on env
            try:  # torchvision absent: the IDX files may still be here
                return load_mnist_idx(data_dir)
            except Exception:
                pass
            print(f"torchvision MNIST unavailable ({e}); using synthetic.")
    train = SyntheticMNIST(train_samples, seed=seed)
    val = SyntheticMNIST(val_samples, seed=seed + 1)
    return train, val


def split_train_set(train_set: Dataset, N: int, split_type: str):
    """Partition the train set across N nodes.

    Parity with the reference driver (experiments/dist_mnist_ex.py:107-127):
      random — N equal random subsets;
      hetero — the 10 digit classes partitioned contiguously over nodes
               (requires N <= 10). Quirk kept for parity: with N not
               dividing 10, the trailing `10 - N*(10//N)` classes are
               assigned to NO node (the reference's torch.split +
               first-N-chunks loop does the same — its paper configs
               use N = 10);
      hetero_sorted — label-sorted chunks for arbitrary N (the scaling
               driver's variant, experiments/dist_mnist_scaling.py:122-129).
    """
    targets = train_set.targets
    if not torch.is_tensor(targets):
        targets = torch.as_tensor(targets)

    if split_type == "random":
        per = len(targets) // N
        sizes = [per] * N
        sizes[-1] += len(targets) - per * N
        return list(random_split(train_set, sizes))

    if split_type == "hetero":
        classes = torch.unique(targets)
        if N > len(classes):
            raise NameError("Hetero MNIST N > 10 not supported.")
        node_classes = torch.split(classes, len(classes) // N)
        subsets = []
        for i in range(N):
            mask = torch.isin(targets, node_classes[i])
            idx = torch.nonzero(mask).reshape(-1)
            subsets.append(Subset(train_set, idx))
        return subsets

    if split_type == "hetero_sorted":
        order = torch.argsort(targets)
        chunks = torch.chunk(order, N)
        return [Subset(train_set, c) for c in chunks]

    raise NameError(f"Unknown data split type: {split_type}")
