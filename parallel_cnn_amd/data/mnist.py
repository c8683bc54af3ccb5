"""MNIST IDX loader + synthetic data.

Fresh implementation of the public IDX format (big-endian headers, magic
2051 for images / 2049 for labels) with the reference's semantics: pixels
scaled to [0, 1] fp32 (the MNIST_DOUBLE path of the reference's loader,
SURVEY.md §1 "Data layer").  The reference snapshot is missing the image
blobs, and the benchmark measures on synthetic data anyway, so
`synthetic_mnist` generates 28x28 images of the same shape/range.
"""
from __future__ import annotations

import struct
from typing import Tuple

import numpy as np
import torch

IMAGE_MAGIC = 2051
LABEL_MAGIC = 2049


def load_idx_images(path: str) -> np.ndarray:
    """Returns fp32 [N, 28*28] in [0, 1]."""
    with open(path, "rb") as f:
        magic, n, rows, cols = struct.unpack(">iiii", f.read(16))
        if magic != IMAGE_MAGIC:
            raise ValueError(f"{path}: bad image magic {magic}")
        raw = np.frombuffer(f.read(n * rows * cols), dtype=np.uint8)
    if raw.size != n * rows * cols:
        raise ValueError(f"{path}: truncated image data")
    return (raw.astype(np.float32) / 255.0).reshape(n, rows * cols)


def load_idx_labels(path: str) -> np.ndarray:
    """Returns int64 [N]."""
    with open(path, "rb") as f:
        magic, n = struct.unpack(">ii", f.read(8))
        if magic != LABEL_MAGIC:
            raise ValueError(f"{path}: bad label magic {magic}")
        raw = np.frombuffer(f.read(n), dtype=np.uint8)
    if raw.size != n:
        raise ValueError(f"{path}: truncated label data")
    return raw.astype(np.int64)


def load_mnist(images_path: str, labels_path: str
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    x = torch.from_numpy(load_idx_images(images_path))
    y = torch.from_numpy(load_idx_labels(labels_path))
    if x.shape[0] != y.shape[0]:
        raise ValueError("image/label count mismatch")
    return x, y


def synthetic_images(n: int, h: int, w: int, c: int, seed: int = 0,
                     structured: bool = True, n_classes: int = 10
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """General synthetic image set: fp32 [n, h*w*c] (NHWC flat) in [0,1] +
    labels.  structured=True adds a label-dependent bright band."""
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, h, w, c, generator=g)
    y = torch.randint(0, n_classes, (n,), generator=g)
    if structured:
        x *= 0.3
        band = max(1, h // (n_classes + 2))
        for lbl in range(n_classes):
            rows = slice(1 + lbl * band, 1 + lbl * band + band)
            x[y == lbl, rows, :, :] += 0.7
        x.clamp_(0, 1)
    return x.reshape(n, h * w * c), y


def synthetic_mnist(n: int, seed: int = 0, structured: bool = True
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Synthetic 28x28x1 images in [0,1] + labels in [0,10).

    With structured=True each image carries a simple label-dependent pattern
    (a bright band whose position encodes the label) on top of noise, so
    convergence tests have something learnable; structured=False is pure
    noise for pure-throughput benchmarking.
    """
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, 28 * 28, generator=g)
    y = torch.randint(0, 10, (n,), generator=g)
    if structured:
        xi = x.view(n, 28, 28)
        xi *= 0.3
        for lbl in range(10):
            rows = slice(2 + lbl * 2, 2 + lbl * 2 + 3)
            mask = y == lbl
            xi[mask, rows, :] += 0.7
        x = xi.reshape(n, 28 * 28).clamp_(0, 1)
    return x, y
