"""MNIST IDX loader + synthetic data tests."""
import struct

import numpy as np
import pytest
import torch

from parallel_cnn_amd.data.mnist import (load_idx_images, load_idx_labels,
                                         load_mnist, synthetic_mnist)


def write_idx(tmp_path, n=10):
    rng = np.random.default_rng(0)
    imgs = rng.integers(0, 256, size=(n, 28, 28), dtype=np.uint8)
    lbls = rng.integers(0, 10, size=n, dtype=np.uint8)
    ip = tmp_path / "imgs.idx3-ubyte"
    lp = tmp_path / "lbls.idx1-ubyte"
    with open(ip, "wb") as f:
        f.write(struct.pack(">iiii", 2051, n, 28, 28))
        f.write(imgs.tobytes())
    with open(lp, "wb") as f:
        f.write(struct.pack(">ii", 2049, n))
        f.write(lbls.tobytes())
    return ip, lp, imgs, lbls


def test_idx_roundtrip(tmp_path):
    ip, lp, imgs, lbls = write_idx(tmp_path)
    x = load_idx_images(str(ip))
    y = load_idx_labels(str(lp))
    assert x.shape == (10, 784) and x.dtype == np.float32
    assert np.allclose(x, imgs.reshape(10, 784).astype(np.float32) / 255.0)
    assert (y == lbls).all()
    xt, yt = load_mnist(str(ip), str(lp))
    assert isinstance(xt, torch.Tensor) and xt.shape == (10, 784)
    assert yt.dtype == torch.int64


def test_idx_bad_magic(tmp_path):
    p = tmp_path / "bad"
    with open(p, "wb") as f:
        f.write(struct.pack(">iiii", 1234, 1, 28, 28))
        f.write(b"\0" * 784)
    with pytest.raises(ValueError, match="magic"):
        load_idx_images(str(p))


def test_idx_truncated(tmp_path):
    p = tmp_path / "trunc"
    with open(p, "wb") as f:
        f.write(struct.pack(">iiii", 2051, 10, 28, 28))
        f.write(b"\0" * 100)
    with pytest.raises(ValueError, match="truncated"):
        load_idx_images(str(p))


def test_synthetic_deterministic_and_ranged():
    x1, y1 = synthetic_mnist(100, seed=5)
    x2, y2 = synthetic_mnist(100, seed=5)
    assert torch.equal(x1, x2) and torch.equal(y1, y2)
    assert x1.min() >= 0 and x1.max() <= 1
    assert y1.min() >= 0 and y1.max() <= 9
    x3, _ = synthetic_mnist(100, seed=6)
    assert not torch.equal(x1, x3)
