"""Model state: init distribution, views, checkpoint format."""
import numpy as np
import pytest
import torch

from parallel_cnn_amd.models.lenet import LeNet5
from parallel_cnn_amd.ops import shapes as S


def test_reference_init_distribution():
    m = LeNet5(seed=1)
    p = m.params
    assert p.shape == (S.N_PARAMS,)
    assert p.min() > -0.5 - 1e-6 and p.max() <= 0.5 + 1e-6
    assert abs(p.mean().item()) < 0.05  # roughly centered


def test_init_seeded_deterministic():
    assert torch.equal(LeNet5(seed=3).params, LeNet5(seed=3).params)
    assert not torch.equal(LeNet5(seed=3).params, LeNet5(seed=4).params)


def test_views_alias_flat_params():
    m = LeNet5(seed=0)
    m.c1_weight.zero_()
    assert m.params[S.OFF_C1W:S.OFF_C1B].abs().sum() == 0
    m.f_bias.fill_(2.0)
    assert (m.params[S.OFF_FB:] == 2.0).all()
    assert m.s1_weight.shape == (4, 4)
    assert m.f_weight.shape == (10, 216)


def test_checkpoint_roundtrip(tmp_path):
    m = LeNet5(seed=9)
    path = str(tmp_path / "w.bin")
    m.save(path)
    # format: exactly 2343 little-endian float32
    raw = np.fromfile(path, dtype="<f4")
    assert raw.size == S.N_PARAMS
    m2 = LeNet5(seed=0)
    m2.load(path)
    assert torch.equal(m.params, m2.params)


def test_checkpoint_wrong_size(tmp_path):
    path = str(tmp_path / "bad.bin")
    np.zeros(10, dtype="<f4").tofile(path)
    with pytest.raises(ValueError, match="expected"):
        LeNet5().load(path)
