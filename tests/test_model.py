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


def test_checkpoint_sidecar_roundtrip(tmp_path):
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.utils.checkpoint import (load_checkpoint,
                                                   save_checkpoint)
    from parallel_cnn_amd.data.mnist import synthetic_mnist
    cfg = TrainConfig(device="cpu", backend="cpu", batch_size=8,
                      log_interval=0)
    t = Trainer(cfg)
    x, y = synthetic_mnist(16, seed=1)
    t.step(*t.stage_batch(x[:8], y[:8]))
    t.step(*t.stage_batch(x[8:], y[8:]))
    path = str(tmp_path / "ck.bin")
    save_checkpoint(t, path)
    t2 = Trainer(cfg)
    meta = load_checkpoint(t2, path)
    assert torch.equal(t.model.params, t2.model.params)
    assert t2.global_step == 2
    assert meta["model"] == "lenet5"


def test_checkpoint_sidecar_wrong_model(tmp_path):
    import json
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.utils.checkpoint import (load_checkpoint,
                                                   save_checkpoint)
    cfg = TrainConfig(device="cpu", backend="cpu", batch_size=8,
                      log_interval=0)
    t = Trainer(cfg)
    path = str(tmp_path / "ck.bin")
    save_checkpoint(t, path)
    with open(path + ".meta.json") as f:
        meta = json.load(f)
    meta["n_params"] = 999
    with open(path + ".meta.json", "w") as f:
        json.dump(meta, f)
    with pytest.raises(ValueError, match="999"):
        load_checkpoint(Trainer(cfg), path)


def test_checkpoint_sidecar_config_mismatch(tmp_path):
    """ADVICE regression: a pool=max / loss=softmax_ce checkpoint must not
    load silently into a trainable/residual config of the same size."""
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.utils.checkpoint import (load_checkpoint,
                                                   save_checkpoint)
    cfg = TrainConfig(device="cpu", backend="cpu", batch_size=8,
                      log_interval=0, pool="max", loss="softmax_ce")
    t = Trainer(cfg)
    path = str(tmp_path / "ck.bin")
    save_checkpoint(t, path)
    other = TrainConfig(device="cpu", backend="cpu", batch_size=8,
                        log_interval=0)  # trainable/residual defaults
    with pytest.raises(ValueError, match="pool"):
        load_checkpoint(Trainer(other), path)


def test_exact_resume_equivalence(tmp_path):
    """train(2 epochs) == train(1) -> save -> load -> train(1): the sidecar
    persists the epoch cursor and RNG state, so resume continues the exact
    trajectory."""
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.utils.checkpoint import (load_checkpoint,
                                                   save_checkpoint)
    from parallel_cnn_amd.data.mnist import synthetic_mnist
    x, y = synthetic_mnist(64, seed=2)
    cfg = TrainConfig(device="cpu", backend="cpu", batch_size=16,
                      log_interval=0, epochs=2)
    full = Trainer(cfg)
    full.train_epoch(x, y, log=lambda *a: None)
    full.train_epoch(x, y, log=lambda *a: None)

    half = Trainer(cfg)
    half.train_epoch(x, y, log=lambda *a: None)
    path = str(tmp_path / "resume.bin")
    save_checkpoint(half, path)

    resumed = Trainer(cfg)
    meta = load_checkpoint(resumed, path)
    assert meta["epoch"] == 1 and resumed.epoch == 1
    assert "rng" in meta and "torch_cpu" in meta["rng"]
    resumed.train_epoch(x, y, log=lambda *a: None)
    assert torch.equal(full.model.params, resumed.model.params)
    assert resumed.epoch == full.epoch == 2
    assert resumed.global_step == full.global_step


def test_resume_restores_rng_stream(tmp_path):
    """RNG draws after load reproduce the draws after save."""
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.utils.checkpoint import (load_checkpoint,
                                                   save_checkpoint)
    cfg = TrainConfig(device="cpu", backend="cpu", log_interval=0)
    t = Trainer(cfg)
    torch.manual_seed(77)
    torch.rand(3)  # advance the stream
    path = str(tmp_path / "r.bin")
    save_checkpoint(t, path)
    expect = torch.rand(4)
    torch.manual_seed(0)  # clobber
    load_checkpoint(Trainer(cfg), path)
    assert torch.equal(torch.rand(4), expect)
