"""Config system tests."""
import argparse

import pytest

from parallel_cnn_amd.config import TrainConfig


def test_defaults_match_reference():
    cfg = TrainConfig()
    assert cfg.dt == 0.1
    assert cfg.threshold == 1e-2
    assert cfg.epochs == 1


def test_yaml_roundtrip(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("batch_size: 128\ndt: 0.05\nbackend: cpu\n")
    cfg = TrainConfig.from_yaml(str(p))
    assert cfg.batch_size == 128
    assert cfg.dt == 0.05
    assert cfg.backend == "cpu"


def test_yaml_unknown_key(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("not_a_key: 1\n")
    with pytest.raises(ValueError, match="unknown config keys"):
        TrainConfig.from_yaml(str(p))


def test_cli_overrides_yaml(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("batch_size: 128\n")
    ap = argparse.ArgumentParser()
    TrainConfig.add_cli_args(ap)
    args = ap.parse_args(["--config", str(p), "--batch-size", "256",
                          "--dt", "0.2"])
    cfg = TrainConfig.from_args(args)
    assert cfg.batch_size == 256
    assert cfg.dt == 0.2


def test_cli_bool_false_is_false():
    """Regression: bool flags used to be re-coerced through str(), so
    `--overlap-comm false` yielded the truthy string 'False'."""
    ap = argparse.ArgumentParser()
    TrainConfig.add_cli_args(ap)
    args = ap.parse_args(["--overlap-comm", "false",
                          "--fuse-wgrad", "false"])
    cfg = TrainConfig.from_args(args)
    assert cfg.overlap_comm is False
    assert cfg.fuse_wgrad is False
    args = ap.parse_args(["--overlap-comm", "true"])
    cfg = TrainConfig.from_args(args)
    assert cfg.overlap_comm is True


def test_cli_bool_matches_yaml(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("overlap_comm: true\nfuse_wgrad: false\n")
    ap = argparse.ArgumentParser()
    TrainConfig.add_cli_args(ap)
    # CLI overrides YAML; explicit false must win over a YAML true
    args = ap.parse_args(["--config", str(p), "--overlap-comm", "false"])
    cfg = TrainConfig.from_args(args)
    assert cfg.overlap_comm is False
    assert cfg.fuse_wgrad is False


def test_resolution():
    cfg = TrainConfig(device="cpu", backend="auto")
    assert cfg.resolved_backend() == "cpu"
    cfg2 = TrainConfig(device="cpu", backend="torchref")
    assert cfg2.resolved_backend() == "torchref"
