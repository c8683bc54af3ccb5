"""End-to-end CLI driver tests (CPU, tiny synthetic runs)."""
import subprocess
import sys


def run_cli(args, timeout=600):
    return subprocess.run(
        [sys.executable, "-m", "parallel_cnn_amd.train"] + args,
        capture_output=True, text=True, timeout=timeout)


def test_cli_lenet_end_to_end(tmp_path):
    ck = str(tmp_path / "w.bin")
    out = run_cli(["--device", "cpu", "--train-count", "256", "--test-count",
                   "128", "--batch-size", "32", "--epochs", "2",
                   "--log-interval", "0", "--ckpt-save", ck])
    assert out.returncode == 0, out.stdout + out.stderr
    assert "Learning" in out.stdout
    assert "error: " in out.stdout
    assert "Error Rate: " in out.stdout
    # checkpoint reload run
    out2 = run_cli(["--device", "cpu", "--train-count", "64", "--test-count",
                    "64", "--batch-size", "32", "--epochs", "1",
                    "--log-interval", "0", "--ckpt-load", ck])
    assert out2.returncode == 0, out2.stdout + out2.stderr


def test_cli_deepcnn_end_to_end():
    out = run_cli(["--device", "cpu", "--model", "deepcnn", "--train-count",
                   "64", "--test-count", "32", "--batch-size", "16",
                   "--epochs", "1", "--log-interval", "0"])
    assert out.returncode == 0, out.stdout + out.stderr
    assert "Error Rate: " in out.stdout


def test_cli_mnist_idx_files(tmp_path):
    """End-to-end: real IDX files on disk -> --data mnist -> train+test."""
    import struct
    import numpy as np
    rng = np.random.default_rng(0)
    d = tmp_path / "data"
    d.mkdir()
    for name, n, lname in [("train-images.idx3-ubyte", 128,
                            "train-labels.idx1-ubyte"),
                           ("t10k-images.idx3-ubyte", 64,
                            "t10k-labels.idx1-ubyte")]:
        imgs = rng.integers(0, 256, size=(n, 28, 28), dtype=np.uint8)
        lbls = rng.integers(0, 10, size=n, dtype=np.uint8)
        with open(d / name, "wb") as f:
            f.write(struct.pack(">iiii", 2051, n, 28, 28))
            f.write(imgs.tobytes())
        with open(d / lname, "wb") as f:
            f.write(struct.pack(">ii", 2049, n))
            f.write(lbls.tobytes())
    out = run_cli(["--device", "cpu", "--data", "mnist", "--data-dir",
                   str(d), "--batch-size", "32", "--epochs", "1",
                   "--log-interval", "0"])
    assert out.returncode == 0, out.stdout + out.stderr
    assert "Error Rate: " in out.stdout


def test_example_configs_parse():
    """Every YAML under examples/ loads into a valid TrainConfig."""
    import glob
    from parallel_cnn_amd.config import TrainConfig
    paths = sorted(glob.glob("examples/*.yaml"))
    assert len(paths) >= 4
    for p in paths:
        cfg = TrainConfig.from_yaml(p)
        assert cfg.batch_size > 0
        assert cfg.model in ("lenet5", "deepcnn")


def test_cli_runs_with_example_config(tmp_path):
    import subprocess
    import sys
    r = subprocess.run(
        [sys.executable, "-m", "parallel_cnn_amd.train", "--config",
         "examples/lenet_reference.yaml", "--device", "cpu", "--backend",
         "cpu", "--train-count", "64", "--test-count", "32",
         "--log-interval", "0"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    assert "Error Rate" in r.stdout


def test_deepcnn_mnist_mismatch_errors():
    """--model deepcnn --data mnist used to silently feed 784-pixel MNIST
    to a 3072-pixel model; it must fail with a clear shape error."""
    out = run_cli(["--device", "cpu", "--model", "deepcnn",
                   "--data", "mnist", "--epochs", "1"])
    assert out.returncode != 0
    assert "28x28x1" in (out.stdout + out.stderr)


def test_cli_checkpoint_resume(tmp_path):
    """CLI end-to-end exact resume: 2 straight epochs == 1 epoch -> save
    -> --ckpt-load -> remaining epoch (the sidecar's epoch cursor)."""
    ck1 = str(tmp_path / "full.bin")
    ck2 = str(tmp_path / "half.bin")
    ck3 = str(tmp_path / "resumed.bin")
    base = ["--device", "cpu", "--train-count", "64", "--test-count", "32",
            "--batch-size", "16", "--log-interval", "0", "--threshold", "0"]
    out = run_cli(base + ["--epochs", "2", "--ckpt-save", ck1])
    assert out.returncode == 0, out.stdout + out.stderr
    out = run_cli(base + ["--epochs", "1", "--ckpt-save", ck2])
    assert out.returncode == 0, out.stdout + out.stderr
    out = run_cli(base + ["--epochs", "2", "--ckpt-load", ck2,
                          "--ckpt-save", ck3])
    assert out.returncode == 0, out.stdout + out.stderr
    import numpy as np
    full = np.fromfile(ck1, dtype="<f4")
    resumed = np.fromfile(ck3, dtype="<f4")
    assert np.array_equal(full, resumed)
