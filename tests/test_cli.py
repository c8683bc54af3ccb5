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
