"""Inference serving endpoint tests (CPU, in-process TestClient)."""
import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_mnist
from parallel_cnn_amd.serve import create_app


@pytest.fixture(scope="module")
def client(tmp_path_factory):
    # train a tiny model so predictions are meaningful, save, serve it
    from parallel_cnn_amd.engine.trainer import Trainer
    cfg = TrainConfig(device="cpu", backend="cpu", batch_size=32,
                      log_interval=0)
    t = Trainer(cfg)
    x, y = synthetic_mnist(2048, seed=1)
    for _ in range(3):
        t.train_epoch(x, y, log=lambda *a: None)
    ck = str(tmp_path_factory.mktemp("serve") / "w.bin")
    t.model.save(ck)
    app = create_app(TrainConfig(device="cpu", backend="cpu",
                                 log_interval=0), ckpt=ck)
    return TestClient(app)


def test_health_and_info(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"
    r = client.get("/info")
    assert r.json()["input_pixels"] == 784
    assert r.json()["n_params"] == 2343


def test_predict(client):
    x, y = synthetic_mnist(16, seed=2)
    r = client.post("/predict", json={"images": x.tolist(),
                                      "return_probs": True})
    assert r.status_code == 200
    body = r.json()
    assert len(body["labels"]) == 16
    assert all(0 <= v <= 9 for v in body["labels"])
    assert len(body["probs"]) == 16 and len(body["probs"][0]) == 10
    # the trained model should beat chance on the structured bands
    acc = sum(int(a == b) for a, b in zip(body["labels"], y.tolist())) / 16
    assert acc > 0.5, acc


def test_predict_validation(client):
    r = client.post("/predict", json={"images": []})
    assert r.status_code == 400
    r = client.post("/predict", json={"images": [[0.0] * 10]})
    assert r.status_code == 400


def test_deepcnn_serving():
    app = create_app(TrainConfig(device="cpu", backend="torchref",
                                 model="deepcnn", log_interval=0))
    c = TestClient(app)
    assert c.get("/info").json()["input_pixels"] == 32 * 32 * 3
    from parallel_cnn_amd.data.mnist import synthetic_images
    x, _ = synthetic_images(2, 32, 32, 3, seed=1)
    r = c.post("/predict", json={"images": x.tolist()})
    assert r.status_code == 200
    assert len(r.json()["labels"]) == 2


def test_deepcnn_request_larger_than_batch():
    """Regression: a deep /predict with B > cfg.batch_size used to feed the
    raw forward (workspaces sized for batch_size) — OOB GPU writes on the
    hip path.  The chunked classify must return the same labels as
    per-chunk requests."""
    from parallel_cnn_amd.data.mnist import synthetic_images
    cfg = TrainConfig(device="cpu", backend="torchref", model="deepcnn",
                      batch_size=4, log_interval=0)
    app = create_app(cfg)
    c = TestClient(app)
    x, _ = synthetic_images(11, 32, 32, 3, seed=3)  # 11 > 4, ragged tail
    r = c.post("/predict", json={"images": x.tolist()})
    assert r.status_code == 200
    big = r.json()["labels"]
    assert len(big) == 11
    small = []
    for i in range(0, 11, 4):
        rr = c.post("/predict", json={"images": x[i:i + 4].tolist()})
        small += rr.json()["labels"]
    assert big == small


def test_lenet_request_larger_than_batch(client):
    x, _ = synthetic_mnist(70, seed=5)  # 70 > server batch of 64
    r = client.post("/predict", json={"images": x.tolist()})
    assert r.status_code == 200
    assert len(r.json()["labels"]) == 70
