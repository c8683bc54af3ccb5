"""Max-pool and softmax-cross-entropy modes: native CPU ops vs the torch
oracle, plus an autograd check of the softmax-CE gradient."""
import pytest
import torch
import torch.nn.functional as F

from parallel_cnn_amd import _C
from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_mnist
from parallel_cnn_amd.engine.trainer import Trainer
from parallel_cnn_amd.ops import shapes as S
from parallel_cnn_amd.ops import torch_ref

MODES = [("trainable", "residual"), ("max", "residual"),
         ("trainable", "softmax_ce"), ("max", "softmax_ce")]


def make_case(B, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(B, S.IN_PIX, generator=g)
    labels = torch.randint(0, 10, (B,), generator=g)
    params = (0.5 - torch.rand(S.N_PARAMS, generator=g)).float()
    return x, labels, params


@pytest.mark.parametrize("pool,loss", MODES)
def test_native_matches_torch_ref_modes(pool, loss):
    B = 5
    x, labels, params = make_case(B, seed=hash((pool, loss)) % 1000)
    pm = 1 if pool == "max" else 0
    lm = 1 if loss == "softmax_ce" else 0
    a1r, a2r, yr = torch_ref.forward(x, params, pool, loss)
    dzr, dz2r, dz1r, gr, lossr = torch_ref.backward(x, params, a1r, a2r, yr,
                                                    labels, pool, loss)
    a1 = torch.empty(B, S.C1_OUT)
    a2 = torch.empty(B, S.S1_OUT)
    y = torch.empty(B, S.FC_OUT)
    _C.cpu_forward(x, params, a1, a2, y, pm, lm)
    dz = torch.empty(B, S.FC_OUT)
    dz2 = torch.empty(B, S.S1_OUT)
    dz1 = torch.empty(B, S.C1_OUT)
    g = torch.zeros(S.N_PARAMS)
    lossc = _C.cpu_backward(x, params, a1, a2, y, labels, dz, dz2, dz1, g,
                            pm, lm)
    for name, a, b in [("a1", a1r.reshape(B, -1), a1),
                       ("a2", a2r.reshape(B, -1), a2), ("y", yr, y),
                       ("dz", dzr, dz), ("dz1", dz1r, dz1), ("grads", gr, g)]:
        d = (a.reshape(-1) - b.reshape(-1)).abs().max().item()
        assert d < 1e-4, f"{pool}/{loss} {name}: {d}"
    assert abs(lossc - lossr) < 1e-3 * max(1.0, abs(lossr))
    if pool == "max":
        # max pool has no parameters -> its grads must be exactly zero
        assert g[S.OFF_S1W:S.OFF_FW].abs().sum() == 0


def test_softmax_ce_gradient_is_true_gradient():
    """With softmax-CE the update IS plain gradient descent on CE (up to
    the ascent sign convention and the conv 1/576 quirk): check the fc
    grads against autograd exactly."""
    B = 4
    x, labels, params = make_case(B, seed=42)
    p = params.clone().requires_grad_(True)
    c1w = p[S.OFF_C1W:S.OFF_C1B].view(6, 1, 5, 5)
    c1b = p[S.OFF_C1B:S.OFF_S1W]
    s1w = p[S.OFF_S1W:S.OFF_S1B].view(4, 4)
    s1b = p[S.OFF_S1B:S.OFF_FW]
    fw = p[S.OFF_FW:S.OFF_FB].view(10, 216)
    fb = p[S.OFF_FB:]
    a1 = torch.sigmoid(F.conv2d(x.view(B, 1, 28, 28), c1w, c1b))
    a2 = torch.sigmoid(F.conv2d(a1, s1w.expand(6, 1, 4, 4), s1b.expand(6),
                                stride=4, groups=6))
    z = F.linear(a2.reshape(B, 216), fw, fb)
    ce = F.cross_entropy(z, labels, reduction="sum")
    ce.backward()
    ag = -p.grad  # ascent convention
    a1r, a2r, yr = torch_ref.forward(x, params, "trainable", "softmax_ce")
    _, _, _, gr, lossv = torch_ref.backward(x, params, a1r, a2r, yr, labels,
                                            "trainable", "softmax_ce")
    assert abs(lossv - ce.item()) < 1e-3
    fc_slice = slice(S.OFF_FW, S.N_PARAMS)
    d = (gr[fc_slice] - ag[fc_slice]).abs().max().item()
    assert d < 1e-4, d


def test_softmax_trainer_learns():
    """softmax-CE learns the structured bands (sigmoid hidden layers are
    slow, so a few epochs and a larger dt)."""
    xtr, ytr = synthetic_mnist(2048, seed=1)
    xte, yte = synthetic_mnist(512, seed=2)
    cfg = TrainConfig(backend="cpu", device="cpu", batch_size=32,
                      log_interval=0, pool="trainable", loss="softmax_ce",
                      dt=0.5)
    t = Trainer(cfg)
    before = t.evaluate(xte, yte)
    for _ in range(4):
        t.train_epoch(xtr, ytr, log=lambda *a: None)
    err = t.evaluate(xte, yte)
    assert err < before and err < 50.0, (before, err)


def test_maxpool_trainer_loss_decreases():
    xtr, ytr = synthetic_mnist(1024, seed=3)
    cfg = TrainConfig(backend="cpu", device="cpu", batch_size=32,
                      log_interval=0, pool="max", loss="softmax_ce", dt=0.2)
    t = Trainer(cfg)
    losses = []
    for _ in range(3):
        for s in range(0, 1024, 32):
            t.step(*t.stage_batch(xtr[s:s + 32], ytr[s:s + 32]))
        l, n = t.consume_loss()
        losses.append(l / n)
    assert losses[-1] < losses[0], losses
