"""DeepCNN family (config #4): spec, oracle correctness (vs an independent
autograd surrogate), torchref engine, checkpointing."""
import numpy as np
import pytest
import torch
import torch.nn.functional as F

from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_images
from parallel_cnn_amd.engine.deep import DeepTrainer
from parallel_cnn_amd.models.deepcnn import DeepCNN, DeepCNNSpec
from parallel_cnn_amd.ops import deep_ref


def test_spec_layout():
    spec = DeepCNNSpec()
    st = spec.stages
    assert [s.cout for s in st] == [32, 64, 64]
    assert [(s.h, s.w) for s in st] == [(32, 32), (16, 16), (8, 8)]
    assert [s.kcp for s in st] == [96, 800, 1600]
    assert spec.fc_in == 1024
    # offsets tile the flat vector exactly
    spans = sorted(spec.offsets.values())
    pos = 0
    for off, n in spans:
        assert off == pos
        pos += n
    assert pos == spec.n_params == 167097


def test_init_pad_rows_zero():
    m = DeepCNN(seed=3)
    for i, st in enumerate(m.stages()):
        w = m.view(f"conv{i}_w").view(st.kcp, st.cout)
        if st.kcp > st.kc:
            assert w[st.kc:].abs().sum() == 0
        assert w[:st.kc].abs().sum() > 0


def test_checkpoint_roundtrip(tmp_path):
    m = DeepCNN(seed=5)
    p = str(tmp_path / "deep.bin")
    m.save(p)
    assert np.fromfile(p, dtype="<f4").size == m.spec.n_params
    m2 = DeepCNN(seed=0)
    m2.load(p)
    assert torch.equal(m.params, m2.params)


def autograd_surrogate_grads(x_nhwc, model, labels):
    """Independent check: all parameter grads are the exact chain-rule
    grads given the residual dz at the fc preact.  Build the same network
    with autograd leaves and the surrogate loss Ls = -sum((onehot -
    y.detach()) * z_fc): dLs/dz_fc = -(onehot - y), so -autograd grads ==
    deep_ref.backward grads."""
    spec = model.spec
    B = x_nhwc.shape[0]
    leaves = {}
    for name in spec.offsets:
        t = model.view(name).clone().requires_grad_(True)
        leaves[name] = t
    xc = x_nhwc.permute(0, 3, 1, 2)
    for i, st in enumerate(model.stages()):
        w = leaves[f"conv{i}_w"].view(st.kcp, st.cout)[:st.kc]
        w = w.view(st.k * st.k, st.cin, st.cout).permute(2, 1, 0).reshape(
            st.cout, st.cin, st.k, st.k)
        a = torch.sigmoid(
            F.conv2d(xc, w, leaves[f"conv{i}_b"], padding=st.pad))
        pw = leaves[f"pool{i}_w"].view(st.pool_k, st.pool_k)
        pwc = pw.view(1, 1, st.pool_k, st.pool_k).expand(
            st.cout, 1, st.pool_k, st.pool_k)
        xc = torch.sigmoid(
            F.conv2d(a, pwc, leaves[f"pool{i}_b"].expand(st.cout),
                     stride=st.pool_k, groups=st.cout))
    flat = xc.permute(0, 2, 3, 1).reshape(B, spec.fc_in)
    fw = leaves["fc_w"].view(spec.n_classes, spec.fc_in)
    z = F.linear(flat, fw, leaves["fc_b"])
    y = torch.sigmoid(z)
    onehot = F.one_hot(labels, spec.n_classes).float()
    ls = -((onehot - y.detach()) * z).sum()
    ls.backward()
    g = torch.zeros(spec.n_params)
    for name, t in leaves.items():
        off, n = spec.offsets[name]
        g[off:off + n] = -t.grad.reshape(-1)
    return g


def test_backward_matches_autograd_surrogate():
    torch.manual_seed(0)
    m = DeepCNN(seed=2)
    B = 3
    x = torch.rand(B, 32, 32, 3)
    labels = torch.randint(0, 10, (B,))
    acts, pouts, y = deep_ref.forward(x, m)
    grads, _loss = deep_ref.backward(x, m, acts, pouts, y, labels)
    g2 = autograd_surrogate_grads(x, m, labels)
    diff = (grads - g2).abs().max().item()
    ref = g2.abs().max().item()
    assert diff < 1e-3 * max(1.0, ref), (diff, ref)


def test_torchref_trainer_loss_decreases():
    """Per-epoch mean error norm must fall (a 3x sigmoid CNN moves slowly —
    the deep family's benchmark is throughput; LeNet covers accuracy)."""
    xtr, ytr = synthetic_images(512, 32, 32, 3, seed=1)
    cfg = TrainConfig(backend="torchref", device="cpu", batch_size=32,
                      log_interval=0, model="deepcnn")
    t = DeepTrainer(cfg)
    losses = []
    for ep in range(3):
        for s in range(0, 512, 32):
            t.step(*t.stage_batch(xtr[s:s + 32], ytr[s:s + 32]))
        l, n = t.consume_loss()
        losses.append(l / n)
    assert losses[-1] < losses[0] * 0.99, losses


def test_im2col_ref_matches_manual():
    st = DeepCNNSpec().stages[0]
    B = 2
    x = torch.rand(B, st.cin, st.h, st.w)
    cols = deep_ref.im2col_ref(x, st)
    assert cols.shape == (B * st.h * st.w, st.kcp)
    # spot-check a few entries
    for (b, oh, ow, i, j, ci) in [(0, 0, 0, 2, 2, 0), (1, 5, 7, 0, 4, 2),
                                  (1, 31, 31, 4, 4, 1)]:
        m = (b * st.h + oh) * st.w + ow
        kc = (i * st.k + j) * st.cin + ci
        ih, iw = oh + i - st.pad, ow + j - st.pad
        want = (x[b, ci, ih, iw].item()
                if 0 <= ih < st.h and 0 <= iw < st.w else 0.0)
        assert abs(cols[m, kc].item() - want) < 1e-6
    assert cols[:, st.kc:].abs().sum() == 0


def test_custom_channel_widths():
    from parallel_cnn_amd.models.deepcnn import DeepCNNSpec
    spec = DeepCNNSpec(channels=(16, 32, 48))
    assert [s.cout for s in spec.stages] == [16, 32, 48]
    assert spec.fc_in == 4 * 4 * 48
    cfg = TrainConfig(backend="torchref", device="cpu", batch_size=4,
                      log_interval=0, deep_channels="16,32,48")
    t = DeepTrainer(cfg)
    assert t.model.spec.n_params == spec.n_params
    x, y = synthetic_images(4, 32, 32, 3, seed=1)
    t.step(*t.stage_batch(x, y))


def test_channel_width_validation():
    from parallel_cnn_amd.models.deepcnn import DeepCNNSpec
    with pytest.raises(ValueError, match="multiples of 16"):
        DeepCNNSpec(channels=(20, 64, 64))


def test_deep_grad_accumulation_matches_big_batch():
    """ADVICE regression: DeepTrainer used to ignore cfg.grad_accum.
    grad_accum=2 over two bs=4 micro-batches == one bs=8 step with mean
    reduction (deep wgrad accumulates; the update scale folds in the
    accumulation count)."""
    x, y = synthetic_images(8, 32, 32, 3, seed=11)
    cfg_a = TrainConfig(backend="torchref", device="cpu", batch_size=4,
                        grad_accum=2, log_interval=0)
    cfg_b = TrainConfig(backend="torchref", device="cpu", batch_size=8,
                        log_interval=0)
    ta, tb = DeepTrainer(cfg_a), DeepTrainer(cfg_b)
    ta.step(*ta.stage_batch(x[:4], y[:4]))
    ta.step(*ta.stage_batch(x[4:], y[4:]))
    tb.step(*tb.stage_batch(x, y))
    assert torch.allclose(ta.model.params, tb.model.params, atol=1e-6)
    assert ta.model.grads.abs().sum() == 0  # consumed at the boundary


def test_four_stage_spec():
    """Deeper-than-default family: 4 stages halve 32 -> 2; the spec, the
    oracle and the torchref engine all compose."""
    from parallel_cnn_amd.models.deepcnn import DeepCNNSpec
    spec = DeepCNNSpec(channels=(16, 32, 32, 64))
    assert [(s.h, s.w) for s in spec.stages] == [(32, 32), (16, 16),
                                                (8, 8), (4, 4)]
    assert spec.fc_in == 2 * 2 * 64
    cfg = TrainConfig(backend="torchref", device="cpu", batch_size=4,
                      log_interval=0, deep_channels="16,32,32,64")
    t = DeepTrainer(cfg)
    x, y = synthetic_images(8, 32, 32, 3, seed=2)
    t.step(*t.stage_batch(x[:4], y[:4]))
    l, n = t.consume_loss()
    assert n == 4 and l > 0
