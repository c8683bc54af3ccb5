"""Native CPU ops vs the pure-PyTorch fp32 oracle (two independent
implementations of the reference semantics)."""
import numpy as np
import pytest
import torch

from parallel_cnn_amd import _C
from parallel_cnn_amd.ops import shapes as S
from parallel_cnn_amd.ops import torch_ref


def make_case(B, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(B, S.IN_PIX, generator=g)
    labels = torch.randint(0, 10, (B,), generator=g)
    params = (0.5 - torch.rand(S.N_PARAMS, generator=g)).float()
    return x, labels, params


def run_native(x, labels, params):
    B = x.shape[0]
    a1 = torch.empty(B, S.C1_OUT)
    a2 = torch.empty(B, S.S1_OUT)
    y = torch.empty(B, S.FC_OUT)
    _C.cpu_forward(x, params, a1, a2, y)
    dz = torch.empty(B, S.FC_OUT)
    dz2 = torch.empty(B, S.S1_OUT)
    dz1 = torch.empty(B, S.C1_OUT)
    g = torch.zeros(S.N_PARAMS)
    loss = _C.cpu_backward(x, params, a1, a2, y, labels, dz, dz2, dz1, g)
    return a1, a2, y, dz, dz2, dz1, g, loss


@pytest.mark.parametrize("B", [1, 3, 64])
def test_native_matches_torch_ref(B):
    x, labels, params = make_case(B, seed=B)
    a1r, a2r, yr = torch_ref.forward(x, params)
    dzr, dz2r, dz1r, gr, lossr = torch_ref.backward(x, params, a1r, a2r, yr,
                                                    labels)
    a1, a2, y, dz, dz2, dz1, g, loss = run_native(x, labels, params)
    for name, a, b in [("a1", a1r.reshape(B, -1), a1),
                       ("a2", a2r.reshape(B, -1), a2), ("y", yr, y),
                       ("dz", dzr, dz), ("dz2", dz2r, dz2),
                       ("dz1", dz1r, dz1), ("grads", gr, g)]:
        assert torch.allclose(a.reshape(-1), b.reshape(-1), atol=1e-4), name
    assert abs(loss - lossr) < 1e-3


def test_forward_against_naive_numpy():
    """Third, fully independent scalar-loop check of one forward pass."""
    x, _labels, params = make_case(1, seed=42)
    xn = x[0].numpy().reshape(28, 28)
    p = params.numpy()
    sig = lambda v: 1.0 / (1.0 + np.exp(-v))
    a1 = np.zeros((6, 24, 24), dtype=np.float64)
    for o in range(6):
        w = p[S.OFF_C1W + o * 25:S.OFF_C1W + (o + 1) * 25].reshape(5, 5)
        for r in range(24):
            for c in range(24):
                a1[o, r, c] = sig((w * xn[r:r + 5, c:c + 5]).sum()
                                  + p[S.OFF_C1B + o])
    s1w = p[S.OFF_S1W:S.OFF_S1W + 16].reshape(4, 4)
    a2 = np.zeros((6, 6, 6))
    for o in range(6):
        for pr in range(6):
            for pc in range(6):
                a2[o, pr, pc] = sig(
                    (s1w * a1[o, 4 * pr:4 * pr + 4, 4 * pc:4 * pc + 4]).sum()
                    + p[S.OFF_S1B])
    fw = p[S.OFF_FW:S.OFF_FB].reshape(10, 216)
    y = sig(fw @ a2.reshape(216) + p[S.OFF_FB:])

    a1n, a2n, yn = run_native(x, torch.zeros(1, dtype=torch.int64), params)[:3]
    assert np.allclose(a1.reshape(-1), a1n[0].numpy(), atol=1e-5)
    assert np.allclose(a2.reshape(-1), a2n[0].numpy(), atol=1e-5)
    assert np.allclose(y, yn[0].numpy(), atol=1e-5)


def test_batched_grads_are_sum_of_per_sample():
    """Batched backward == sum of bs=1 backwards at the same weights."""
    B = 5
    x, labels, params = make_case(B, seed=7)
    _, _, _, _, _, _, g_batch, loss_batch = run_native(x, labels, params)
    g_sum = torch.zeros(S.N_PARAMS)
    loss_sum = 0.0
    for b in range(B):
        *_rest, g1, l1 = run_native(x[b:b + 1], labels[b:b + 1], params)
        g_sum += g1
        loss_sum += l1
    assert torch.allclose(g_batch, g_sum, atol=1e-4)
    assert abs(loss_batch - loss_sum) < 1e-4


def test_update_rule():
    """p += dt*scale*g, grads zeroed afterwards (gradient ascent on the
    residual, Sequential/layer.h:97-101)."""
    params = torch.zeros(S.N_PARAMS)
    grads = torch.ones(S.N_PARAMS)
    _C.cpu_update(params, grads, 0.1, 0.5)
    assert torch.allclose(params, torch.full((S.N_PARAMS,), 0.05))
    assert grads.abs().sum() == 0


def test_conv_grad_normalization_quirk():
    """conv1 grads carry the reference's 1/(24*24) factor; fc grads don't
    (SURVEY.md §0.1 item 5).  Scale dz1 by 576 -> conv wgrad scales by 576."""
    B = 2
    x, labels, params = make_case(B, seed=3)
    a1r, a2r, yr = torch_ref.forward(x, params)
    dzr, dz2r, dz1r, gr, _ = torch_ref.backward(x, params, a1r, a2r, yr,
                                                labels)
    # conv bias grad == mean over batch-summed dz1 / 576
    expected = dz1r.reshape(B, 6, 576).sum(dim=(0, 2)) / 576.0
    assert torch.allclose(gr[S.OFF_C1B:S.OFF_C1B + 6], expected, atol=1e-5)
    # fc bias grad == plain sum of dz
    assert torch.allclose(gr[S.OFF_FB:], dzr.sum(0), atol=1e-5)
