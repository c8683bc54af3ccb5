"""Pure-PyTorch fp32 oracle for the DeepCNN family — the numerics reference
for the gfx950 im2col+MFMA path, and the model's CPU execution path
("torchref" backend).

Semantics: sigmoid after every conv/pool/fc; residual loss dz = onehot - y
used directly as the fc preact gradient; batched grads are SUMS over the
batch; update p += dt*scale*g.  No LeNet normalization quirks.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from ..models.deepcnn import DeepCNN, DeepCNNSpec


def conv_weight_nchw(model: DeepCNN, i: int) -> torch.Tensor:
    """[KcP][Cout] flat (kc = (i*K+j)*Cin+ci) -> F.conv2d weight
    [Cout][Cin][K][K]."""
    st = model.stages()[i]
    w = model.view(f"conv{i}_w").view(st.kcp, st.cout)[:st.kc]
    w = w.view(st.k * st.k, st.cin, st.cout)      # [(i,j)][ci][co]
    return w.permute(2, 1, 0).reshape(st.cout, st.cin, st.k, st.k)


def im2col_ref(x_nchw: torch.Tensor, st) -> torch.Tensor:
    """cols [M, KcP] in the kernel's (i*K+j)*Cin+ci order, zero-padded."""
    B = x_nchw.shape[0]
    u = F.unfold(x_nchw, st.k, padding=st.pad)    # [B, Cin*K*K, L] (ci,i,j)
    u = u.view(B, st.cin, st.k * st.k, -1)
    u = u.permute(0, 3, 2, 1).reshape(B * st.h * st.w, st.kc)
    cols = torch.zeros(u.shape[0], st.kcp, dtype=u.dtype)
    cols[:, :st.kc] = u
    return cols


def pool_fwd_ref(a_nchw: torch.Tensor, pw: torch.Tensor, pb: torch.Tensor,
                 K: int) -> torch.Tensor:
    C = a_nchw.shape[1]
    w = pw.view(1, 1, K, K).expand(C, 1, K, K)
    return torch.sigmoid(
        F.conv2d(a_nchw, w, pb.expand(C), stride=K, groups=C))


def forward(x_nhwc: torch.Tensor, model: DeepCNN):
    """x: [B, H, W, Cin] fp32.  Returns (acts, pouts, y): acts/pouts are
    NCHW fp32 per stage; y [B, 10]."""
    B = x_nhwc.shape[0]
    spec = model.spec
    xc = x_nhwc.permute(0, 3, 1, 2).contiguous()
    acts, pouts = [], []
    for i, st in enumerate(model.stages()):
        a = torch.sigmoid(
            F.conv2d(xc, conv_weight_nchw(model, i),
                     model.view(f"conv{i}_b"), padding=st.pad))
        p = pool_fwd_ref(a, model.view(f"pool{i}_w"),
                         model.view(f"pool{i}_b"), st.pool_k)
        acts.append(a)
        pouts.append(p)
        xc = p
    flat = pouts[-1].permute(0, 2, 3, 1).reshape(B, spec.fc_in)  # NHWC flat
    fw = model.view("fc_w").view(spec.n_classes, spec.fc_in)
    y = torch.sigmoid(F.linear(flat, fw, model.view("fc_b")))
    return acts, pouts, y


def backward(x_nhwc: torch.Tensor, model: DeepCNN, acts, pouts,
             y: torch.Tensor, labels: torch.Tensor):
    """Returns (grads_flat [n_params] SUM over batch, loss)."""
    B = x_nhwc.shape[0]
    spec = model.spec
    grads = torch.zeros(spec.n_params, dtype=torch.float32)

    def gview(name):
        off, n = spec.offsets[name]
        return grads[off:off + n]

    onehot = F.one_hot(labels, spec.n_classes).float()
    dz = onehot - y
    loss = dz.norm(dim=1).sum().item()

    flat = pouts[-1].permute(0, 2, 3, 1).reshape(B, spec.fc_in)
    fw = model.view("fc_w").view(spec.n_classes, spec.fc_in)
    gview("fc_w").add_(torch.einsum("bk,bm->km", dz, flat).reshape(-1))
    gview("fc_b").add_(dz.sum(0))
    dflat = (dz @ fw) * flat * (1 - flat)
    last = model.stages()[-1]
    dppre = dflat.view(B, last.oh, last.ow, last.cout).permute(
        0, 3, 1, 2).contiguous()  # NCHW grad wrt pool preact

    for i in range(len(model.stages()) - 1, -1, -1):
        st = model.stages()[i]
        a = acts[i]
        K = st.pool_k
        pw = model.view(f"pool{i}_w").view(K, K)
        # pool wgrad: windows of a
        aw = a.view(B, st.cout, st.oh, K, st.ow, K)
        gview(f"pool{i}_w").add_(
            torch.einsum("bcpq,bcpiqj->ij", dppre, aw).reshape(-1))
        gview(f"pool{i}_b").add_(dppre.sum().reshape(1))
        # pool bwd-data -> conv preact grad
        up = dppre.repeat_interleave(K, dim=2).repeat_interleave(K, dim=3)
        da = up * pw.repeat(st.h // K, st.w // K)
        dapre = da * a * (1 - a)                     # NCHW [B,Cout,H,W]
        # conv wgrad (cols^T @ dapre) in the kernel's kc order
        x_in = (x_nhwc.permute(0, 3, 1, 2).contiguous() if i == 0
                else pouts[i - 1])
        cols = im2col_ref(x_in, st)                  # [M, KcP]
        dapre_f = dapre.permute(0, 2, 3, 1).reshape(-1, st.cout)  # [M, Cout]
        gview(f"conv{i}_w").add_(
            (cols.T @ dapre_f).reshape(-1))
        gview(f"conv{i}_b").add_(dapre_f.sum(0))
        if i > 0:
            # dgrad -> previous pool output grad, with sigmoid'
            w = model.view(f"conv{i}_w").view(st.kcp, st.cout)
            dcols = dapre_f @ w.T                    # [M, KcP]
            dcols_u = dcols[:, :st.kc].view(
                B, st.h * st.w, st.k * st.k, st.cin).permute(0, 3, 2, 1)
            dx = F.fold(dcols_u.reshape(B, st.cin * st.k * st.k, -1),
                        (st.h, st.w), st.k, padding=st.pad)  # NCHW
            pprev = pouts[i - 1]
            dppre = dx * pprev * (1 - pprev)
    return grads, loss
