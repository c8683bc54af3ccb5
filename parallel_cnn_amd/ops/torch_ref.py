"""Pure-PyTorch fp32 reference ops — the numerics oracle for every backend.

Implements the exact semantics catalogued in SURVEY.md §0.1 (reference
behavior defined by Sequential/layer.h):

* sigmoid after conv1, pool and fc;
* the "pool" is a trainable weighted-sum downsample: ONE shared 4x4 kernel,
  stride 4, plus one scalar bias (Sequential/layer.h:154-180);
* loss gradient dz = onehot(label) - y used directly as the fc preact
  gradient (no output sigmoid derivative, Sequential/layer.h:91-95); the loss
  metric is sum_b ||dz_b||_2 (Sequential/Main.cpp:167-168);
* conv1 weight/bias grads normalized by 1/(24*24)
  (Sequential/layer.h:381-414); pool bias grad averaged over 216; pool/fc
  weight grads unnormalized;
* batched grads are SUMS over the batch; update is p += dt * scale * g.

Every tensor op here is a different implementation strategy from the native
C++ ops (unfold/einsum vs scalar loops) so the two act as independent
cross-checks in the tests.
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F

from .shapes import (C1_CH, C1_H, C1_K, C1_PIX, C1_W, FC_IN, FC_OUT, IN_H,
                     IN_W, N_PARAMS, OFF_C1B, OFF_C1W, OFF_FB, OFF_FW, OFF_S1B,
                     OFF_S1W, S1_H, S1_K, S1_W)


def split_params(params: torch.Tensor):
    """Views of the flat parameter vector."""
    assert params.numel() == N_PARAMS
    c1w = params[OFF_C1W:OFF_C1B].view(C1_CH, 1, C1_K, C1_K)
    c1b = params[OFF_C1B:OFF_S1W]
    s1w = params[OFF_S1W:OFF_S1B].view(S1_K, S1_K)
    s1b = params[OFF_S1B:OFF_FW]
    fw = params[OFF_FW:OFF_FB].view(FC_OUT, FC_IN)
    fb = params[OFF_FB:]
    return c1w, c1b, s1w, s1b, fw, fb


def forward(x: torch.Tensor, params: torch.Tensor, pool: str = "trainable",
            loss: str = "residual"
            ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """x: [B, 784] fp32 -> (a1 [B,6,24,24], a2 [B,6,6,6], y [B,10]).

    pool: "trainable" (reference: shared 4x4 kernel + scalar bias) or
    "max" (max over the 4x4 window; the s1 parameters are unused).  Both
    apply sigmoid to the pool output.  loss: "residual" -> y = sigmoid(z);
    "softmax_ce" -> y = softmax(z)."""
    B = x.shape[0]
    c1w, c1b, s1w, s1b, fw, fb = split_params(params)
    xi = x.view(B, 1, IN_H, IN_W)
    a1 = torch.sigmoid(F.conv2d(xi, c1w, c1b))
    if pool == "max":
        a2 = torch.sigmoid(F.max_pool2d(a1, S1_K))
    else:
        pw = s1w.expand(C1_CH, 1, S1_K, S1_K)
        a2 = torch.sigmoid(
            F.conv2d(a1, pw, s1b.expand(C1_CH), stride=S1_K, groups=C1_CH))
    z = F.linear(a2.reshape(B, FC_IN), fw, fb)
    y = torch.softmax(z, dim=1) if loss == "softmax_ce" else torch.sigmoid(z)
    return a1, a2, y


def backward(x: torch.Tensor, params: torch.Tensor, a1: torch.Tensor,
             a2: torch.Tensor, y: torch.Tensor, labels: torch.Tensor,
             pool: str = "trainable", loss: str = "residual"):
    """Returns (dz, dz2, dz1, grads, loss_value).

    grads is the flat [N_PARAMS] SUM-over-batch gradient.  With
    loss="residual": dz = onehot - y, loss_value = sum_b ||dz_b||_2 (the
    reference's metric).  With loss="softmax_ce": dz = onehot - softmax
    (the ascent-convention CE gradient — same algebraic form), loss_value
    = sum_b -log y_b[label].
    """
    B = x.shape[0]
    _c1w, _c1b, s1w, _s1b, fw, _fb = split_params(params)
    onehot = F.one_hot(labels, FC_OUT).to(y.dtype)
    dz = onehot - y                                   # [B,10]
    if loss == "softmax_ce":
        loss_v = -torch.log(
            y.gather(1, labels.view(-1, 1)).clamp_min(1e-30)).sum().item()
    else:
        loss_v = dz.norm(dim=1).sum().item()

    grads = torch.zeros(N_PARAMS, dtype=params.dtype, device=params.device)
    g_c1w, g_c1b, g_s1w, g_s1b, g_fw, g_fb = split_params(grads)

    a2f = a2.reshape(B, FC_IN)
    g_fw += torch.einsum("bk,bm->km", dz, a2f).view(FC_OUT, FC_IN)
    g_fb += dz.sum(0)

    da2 = dz @ fw                                     # [B,216]
    dz2 = (da2 * a2f * (1 - a2f)).view(B, C1_CH, S1_H, S1_W)

    if pool == "max":
        # route the pool gradient to the argmax of each 4x4 window
        a1w = a1.view(B, C1_CH, S1_H, S1_K, S1_W, S1_K)
        flat = a1w.permute(0, 1, 2, 4, 3, 5).reshape(B, C1_CH, S1_H, S1_W,
                                                     S1_K * S1_K)
        amax = flat.argmax(dim=-1, keepdim=True)
        da1w = torch.zeros_like(flat)
        da1w.scatter_(-1, amax, dz2.unsqueeze(-1))
        da1 = da1w.view(B, C1_CH, S1_H, S1_W, S1_K, S1_K).permute(
            0, 1, 2, 4, 3, 5).reshape(B, C1_CH, C1_H, C1_W)
    else:
        g_s1b += dz2.sum() / FC_IN
        # pool wgrad: windows of a1
        a1w = a1.view(B, C1_CH, S1_H, S1_K, S1_W, S1_K)  # [b,o,pr,i,pc,j]
        g_s1w += torch.einsum("bopq,bopiqj->ij", dz2, a1w)
        # pool backward-data (stride == kernel: pure upsample * kernel)
        up = dz2.repeat_interleave(S1_K, dim=2).repeat_interleave(S1_K, dim=3)
        ktile = s1w.repeat(S1_H, S1_W)                # [24,24]
        da1 = up * ktile
    dz1 = da1 * a1 * (1 - a1)                         # [B,6,24,24]

    # conv1 wgrad via unfold, normalized by 1/(24*24)
    patches = F.unfold(x.view(B, 1, IN_H, IN_W), C1_K)  # [B,25,576]
    dz1f = dz1.reshape(B, C1_CH, C1_PIX)
    g_c1w += (torch.einsum("bot,bwt->ow", dz1f, patches) / C1_PIX).view(
        C1_CH, 1, C1_K, C1_K)
    g_c1b += dz1f.sum(dim=(0, 2)) / C1_PIX

    return dz, dz2.reshape(B, -1), dz1.reshape(B, C1_CH * C1_PIX), grads, loss_v


def update(params: torch.Tensor, grads: torch.Tensor, dt: float,
           scale: float) -> None:
    params += dt * scale * grads
    grads.zero_()
