"""DeepCNN — the framework's larger model family (BASELINE.json config #4):
3 x (conv5x5 same-pad + sigmoid + trainable 2x2 pool + sigmoid) + fc -> 10,
on 32x32x3 inputs, trained with the same residual-loss SGD semantics as the
LeNet path (no LeNet-specific normalization quirks — those are reference
parity only).

Layouts are MFMA-first (see csrc/hip/conv_kernels.hip):
  * activations NHWC;
  * conv weights [KcP][Cout] fp32 with kc = (i*K+j)*Cin + ci and KcP = Kc
    rounded up to 32; pad rows are zero and stay zero;
  * pool = ONE shared KxK kernel + scalar bias per stage (the reference's
    trainable-pool generalized);
  * fc weights [10][FCIN] row-major.

The flat parameter vector (weights then bias per layer, in forward order)
is also the checkpoint format and the single fused DP all-reduce bucket.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Tuple

import numpy as np
import torch


def _pad32(n: int) -> int:
    return (n + 31) // 32 * 32


@dataclass
class ConvStage:
    cin: int
    h: int
    w: int
    cout: int
    k: int = 5
    pad: int = 2
    pool_k: int = 2

    @property
    def kc(self) -> int:
        return self.k * self.k * self.cin

    @property
    def kcp(self) -> int:
        return _pad32(self.kc)

    @property
    def m_rows(self):
        return self.h * self.w  # per image (same-padding conv)

    @property
    def oh(self) -> int:
        return self.h // self.pool_k

    @property
    def ow(self) -> int:
        return self.w // self.pool_k


@dataclass
class DeepCNNSpec:
    in_ch: int = 3
    in_h: int = 32
    in_w: int = 32
    n_classes: int = 10
    channels: Tuple[int, ...] = (32, 64, 64)
    stages: List[ConvStage] = field(default_factory=list)

    def __post_init__(self):
        for c in self.channels:
            if c % 16 != 0:
                raise ValueError(
                    f"DeepCNN channel widths must be multiples of 16 "
                    f"(MFMA fragment width), got {self.channels}")
        if not self.stages:
            cin, h, w = self.in_ch, self.in_h, self.in_w
            for cout in self.channels:
                st = ConvStage(cin=cin, h=h, w=w, cout=cout)
                self.stages.append(st)
                cin, h, w = cout, st.oh, st.ow
        last = self.stages[-1]
        self.fc_in = last.oh * last.ow * last.cout
        # flat parameter offsets
        off = 0
        self.offsets = {}
        for i, st in enumerate(self.stages):
            self.offsets[f"conv{i}_w"] = (off, st.kcp * st.cout)
            off += st.kcp * st.cout
            self.offsets[f"conv{i}_b"] = (off, st.cout)
            off += st.cout
            self.offsets[f"pool{i}_w"] = (off, st.pool_k * st.pool_k)
            off += st.pool_k * st.pool_k
            self.offsets[f"pool{i}_b"] = (off, 1)
            off += 1
        self.offsets["fc_w"] = (off, self.n_classes * self.fc_in)
        off += self.n_classes * self.fc_in
        self.offsets["fc_b"] = (off, self.n_classes)
        off += self.n_classes
        self.n_params = off


class DeepCNN:
    def __init__(self, device: str | torch.device = "cpu", seed: int = 0,
                 spec: DeepCNNSpec | None = None):
        self.spec = spec or DeepCNNSpec()
        self.device = torch.device(device)
        self.params = torch.zeros(self.spec.n_params, dtype=torch.float32,
                                  device=self.device)
        self.grads = torch.zeros_like(self.params)
        self.init_params(seed)

    def view(self, name: str) -> torch.Tensor:
        off, n = self.spec.offsets[name]
        return self.params[off:off + n]

    def grad_view(self, name: str) -> torch.Tensor:
        off, n = self.spec.offsets[name]
        return self.grads[off:off + n]

    def init_params(self, seed: int = 0) -> None:
        """Uniform(-1/sqrt(fan_in), +1/sqrt(fan_in)) for conv/fc weights and
        biases; pool kernels uniform(-0.5, 0.5) like the reference's pool.
        Conv pad rows (kc >= Kc) stay exactly zero."""
        g = torch.Generator().manual_seed(seed)

        def u(n, scale):
            return (torch.rand(n, generator=g) * 2.0 - 1.0) * scale

        with torch.no_grad():
            self.params.zero_()
            for i, st in enumerate(self.stages()):
                s = 1.0 / float(np.sqrt(st.kc))
                w = torch.zeros(st.kcp, st.cout)
                w[:st.kc] = u((st.kc, st.cout), s)
                self.view(f"conv{i}_w").copy_(w.reshape(-1).to(self.device))
                self.view(f"conv{i}_b").copy_(u(st.cout, s).to(self.device))
                self.view(f"pool{i}_w").copy_(
                    u(st.pool_k * st.pool_k, 0.5).to(self.device))
                self.view(f"pool{i}_b").copy_(u(1, 0.5).to(self.device))
            s = 1.0 / float(np.sqrt(self.spec.fc_in))
            self.view("fc_w").copy_(
                u(self.spec.n_classes * self.spec.fc_in, s).to(self.device))
            self.view("fc_b").copy_(u(self.spec.n_classes, s).to(self.device))
            self.grads.zero_()

    def stages(self):
        return self.spec.stages

    def save(self, path: str) -> None:
        self.params.detach().cpu().numpy().astype("<f4").tofile(path)

    def load(self, path: str) -> None:
        arr = np.fromfile(path, dtype="<f4")
        if arr.size != self.spec.n_params:
            raise ValueError(f"checkpoint {path!r} has {arr.size} floats, "
                             f"expected {self.spec.n_params}")
        with torch.no_grad():
            self.params.copy_(torch.from_numpy(arr.copy()).to(self.device))
