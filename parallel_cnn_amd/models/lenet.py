"""LeNet-5-class model state: flat fp32 parameters + flat gradient bucket.

The flat gradient vector doubles as the single fused RCCL all-reduce payload
(one all-reduce per step, not 16 per-op reduces like the
reference MPI variant).

Checkpoint format (designed — the reference has none):
flat little-endian float32, per layer weight then bias, in Layer-ctor order:
c1.weight[150], c1.bias[6], s1.weight[16], s1.bias[1], f.weight[2160],
f.bias[10].  2343 floats, 9372 bytes.
"""
from __future__ import annotations

import numpy as np
import torch

from ..ops import shapes as S


class LeNet5:
    def __init__(self, device: str | torch.device = "cpu", seed: int = 0):
        self.device = torch.device(device)
        self.params = torch.empty(S.N_PARAMS, dtype=torch.float32,
                                  device=self.device)
        self.grads = torch.zeros(S.N_PARAMS, dtype=torch.float32,
                                 device=self.device)
        self.init_reference(seed)

    def init_reference(self, seed: int = 0) -> None:
        """Reference init: every weight and bias uniform in (-0.5, 0.5]
        (`0.5f - rand()/RAND_MAX`, Sequential/layer.h:48-54)."""
        g = torch.Generator(device="cpu").manual_seed(seed)
        vals = 0.5 - torch.rand(S.N_PARAMS, generator=g, dtype=torch.float32)
        with torch.no_grad():
            self.params.copy_(vals.to(self.device))
            self.grads.zero_()

    # --- named views -------------------------------------------------------
    @property
    def c1_weight(self):
        return self.params[S.OFF_C1W:S.OFF_C1B].view(S.C1_CH, S.C1_K, S.C1_K)

    @property
    def c1_bias(self):
        return self.params[S.OFF_C1B:S.OFF_S1W]

    @property
    def s1_weight(self):
        return self.params[S.OFF_S1W:S.OFF_S1B].view(S.S1_K, S.S1_K)

    @property
    def s1_bias(self):
        return self.params[S.OFF_S1B:S.OFF_FW]

    @property
    def f_weight(self):
        return self.params[S.OFF_FW:S.OFF_FB].view(S.FC_OUT, S.FC_IN)

    @property
    def f_bias(self):
        return self.params[S.OFF_FB:]

    # --- checkpoint --------------------------------------------------------
    def save(self, path: str) -> None:
        arr = self.params.detach().cpu().numpy().astype("<f4")
        arr.tofile(path)

    def load(self, path: str) -> None:
        arr = np.fromfile(path, dtype="<f4")
        if arr.size != S.N_PARAMS:
            raise ValueError(
                f"checkpoint {path!r} has {arr.size} floats, "
                f"expected {S.N_PARAMS}")
        with torch.no_grad():
            self.params.copy_(torch.from_numpy(arr.copy()).to(self.device))
