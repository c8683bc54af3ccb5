"""Stretch probe: measure the im2col + MFMA GEMM
machinery on the LeNet conv1 scale, against the direct fused kernels
that the LeNet path actually uses.

The north star names both "im2col + GEMM and direct 5x5" for the conv
hot path; the framework ships the direct kernels for LeNet (3 fused
launches/step) and the GEMM machinery for the DeepCNN family.  This
probe runs the deep GEMM machinery on the LeNet conv1 SHAPE so the
choice is backed by a number, not an assumption.

Geometry note: the deep kernels implement same-padding (output grid ==
input grid), so the probe uses 28x28 same-pad with Cin=1 (M = B*784,
KcP = 32 >= 25) and Cout padded 6 -> 16 (MFMA fragment width).  That is
MORE work than LeNet's valid conv (24x24, 6 ch), but the comparison is
against the ENTIRE fused LeNet step (fwd+bwd-data for conv1+pool+fc in
one kernel), so a GEMM path losing here loses with margin.

Run on a GPU box:  python tools/lenet_mfma_probe.py [--batch 64]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def t_kernel(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=64)
    args = p.parse_args()
    assert torch.cuda.is_available(), "probe needs a GPU"
    from parallel_cnn_amd import _C
    from parallel_cnn_amd.ops import native

    B = args.batch
    dev = torch.device("cuda")
    st = native.current_stream_handle()
    H = W = 28
    K, PAD, KCP, N = 5, 2, 32, 16  # Kc=25 padded to 32; Cout 6 padded to 16
    M = B * H * W

    x = torch.rand(B, H * W, dtype=torch.bfloat16, device=dev)
    cols = torch.empty(M, KCP, dtype=torch.bfloat16, device=dev)
    wsrc = torch.randn(KCP, N, dtype=torch.float32, device=dev)
    bias = torch.randn(N, dtype=torch.float32, device=dev)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    c32 = torch.empty(16 * M * N, dtype=torch.float32, device=dev)

    t_im2col = t_kernel(lambda: _C.deep_im2col(x, cols, B, H, W, 1, K, PAD,
                                               KCP, st))
    t_gemm = t_kernel(lambda: _C.deep_gemm(cols, wsrc, bias, out, M, KCP, N,
                                           KCP, N, 1, 1, st, c32=c32))

    # the direct path's number: one FULL fused LeNet training step
    # (fwd+bwd-data kernel + wgrad kernel + update kernel)
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.data.mnist import synthetic_mnist
    cfg = TrainConfig(batch_size=B, device="cuda", backend="hip",
                      log_interval=0)
    tr = Trainer(cfg)
    xh, yh = synthetic_mnist(B, seed=0)
    xb, lb = tr.stage_batch(xh, yh)
    t_step = t_kernel(lambda: tr.step(xb, lb))

    print(f"LeNet conv1-shape probe, B={B} (us per call):")
    print(f"  im2col (M={M}, KcP={KCP}):            {t_im2col:7.2f}")
    print(f"  MFMA GEMM (M={M}, K={KCP}, N={N}):    {t_gemm:7.2f}")
    print(f"  im2col+GEMM conv1 fwd only:           {t_im2col + t_gemm:7.2f}")
    print(f"  direct path, ENTIRE fused train step: {t_step:7.2f}")
    print(json.dumps({
        "probe": "lenet_mfma", "batch": B, "im2col_us": t_im2col,
        "gemm_us": t_gemm, "gemm_path_conv1_fwd_us": t_im2col + t_gemm,
        "direct_full_step_us": t_step}))
    return 0


import json  # noqa: E402

if __name__ == "__main__":
    sys.exit(main())
