"""HBM streaming-bandwidth calibration on a live MI355X: what do
torch's own copy/elementwise kernels achieve vs our im2col/GEMM-shaped
kernels on the same tensor sizes?  Establishes the realistic ceiling
(theoretical HBM3E is ~8 TB/s; achievable STREAM is typically 60-75%).
Usage: gpurun -- 'PYTHONPATH=. python tools/membench.py'
"""
import time

import torch

from parallel_cnn_amd import _C
from parallel_cnn_amd.ops import native


def t_op(fn, reps=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    dev = "cuda:0"
    for mb in (32, 105, 512):
        n = mb * 1024 * 1024 // 2  # bf16 elements
        a = torch.randn(n, device=dev).to(torch.bfloat16)
        b = torch.empty_like(a)
        dt = t_op(lambda: b.copy_(a))
        gbs = 2 * n * 2 / dt / 1e9  # read + write
        dt2 = t_op(lambda: torch.add(a, 1, out=b))
        gbs2 = 2 * n * 2 / dt2 / 1e9
        print("size %4d MB  copy: %6.0f GB/s   add: %6.0f GB/s" %
              (mb, gbs, gbs2))

    # our im2col, stage-1 shape (writes 105 MB, reads 4.2 MB)
    B = 256
    h = w = 16
    cin, k, pad, kcp = 32, 5, 2, 800
    x = torch.randn(B * h * w * cin, device=dev).to(torch.bfloat16)
    cols = torch.empty(B * h * w * kcp, dtype=torch.bfloat16, device=dev)
    st = native.current_stream_handle()
    dt = t_op(lambda: _C.deep_im2col(x, cols, B, h, w, cin, k, pad, kcp, st))
    traffic = (cols.numel() * 2 + x.numel() * 2 * 25)  # writes + 25x re-read
    print("im2col s1: %.1f us  write-side %4.0f GB/s (incl %4.0f GB/s "
          "gather reads)" % (dt * 1e6, cols.numel() * 2 / dt / 1e9,
                             traffic / dt / 1e9))

    # our forward GEMM, stage-1 shape (reads 105 MB cols, writes 16.8 MB)
    M, K, N = B * h * w, kcp, 64
    wt = torch.randn(N, K, device=dev).to(torch.bfloat16)
    bias = torch.randn(N, device=dev)
    wsrc = torch.randn(K, N, device=dev)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    cols2 = cols.view(M, K)
    dt = t_op(lambda: _C.deep_gemm(cols2, wsrc, bias, out, M, K, N, K, N,
                                   1, 1, st, wt))
    gbs = (M * K * 2 + M * N * 2) / dt / 1e9
    fl = 2 * M * K * N / dt / 1e12
    print("k_gemm s1 fwd: %.1f us  %4.0f GB/s  %5.1f TFLOP/s" %
          (dt * 1e6, gbs, fl))

    # all five conv GEMM shapes at bs=256: ours vs torch.matmul(out=)
    B = 256
    shapes = [  # (label, M, K, N)
        ("s0 fwd ", B * 1024, 96, 32),
        ("s1 fwd ", B * 256, 800, 64),
        ("s2 fwd ", B * 64, 1600, 64),
        ("s1 dgrad", B * 256, 64, 800),
        ("s2 dgrad", B * 64, 64, 1600),
    ]
    for label, M, K, N in shapes:
        A = torch.randn(M, K, device=dev).to(torch.bfloat16)
        Bp = torch.randn(N, K, device=dev).to(torch.bfloat16)  # [N][K] rows
        bias = torch.randn(N, device=dev)
        C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
        wsrc = torch.empty(K, N, device=dev)
        d1 = t_op(lambda: _C.deep_gemm(A, wsrc, bias, C, M, K, N, K, N,
                                       1, 1, st, Bp))
        d2 = t_op(lambda: torch.matmul(A, Bp.t(), out=C))
        print("%s M=%6d K=%4d N=%4d  ours %6.1f us  matmul %6.1f us  "
              "(%.2fx)" % (label, M, K, N, d1 * 1e6, d2 * 1e6, d1 / d2))


if __name__ == "__main__":
    main()
