"""Sweep engine-side grid knobs for the DeepCNN kernels on a live GPU:
k_wgrad_gemm MS (K-slices, atomics combine) and k_colsum G (workgroups).
The GEMM-shaped kernels are latency-bound at small grids (256 CUs want
>=2048 workgroups); this measures where the atomic-combine cost starts
to win/lose.  Usage: gpurun -- 'python tools/deep_sweep.py'
"""
import time

import torch

from parallel_cnn_amd import _C
from parallel_cnn_amd.ops import native

REPS = 200


def t_kernel(fn, reps=REPS):
    st = native.current_stream_handle()
    for _ in range(10):
        fn(st)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn(st)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    B = 256
    # (H, W, Cin, Cout, KcP) per stage of the default DeepCNN
    stages = [(32, 32, 3, 32, 96), (16, 16, 32, 64, 800), (8, 8, 64, 64, 1600)]
    print("== k_wgrad_gemm MS sweep (bs=%d) ==" % B)
    for si, (h, w, cin, cout, kcp) in enumerate(stages):
        M = B * h * w
        cols = torch.randn(M, kcp, device=dev).to(torch.bfloat16)
        dpre = torch.randn(M, cout, device=dev).to(torch.bfloat16)
        dW = torch.zeros(kcp, cout, device=dev)
        ktiles = (kcp + 63) // 64
        empty = torch.empty(0, device=dev)
        base = max(1, min(128, 512 // ktiles))
        cands = sorted({base, 64, 128, 256,
                        max(1, 1024 // ktiles), max(1, 2048 // ktiles),
                        max(1, 4096 // ktiles)})
        for slab in (False, True):
            part = (torch.empty(max(cands) * kcp * cout, device=dev)
                    if slab else torch.empty(0, device=dev))
            row = []
            for ms in cands:
                if M // ms < 64:  # slice thinner than one BK chunk: skip
                    continue
                us = t_kernel(lambda st, ms=ms: _C.deep_wgrad_gemm(
                    cols, dpre, dW, M, kcp, cout, ms, st, empty, h, w, cin,
                    5, 2, part))
                row.append((ms, us))
            best = min(row, key=lambda p: p[1])
            print("stage%d %s M=%6d KcP=%4d ktiles=%2d best(ms=%d)=%.1fus" %
                  (si, "slab" if slab else "atom", M, kcp, ktiles,
                   best[0], best[1]))
            print("   ", " ".join("ms=%d:%.1f" % p for p in row))

    print("== k_colsum G sweep ==")
    for si, (h, w, cin, cout, kcp) in enumerate(stages):
        M = B * h * w
        dpre = torch.randn(M, cout, device=dev).to(torch.bfloat16)
        db = torch.zeros(cout, device=dev)
        part = torch.empty(512 * cout, device=dev)
        base = max(32, min(512, (M * cout) // (256 * 96)))
        row = []
        for g in sorted({base, 64, 128, 256, 512}):
            us = t_kernel(lambda st, g=g: _C.deep_colsum(dpre, part, db, M,
                                                         cout, g, st))
            row.append((g, us))
        best = min(row, key=lambda p: p[1])
        print("stage%d M=%6d N=%2d cur(G=%d)=%.1fus best(G=%d)=%.1fus" %
              (si, M, cout, base, dict(row)[base], best[0], best[1]))
        print("   ", " ".join("G=%d:%.1f" % p for p in row))


def pool_sweep():
    dev = "cuda:0"
    B = 256
    print("== k_pool_wgrad8 G sweep (bs=%d) ==" % B)
    for si, (h, w, cout) in enumerate([(32, 32, 32), (16, 16, 64),
                                       (8, 8, 64)]):
        dppre = torch.randn(B * (h // 2) * (w // 2) * cout,
                            device=dev).to(torch.bfloat16)
        acts = torch.rand(B * h * w * cout, device=dev).to(torch.bfloat16)
        dpw = torch.zeros(5, device=dev)
        base = max(8, min(512, (B * (h // 2) * (w // 2) * cout) // (256 * 16)))
        row = []
        for g in sorted({base, 32, 64, 128, 256, 512}):
            us = t_kernel(lambda st, g=g: _C.deep_pool_wgrad(
                dppre, acts, dpw, B, h, w, cout, 2, g, st))
            row.append((g, us))
        best = min(row, key=lambda p: p[1])
        print("stage%d cur(G=%d)=%.1fus best(G=%d)=%.1fus" %
              (si, base, dict(row)[base], best[0], best[1]))
        print("   ", " ".join("G=%d:%.1f" % p for p in row))


if __name__ == "__main__":
    main()
    pool_sweep()
