"""Per-kernel / per-role launch timing on one GPU (run via gpurun).

Times each kernel in isolation with CUDA events around an N-launch loop
(within-probe A/B, avoids cross-process noise)."""
import sys

import torch

sys.path.insert(0, ".")
from parallel_cnn_amd import _C  # noqa: E402
from parallel_cnn_amd.ops import native, shapes as S  # noqa: E402

DEV = "cuda:0"
REPS = 500


def time_loop(fn, reps=REPS):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    t1.synchronize()
    return t0.elapsed_time(t1) * 1e3 / reps  # us per launch


def main():
    torch.manual_seed(0)
    for B in (64, 512, 4096):
        ad = torch.bfloat16
        x = torch.rand(B, S.IN_PIX, device=DEV).to(ad)
        labels = torch.randint(0, 10, (B,), device=DEV, dtype=torch.int32)
        params = (0.5 - torch.rand(S.N_PARAMS, device=DEV))
        a1 = torch.empty(B, S.C1_OUT, dtype=ad, device=DEV)
        a2 = torch.empty(B, S.S1_OUT, dtype=ad, device=DEV)
        y = torch.empty(B, S.FC_OUT, device=DEV)
        dz = torch.empty(B, S.FC_OUT, device=DEV)
        dz2 = torch.empty(B, S.S1_OUT, device=DEV)
        dz1 = torch.empty(B, S.C1_OUT, dtype=ad, device=DEV)
        loss = torch.zeros(1, device=DEV)
        corr = torch.zeros(1, dtype=torch.int32, device=DEV)
        grads = torch.zeros(S.N_PARAMS, device=DEV)
        st = native.current_stream_handle()

        fwd = lambda: _C.hip_fwdbwd(x, params, a1, a2, y, dz, dz2, dz1,
                                    labels, loss, corr, B, 0, st)
        fwd_fused = lambda: _C.hip_fwdbwd(x, params, a1, a2, y, dz, dz2,
                                          dz1, labels, loss, corr, B, 0, st,
                                          0, 0, grads, 1)
        fwd()
        print(f"== B={B}")
        print(f"  fwdbwd          : {time_loop(fwd):8.2f} us")
        print(f"  fwdbwd+cs-wgrad : {time_loop(fwd_fused):8.2f} us")
        t = time_loop(lambda: _C.hip_wgrad_roles(
            x, a1, a2, dz, dz2, dz1, grads, B, 0, 4, st))
        print(f"  wgrad fc-only   : {t:8.2f} us")
        for roles, name in [(7, "wgrad all"), (1, "wgrad c1"),
                            (2, "wgrad s1"), (4, "wgrad fc")]:
            t = time_loop(lambda: _C.hip_wgrad_roles(
                x, a1, a2, dz, dz2, dz1, grads, B, 0, roles, st))
            print(f"  {name:16s}: {t:8.2f} us")
        for gc in (8, 12, 16, 20, 24, 32):
            t = time_loop(lambda: _C.hip_wgrad_roles(
                x, a1, a2, dz, dz2, dz1, grads, B, gc, 1, st))
            print(f"  wgrad c1 GC={gc:3d}: {t:8.2f} us")
        t = time_loop(lambda: _C.hip_update(params, grads, 0.0, st))
        print(f"  update          : {t:8.2f} us")
        # eval / infer modes
        t = time_loop(lambda: _C.hip_fwdbwd(x, params, a1, a2, y, dz, dz2,
                                            dz1, labels, loss, corr, B, 1,
                                            st))
        print(f"  fwd eval        : {t:8.2f} us")


if __name__ == "__main__":
    main()
