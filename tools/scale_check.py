"""Multi-GPU preflight — run BEFORE the first 8-GPU job to catch env,
binding and collective problems while they are cheap to debug.

Single process:  python tools/scale_check.py
Full node:       torchrun --standalone --local-addr 127.0.0.1 \
                     --nproc-per-node 8 tools/scale_check.py

Checks, in order:
  1. environment  (HSA_ENABLE_IPC_MODE_LEGACY=0 for dmabuf IPC, torchrun
     rendezvous vars when launched distributed)
  2. device binding (one GPU per rank, LOCAL_RANK -> device index)
  3. process-group init with the backend the trainer would pick
  4. all-reduce of the EXACT gradient bucket the trainer ships
     (LeNet: 2343 fp32; deep: n_params fp32) + a bf16 activation-sized
     tensor; verifies sums and prints per-collective latency
  5. async all-reduce (the overlap_comm path) wait-ordering

Exit code 0 = ready for `torchrun --nproc-per-node 8 bench.py --gpus 8`.
"""
from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def log(rank: int, msg: str) -> None:
    print(f"[rank {rank}] {msg}", flush=True)


def main() -> int:
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    ok = True

    # 1. environment
    ipc = os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY")
    if torch.cuda.is_available() and world > 1 and ipc != "0":
        log(rank, f"FAIL env: HSA_ENABLE_IPC_MODE_LEGACY={ipc!r} (want '0': "
                  "the host driver only supports dmabuf IPC; RCCL fails "
                  "with hipIpcGetMemHandle errors without it)")
        ok = False
    else:
        log(rank, "PASS env: IPC mode")
    if world > 1:
        for var in ("MASTER_ADDR", "MASTER_PORT"):
            if not os.environ.get(var):
                log(rank, f"FAIL env: {var} unset (launch via torchrun "
                          "--master-addr 127.0.0.1)")
                ok = False

    # 2. device binding
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        n = torch.cuda.device_count()
        local = int(os.environ.get("LOCAL_RANK", str(rank)))
        if world > n:
            log(rank, f"WARN binding: {world} ranks > {n} visible GPUs "
                      "(ranks will share devices; gloo fallback)")
        torch.cuda.set_device(local % n)
        log(rank, f"PASS binding: rank {rank} -> cuda:{local % n} of {n} "
                  f"({torch.cuda.get_device_name(local % n)})")
    else:
        log(rank, "WARN binding: no GPU visible (CPU/gloo checks only)")

    # 3. process group with the trainer's backend choice
    from parallel_cnn_amd.parallel import dist as pdist
    if world <= 1 and not os.environ.get("PCNN_DIST_BACKEND"):
        # force a real group even single-process so the collective path runs
        os.environ["PCNN_DIST_BACKEND"] = "nccl" if use_gpu else "gloo"
    ctx = pdist.init_from_env("auto" if use_gpu else "cpu")
    backend = torch.distributed.get_backend()
    log(rank, f"PASS init: backend={backend} world={ctx.world_size}")
    if use_gpu and ctx.world_size > 1 and backend != "nccl":
        log(rank, "WARN init: multi-rank on GPU but backend is not nccl "
                  "(PCNN_DIST_BACKEND override? xGMI will not be used)")

    dev = torch.device("cuda") if use_gpu else torch.device("cpu")

    # 4. bucket all-reduce — the exact payloads the trainers ship
    from parallel_cnn_amd.ops import shapes as S
    payloads = [("lenet-bucket", torch.full((S.N_PARAMS,), 1.0,
                                            dtype=torch.float32, device=dev))]
    try:
        from parallel_cnn_amd.models.deepcnn import DeepCNNSpec
        payloads.append(("deep-bucket", torch.full(
            (DeepCNNSpec().n_params,), 1.0, dtype=torch.float32,
            device=dev)))
    except Exception as e:  # spec import must not kill the preflight
        log(rank, f"WARN deep spec unavailable: {e}")
    if backend == "nccl":
        payloads.append(("bf16-activation", torch.full(
            (64 * 3456,), 1.0, dtype=torch.bfloat16, device=dev)))
    for name, t in payloads:
        if use_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        pdist.allreduce_grads(t)
        if use_gpu:
            torch.cuda.synchronize()
        us = (time.perf_counter() - t0) * 1e6
        want = float(ctx.world_size)
        if torch.allclose(t.float(), torch.full_like(t.float(), want)):
            log(rank, f"PASS all-reduce {name}: {t.numel()} x {t.dtype}, "
                      f"{us:.0f} us (first call includes comm setup)")
        else:
            log(rank, f"FAIL all-reduce {name}: expected {want}, got "
                      f"{t.float().mean().item()}")
            ok = False

    # 5. async all-reduce (overlap_comm wait-ordering)
    t = torch.full((S.N_PARAMS,), 2.0, dtype=torch.float32, device=dev)
    wk = pdist.allreduce_grads_async(t)
    if wk is not None:
        wk.wait()
    if use_gpu:
        torch.cuda.synchronize()
    if torch.allclose(t, torch.full_like(t, 2.0 * ctx.world_size)):
        log(rank, "PASS async all-reduce (overlap_comm path)")
    else:
        log(rank, "FAIL async all-reduce")
        ok = False

    pdist.barrier()
    if ctx.is_main:
        verdict = "READY" if ok else "NOT READY"
        log(rank, f"{verdict}: `torchrun --standalone --local-addr "
                  f"127.0.0.1 --nproc-per-node 8 bench.py --gpus 8 "
                  f"--steps 200 --warmup 20` should "
                  + ("work" if ok else "NOT be attempted yet"))
    torch.distributed.destroy_process_group()
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
