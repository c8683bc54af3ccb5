"""Flagship benchmark — LeNet-5 training step throughput (BASELINE.json).

Single GPU:   python bench.py --gpus 1 --steps 200 --warmup 20
Multi-GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Per-rank batch is 64 (the BASELINE.json config); weak scaling (per-GPU work
fixed).  Synthetic 28x28x1 data, random-init weights (the reference's MNIST
image blobs are absent and the baseline is defined on synthetic data).
Timing: W untimed warmup steps, then exactly K steps bracketed by
barrier + torch.cuda.synchronize on both sides; MAX time over ranks;
rank 0 prints one JSON line.

Reference headline being compared against: the CUDA variant's
2,996.99 ms / 60k-sample epoch on a T4 ~= 20,020 images/sec.
"""
from __future__ import annotations

import argparse
import json
import time

import torch

BASELINE_IMAGES_PER_SEC = 60000.0 / 2.9969857  # BASELINE.md CUDA headline


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--batch-size", type=int, default=64,
                   help="per-GPU batch (BASELINE config: 64)")
    p.add_argument("--act-dtype", type=str, default="bf16",
                   choices=["bf16", "fp16", "fp32"])
    p.add_argument("--device", type=str, default="auto")
    p.add_argument("--wgrad-chunk", type=int, default=0)
    p.add_argument("--model", type=str, default="lenet5",
                   choices=["lenet5", "deepcnn"])
    p.add_argument("--use-graph", action="store_true",
                   help="capture the step in a hipGraph and replay it")
    p.add_argument("--pool", default="trainable",
                   choices=["trainable", "max"])
    p.add_argument("--loss", default="residual",
                   choices=["residual", "softmax_ce"])
    p.add_argument("--overlap-comm", default=None,
                   action=argparse.BooleanOptionalAction,
                   help="two-bucket DP: overlap the fc/pool grad all-reduce "
                        "with the conv wgrad.  Default OFF: the 9.4 KB "
                        "LeNet bucket is pure latency, and splitting it "
                        "pays a second collective latency that the ~7 us "
                        "of remaining wgrad cannot hide — ONE fused bucket "
                        "is the design point (SURVEY 5.8); measure with "
                        "--overlap-comm on multi-GPU nodes")
    args = p.parse_args()

    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.data.mnist import synthetic_images, synthetic_mnist
    from parallel_cnn_amd.engine.deep import DeepTrainer
    from parallel_cnn_amd.engine.trainer import Trainer
    from parallel_cnn_amd.parallel import dist as pdist

    cfg = TrainConfig(batch_size=args.batch_size, act_dtype=args.act_dtype,
                      device=args.device, log_interval=0, data="synthetic",
                      wgrad_chunk=args.wgrad_chunk, model=args.model,
                      overlap_comm=bool(args.overlap_comm), pool=args.pool,
                      loss=args.loss)
    device = cfg.resolved_device()
    ctx = pdist.init_from_env(device)
    n_gpus = ctx.world_size if ctx.world_size > 1 else args.gpus
    if ctx.world_size == 1 and args.gpus > 1:
        raise SystemExit("--gpus N>1 must be launched via torch.distributed.run")

    trainer = (DeepTrainer(cfg, ctx=ctx) if args.model == "deepcnn"
               else Trainer(cfg, ctx=ctx))
    if args.use_graph:
        trainer.enable_graph()
    B = args.batch_size

    # Device-resident synthetic epoch pool (no H2D inside the timed loop; the
    # pool is one epoch's worth of batches, cycled).
    n_pool_batches = max(1, min(args.steps + args.warmup,
                                60000 // max(1, B)))
    if args.model == "deepcnn":
        x_host, y_host = synthetic_images(n_pool_batches * B, 32, 32, 3,
                                          seed=1234 + ctx.rank,
                                          structured=False)
    else:
        x_host, y_host = synthetic_mnist(n_pool_batches * B,
                                         seed=1234 + ctx.rank,
                                         structured=False)
    x_pool, y_pool = trainer.stage_batch(x_host, y_host)
    x_pool, y_pool = x_pool.contiguous(), y_pool.contiguous()
    if device == "cuda":
        torch.cuda.synchronize()

    def run(n_steps: int):
        if args.model == "deepcnn":
            fn = (trainer.step_graph if getattr(trainer, "_graph", None)
                  is not None else trainer.step)
            for st in range(n_steps):
                i = (st % n_pool_batches) * B
                fn(x_pool[i:i + B], y_pool[i:i + B])
        else:
            trainer.run_steps_pooled(x_pool, y_pool, n_steps)

    def timed_block() -> float:
        """One measurement: exactly K steps bracketed by barrier +
        torch.cuda.synchronize on both sides; MAX elapsed over ranks."""
        pdist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        run(args.steps)
        if device == "cuda":
            torch.cuda.synchronize()
        pdist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()
        el = time.perf_counter() - t0
        return pdist.allreduce_max_scalar(
            el, device=device if device == "cuda" else None)

    run(args.warmup)
    # Short driver runs (e.g. --steps 20 ~ 0.4 ms timed region) are noise-
    # limited: repeat the K-step block until >= 50 ms of cumulative timed
    # region (each block still times EXACTLY K steps) and report the
    # median block.
    blocks = [timed_block()]
    while sum(blocks) < 0.050 and len(blocks) < 64:
        blocks.append(timed_block())
    blocks.sort()
    elapsed = blocks[len(blocks) // 2]  # median block

    global_batch = B * n_gpus
    images_per_sec = args.steps * global_batch / elapsed
    dtype = args.act_dtype if device == "cuda" else "fp32"
    # the published baseline is the reference's LeNet CUDA headline; other
    # model families have no reference number
    vs_baseline = (images_per_sec / BASELINE_IMAGES_PER_SEC
                   if args.model == "lenet5" else None)
    result = {
        "metric": "training images/sec (whole node)",
        "value": images_per_sec,
        "unit": "images/sec",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": vs_baseline,
        "dtype": dtype,
        "data": "synthetic",
        "config": {
            "model": ("DeepCNN 32x32x3 (3x(conv5x5+trainable-pool2x2) -> "
                      "fc1024x10), implicit-im2col+MFMA GEMM path"
                      if args.model == "deepcnn" else
                      "LeNet-5 28x28x1 (conv6x5x5 -> trainable-pool4x4 -> fc216x10)"),
            "global_batch": global_batch,
            "per_gpu_batch": B,
            "input": "32x32x3" if args.model == "deepcnn" else "28x28x1",
            "parallelism": f"dp{n_gpus}",
            "dist_backend": ctx.backend or "none",
            "pool": args.pool,
            "loss": args.loss,
            "backend": trainer.backend,
            "hipgraph": bool(getattr(trainer, "_graph", None)),
            "overlap_comm": cfg.overlap_comm,
            "timed_blocks": len(blocks),
            "block_min_s": blocks[0],
            "block_max_s": blocks[-1],
        },
    }
    if ctx.is_main:
        print(json.dumps(result), flush=True)
    pdist.barrier()
    if pdist.is_distributed():
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
