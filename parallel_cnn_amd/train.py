"""CLI training driver — the framework equivalent of the reference's
`main -> loaddata -> learn -> test`, with the same stdout
shape ("Learning", per-epoch "error: %e", "Error Rate: %.2f%%") plus real
flags, checkpointing and DP.

Run:   python -m parallel_cnn_amd.train [--batch-size 64 ...]
DP:    torchrun --standalone --nproc-per-node 8 -m parallel_cnn_amd.train ...
"""
from __future__ import annotations

import argparse
import os
import time

import torch

from .config import TrainConfig
from .data.mnist import load_mnist, synthetic_images, synthetic_mnist
from .engine.deep import DeepTrainer
from .engine.trainer import Trainer
from .parallel import dist as pdist


def load_datasets(cfg: TrainConfig):
    if cfg.data == "mnist":
        if cfg.model == "deepcnn":
            raise ValueError(
                "--model deepcnn expects 32x32x3 inputs; MNIST is 28x28x1 "
                "(use --data synthetic with deepcnn, or --model lenet5 "
                "with --data mnist)")
        xtr, ytr = load_mnist(
            os.path.join(cfg.data_dir, "train-images.idx3-ubyte"),
            os.path.join(cfg.data_dir, "train-labels.idx1-ubyte"))
        xte, yte = load_mnist(
            os.path.join(cfg.data_dir, "t10k-images.idx3-ubyte"),
            os.path.join(cfg.data_dir, "t10k-labels.idx1-ubyte"))
    elif cfg.model == "deepcnn":
        xtr, ytr = synthetic_images(cfg.train_count, 32, 32, 3,
                                    seed=cfg.seed)
        xte, yte = synthetic_images(cfg.test_count, 32, 32, 3,
                                    seed=cfg.seed + 1)
    else:
        xtr, ytr = synthetic_mnist(cfg.train_count, seed=cfg.seed)
        xte, yte = synthetic_mnist(cfg.test_count, seed=cfg.seed + 1)
    return xtr, ytr, xte, yte


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__)
    TrainConfig.add_cli_args(p)
    p.add_argument("--profile", action="store_true",
                   help="per-phase device-synced timing report")
    args = p.parse_args(argv)
    cfg = TrainConfig.from_args(args)

    profile = bool(args.profile)
    ctx = pdist.init_from_env(cfg.resolved_device())
    trainer = (DeepTrainer(cfg, ctx=ctx) if cfg.model == "deepcnn"
               else Trainer(cfg, ctx=ctx))
    if profile and hasattr(trainer, "enable_profiling"):
        trainer.enable_profiling()
    if cfg.ckpt_load:
        from .utils.checkpoint import load_checkpoint
        load_checkpoint(trainer, cfg.ckpt_load)

    xtr, ytr, xte, yte = load_datasets(cfg)

    if ctx.is_main:
        print("Learning", flush=True)
    t0 = time.perf_counter()
    # a loaded checkpoint restores trainer.epoch — resume the remaining
    # epochs, not the full schedule
    for epoch in range(trainer.epoch, cfg.epochs):
        err = trainer.train_epoch(xtr, ytr)
        dt = time.perf_counter() - t0
        if ctx.is_main:
            print(f"error: {err:e}, time_on_gpu: {dt:f}", flush=True)
        if err < cfg.threshold:
            break
    if trainer.device.type == "cuda":
        torch.cuda.synchronize()
    total = time.perf_counter() - t0
    if ctx.is_main:
        print(f"\n Time - {total * 1e3:f} ms", flush=True)

    if cfg.ckpt_save and ctx.is_main:
        from .utils.checkpoint import save_checkpoint
        save_checkpoint(trainer, cfg.ckpt_save)
        print(f"saved checkpoint: {cfg.ckpt_save}", flush=True)

    err_rate = trainer.evaluate(xte, yte)
    if ctx.is_main:
        print(f"Error Rate: {err_rate:.2f}%", flush=True)
        if profile and getattr(trainer, "timers", None) is not None:
            print("-- per-phase timers (device-synced) --")
            print(trainer.timers.report(), flush=True)
    pdist.barrier()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
