"""Randomized cross-validation of the DeepCNN hip path against the fp32
oracle: random channel sets (2-4 stages, multiples of 16), batch sizes,
activation dtypes, and engine modes (implicit / materialized), one full
training step each, parameters compared to the chain-rule reference.

Run on a GPU box:  python tools/fuzz_deep.py [--n 12] [--seed 0]
Exit code != 0 on the first mismatch (prints the failing config).
"""
from __future__ import annotations

import argparse
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=12)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args()
    assert torch.cuda.is_available(), "fuzz needs a GPU"
    from parallel_cnn_amd.config import TrainConfig
    from parallel_cnn_amd.engine.deep import DeepTrainer
    from parallel_cnn_amd.models.deepcnn import DeepCNN, DeepCNNSpec
    from parallel_cnn_amd.ops import deep_ref
    from parallel_cnn_amd.data.mnist import synthetic_images

    rng = random.Random(args.seed)
    fails = 0
    for trial in range(args.n):
        nstages = rng.choice([2, 3, 4])
        channels = tuple(16 * rng.randint(1, 4) for _ in range(nstages))
        B = rng.choice([2, 3, 8, 16, 32])
        dtype = rng.choice(["fp32", "fp32", "bf16"])  # fp32-weighted: the
        # oracle comparison is tight only at fp32; bf16 runs check for
        # crashes/NaNs with a loose bound
        implicit = rng.choice([True, False])
        desc = (f"trial {trial}: channels={channels} B={B} "
                f"dtype={dtype} implicit={implicit}")
        cfg = TrainConfig(batch_size=B, device="cuda", backend="hip",
                          act_dtype=dtype, log_interval=0,
                          deep_channels=",".join(map(str, channels)),
                          deep_implicit=implicit, seed=trial)
        try:
            t = DeepTrainer(cfg)
            x, labels = synthetic_images(B, 32, 32, 3, seed=100 + trial,
                                         structured=False)
            t.step(*t.stage_batch(x, labels))
            torch.cuda.synchronize()
            ref = DeepCNN(seed=cfg.seed, spec=DeepCNNSpec(channels=channels))
            xh = x.view(B, 32, 32, 3)
            acts, pouts, y = deep_ref.forward(xh, ref)
            grads, _ = deep_ref.backward(xh, ref, acts, pouts, y, labels)
            with torch.no_grad():
                ref.params += cfg.dt * (1.0 / B) * grads
            got = t.model.params.cpu()
            assert torch.isfinite(got).all(), "non-finite params"
            diff = (got - ref.params).abs().max().item()
            tol = 5e-3 if dtype == "fp32" else 5e-2
            status = "OK " if diff < tol else "FAIL"
            print(f"{status} {desc}: max|dp|={diff:.2e} (tol {tol:g})",
                  flush=True)
            if diff >= tol:
                fails += 1
        except Exception as e:
            print(f"FAIL {desc}: {type(e).__name__}: {e}", flush=True)
            fails += 1
    print(f"{args.n - fails}/{args.n} configs passed")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
