"""Measure the per-step host overhead of the distributed-style Python loop
vs the C++ enqueue loop (single GPU, world=1 — no all-reduce)."""
import sys
import time

import torch

sys.path.insert(0, ".")
from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_mnist
from parallel_cnn_amd.engine.trainer import Trainer

cfg = TrainConfig(batch_size=64, device="cuda", backend="hip",
                  log_interval=0)
t = Trainer(cfg)
x, y = synthetic_mnist(64 * 64, seed=0, structured=False)
xp, yp = t.stage_batch(x, y)
xp, yp = xp.contiguous(), yp.contiguous()
torch.cuda.synchronize()

def cpp_null_stream(n):
    # identical C++ enqueue loop, HIP null stream instead of torch's stream
    w = t.ws
    t._C.hip_train_steps(xp, yp, t.model.params, t.model.grads, w.a1, w.a2,
                         w.y, w.dz, w.dz2, w.dz1, w.loss_accum, 64, n,
                         t.cfg.wgrad_chunk, t.cfg.dt / 64.0, 0, 0, 0)


for name, fn in [
    ("cpp-loop", lambda n: t.run_steps_pooled(xp, yp, n)),
    ("cpp-null-stream", cpp_null_stream),
    ("py-loop", lambda n: [t.step(xp[(s % 64) * 64:(s % 64 + 1) * 64],
                                  yp[(s % 64) * 64:(s % 64 + 1) * 64])
                           for s in range(n)]),
]:
    fn(100)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    fn(2000)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{name}: {dt / 2000 * 1e6:.1f} us/step")
