"""Multi-process DP tests over gloo (world_size 2, CPU) — covers the
distributed path that runs over RCCL on the GPU node: 2-rank DP with the
fused gradient bucket must reproduce the 1-rank trajectory at the same
global batch."""
import os

import pytest
import torch
import torch.multiprocessing as mp

from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_mnist
from parallel_cnn_amd.engine.trainer import Trainer
from parallel_cnn_amd.parallel import dist as pdist

GLOBAL_BATCH = 16
STEPS = 3


def _worker(rank, world, port, result_q):
    os.environ.update({
        "RANK": str(rank),
        "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port),
    })
    ctx = pdist.init_from_env(device="cpu")
    cfg = TrainConfig(backend="cpu", device="cpu",
                      batch_size=GLOBAL_BATCH // world, log_interval=0)
    t = Trainer(cfg, ctx=ctx)
    x, y = synthetic_mnist(GLOBAL_BATCH * STEPS, seed=0)
    Bl = cfg.batch_size
    for s in range(STEPS):
        lo = s * GLOBAL_BATCH + rank * Bl
        t.step(*t.stage_batch(x[lo:lo + Bl], y[lo:lo + Bl]))
    loss, n = t.consume_loss()
    if rank == 0:
        result_q.put((t.model.params.clone(), loss, n))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world,port", [(2, 29531), (4, 29532)])
def test_dp_matches_single_rank(world, port):
    # single-rank reference trajectory at the same global batch
    cfg = TrainConfig(backend="cpu", device="cpu", batch_size=GLOBAL_BATCH,
                      log_interval=0)
    ref = Trainer(cfg)
    x, y = synthetic_mnist(GLOBAL_BATCH * STEPS, seed=0)
    for s in range(STEPS):
        lo = s * GLOBAL_BATCH
        ref.step(*ref.stage_batch(x[lo:lo + GLOBAL_BATCH],
                                  y[lo:lo + GLOBAL_BATCH]))
    ref_loss, ref_n = ref.consume_loss()

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    params, loss, n = q.get()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0

    assert n == ref_n == GLOBAL_BATCH * STEPS
    assert abs(loss - ref_loss) < 1e-3
    assert torch.allclose(params, ref.model.params, atol=1e-5), \
        (params - ref.model.params).abs().max()

def _accum_worker(rank, world, port, result_q):
    os.environ.update({
        "RANK": str(rank),
        "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port),
    })
    ctx = pdist.init_from_env(device="cpu")
    # 2 ranks x bs=4 x grad_accum=2 == one global-batch-16 step
    cfg = TrainConfig(backend="cpu", device="cpu", batch_size=4,
                      grad_accum=2, log_interval=0)
    t = Trainer(cfg, ctx=ctx)
    x, y = synthetic_mnist(16, seed=0)
    for micro in range(2):
        lo = micro * 8 + rank * 4
        t.step(*t.stage_batch(x[lo:lo + 4], y[lo:lo + 4]))
    if rank == 0:
        result_q.put(t.model.params.clone())
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp_with_grad_accumulation_matches_big_batch():
    """DP(2) x grad_accum(2) at bs=4 == one single-rank bs=16 step: the
    all-reduce must fire only at the accumulation boundary and the update
    scale must fold in world*accum*micro_batch."""
    cfg = TrainConfig(backend="cpu", device="cpu", batch_size=16,
                      log_interval=0)
    ref = Trainer(cfg)
    x, y = synthetic_mnist(16, seed=0)
    ref.step(*ref.stage_batch(x, y))

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_accum_worker, args=(r, 2, 29533, q))
             for r in range(2)]
    for p in procs:
        p.start()
    params = q.get()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert torch.allclose(params, ref.model.params, atol=1e-6), \
        (params - ref.model.params).abs().max()


def _deep_worker(rank, world, port, result_q):
    os.environ.update({
        "RANK": str(rank),
        "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port),
    })
    from parallel_cnn_amd.data.mnist import synthetic_images
    from parallel_cnn_amd.engine.deep import DeepTrainer
    ctx = pdist.init_from_env(device="cpu")
    cfg = TrainConfig(backend="torchref", device="cpu", model="deepcnn",
                      batch_size=8, log_interval=0)
    t = DeepTrainer(cfg, ctx=ctx)
    x, y = synthetic_images(16, 32, 32, 3, seed=0, structured=False)
    lo = rank * 8
    t.step(*t.stage_batch(x[lo:lo + 8], y[lo:lo + 8]))
    if rank == 0:
        result_q.put(t.model.params.clone())
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_deep_dp_matches_single_rank():
    """DeepCNN DP(2) at bs=8/rank == one single-rank bs=16 step (covers
    the deep trainer's fused-bucket all-reduce path the 8-GPU scale run
    exercises with --model deepcnn)."""
    from parallel_cnn_amd.data.mnist import synthetic_images
    from parallel_cnn_amd.engine.deep import DeepTrainer
    cfg = TrainConfig(backend="torchref", device="cpu", model="deepcnn",
                      batch_size=16, log_interval=0)
    ref = DeepTrainer(cfg)
    x, y = synthetic_images(16, 32, 32, 3, seed=0, structured=False)
    ref.step(*ref.stage_batch(x, y))

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_deep_worker, args=(r, 2, 29534, q))
             for r in range(2)]
    for p in procs:
        p.start()
    params = q.get()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert torch.allclose(params, ref.model.params, atol=1e-5), \
        (params - ref.model.params).abs().max()


@pytest.mark.timeout(300)
def test_scale_check_gloo_ws2():
    """The multi-GPU preflight tool runs green over gloo at world_size 2
    (the CPU stand-in for the nccl run the GPU test does)."""
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29653", "tools/scale_check.py"],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "READY" in out.stdout and "FAIL" not in out.stdout
