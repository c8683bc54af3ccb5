"""GPU (MI355X) tests: every HIP kernel against the fp32 PyTorch/native-CPU
oracle, end-to-end step parity, convergence, and the bench path."""
import json
import subprocess
import sys

import pytest
import torch

from parallel_cnn_amd import _C
from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_mnist
from parallel_cnn_amd.engine.trainer import Trainer
from parallel_cnn_amd.models.lenet import LeNet5
from parallel_cnn_amd.ops import native
from parallel_cnn_amd.ops import shapes as S
from parallel_cnn_amd.ops import torch_ref

pytestmark = pytest.mark.gpu


def make_case(B, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(B, S.IN_PIX, generator=g)
    labels = torch.randint(0, 10, (B,), generator=g)
    params = (0.5 - torch.rand(S.N_PARAMS, generator=g)).float()
    return x, labels, params


def run_gpu_step_pieces(x, labels, params, act_dtype, device):
    """Run kernel A (fwdbwd) + kernel B (wgrad) on the GPU; return all
    intermediates on the host."""
    B = x.shape[0]
    ad = act_dtype
    xd = x.to(device, dtype=ad)
    ld = labels.to(device, dtype=torch.int32)
    pd = params.to(device)
    a1 = torch.empty(B, S.C1_OUT, dtype=ad, device=device)
    a2 = torch.empty(B, S.S1_OUT, dtype=ad, device=device)
    y = torch.empty(B, S.FC_OUT, dtype=torch.float32, device=device)
    dz = torch.empty(B, S.FC_OUT, dtype=torch.float32, device=device)
    dz2 = torch.empty(B, S.S1_OUT, dtype=torch.float32, device=device)
    dz1 = torch.empty(B, S.C1_OUT, dtype=ad, device=device)
    loss = torch.zeros(1, dtype=torch.float32, device=device)
    corr = torch.zeros(1, dtype=torch.int32, device=device)
    grads = torch.zeros(S.N_PARAMS, dtype=torch.float32, device=device)
    stream = native.current_stream_handle()
    _C.hip_fwdbwd(xd, pd, a1, a2, y, dz, dz2, dz1, ld, loss, corr, B, 0,
                  stream)
    _C.hip_wgrad(xd, a1, a2, dz, dz2, dz1, grads, B, 8, stream)
    torch.cuda.synchronize()
    return (a1.float().cpu(), a2.float().cpu(), y.cpu(), dz.cpu(), dz2.cpu(),
            dz1.float().cpu(), grads.cpu(), float(loss.item()))


def ref_step_pieces(x, labels, params):
    a1, a2, y = torch_ref.forward(x, params)
    dz, dz2, dz1, grads, loss = torch_ref.backward(x, params, a1, a2, y,
                                                   labels)
    return (a1.reshape(x.shape[0], -1), a2.reshape(x.shape[0], -1), y, dz,
            dz2, dz1, grads, loss)


@pytest.mark.parametrize("B", [1, 64, 100])
def test_fwdbwd_wgrad_fp32_matches_oracle(B, device):
    x, labels, params = make_case(B, seed=B)
    got = run_gpu_step_pieces(x, labels, params, torch.float32, device)
    want = ref_step_pieces(x, labels, params)
    names = ["a1", "a2", "y", "dz", "dz2", "dz1", "grads"]
    for n, a, b in zip(names, got[:7], want[:7]):
        diff = (a.reshape(-1) - b.reshape(-1)).abs().max().item()
        # grads are B-way sums accumulated with atomics: allow summation-
        # order roundoff to scale with the magnitude of the reduction
        tol = 2e-4 * max(1.0, b.abs().max().item()) if n == "grads" else 2e-4
        assert diff < tol, f"{n}: max abs diff {diff}"
    assert abs(got[7] - want[7]) < 1e-2 * max(1.0, abs(want[7]))


@pytest.mark.parametrize("B", [64])
def test_fwdbwd_wgrad_bf16_acts_close(B, device):
    """bf16 activation storage: arithmetic is fp32 so tolerances are the
    bf16 storage quantum (~0.8% relative)."""
    x, labels, params = make_case(B, seed=5)
    got = run_gpu_step_pieces(x, labels, params, torch.bfloat16, device)
    want = ref_step_pieces(x, labels, params)
    for n, a, b, tol in [("y", got[2], want[2], 2e-2),
                         ("dz", got[3], want[3], 2e-2),
                         ("dz2", got[4], want[4], 2e-2),
                         ("dz1", got[5], want[5], 2e-2)]:
        diff = (a.reshape(-1) - b.reshape(-1)).abs().max().item()
        assert diff < tol, f"{n}: max abs diff {diff}"
    gdiff = (got[6] - want[6]).abs().max().item()
    gref = want[6].abs().max().item()
    assert gdiff < 2e-2 * max(1.0, gref), f"grads: {gdiff} vs max {gref}"


def test_update_kernel(device):
    params = torch.zeros(S.N_PARAMS, device=device)
    grads = torch.ones(S.N_PARAMS, device=device)
    _C.hip_update(params, grads, 0.05, native.current_stream_handle())
    torch.cuda.synchronize()
    assert torch.allclose(params.cpu(), torch.full((S.N_PARAMS,), 0.05))
    assert grads.abs().sum().item() == 0


def test_eval_mode_correct_count(device):
    B = 32
    x, labels, params = make_case(B, seed=9)
    # oracle predictions
    _, _, y = torch_ref.forward(x, params)
    want_correct = int((y.argmax(1) == labels).sum().item())
    xd = x.to(device, dtype=torch.float32)
    ld = labels.to(device, dtype=torch.int32)
    pd = params.to(device)
    a1 = torch.empty(B, S.C1_OUT, dtype=torch.float32, device=device)
    a2 = torch.empty(B, S.S1_OUT, dtype=torch.float32, device=device)
    yd = torch.empty(B, S.FC_OUT, dtype=torch.float32, device=device)
    e = torch.empty(0, device=device)
    corr = torch.zeros(1, dtype=torch.int32, device=device)
    loss = torch.zeros(1, dtype=torch.float32, device=device)
    _C.hip_fwdbwd(xd, pd, a1, a2, yd, e, e, e, ld, loss, corr, B, 1,
                  native.current_stream_handle())
    torch.cuda.synchronize()
    assert int(corr.item()) == want_correct


def test_trainer_step_trajectory_matches_cpu(device):
    """5 fp32 steps on GPU == 5 steps of the native CPU engine."""
    cfg_g = TrainConfig(batch_size=16, device="cuda", backend="hip",
                        act_dtype="fp32", log_interval=0)
    cfg_c = TrainConfig(batch_size=16, device="cpu", backend="cpu",
                        log_interval=0)
    tg, tc = Trainer(cfg_g), Trainer(cfg_c)
    x, y = synthetic_mnist(80, seed=3)
    for s in range(5):
        xb, yb = x[s * 16:(s + 1) * 16], y[s * 16:(s + 1) * 16]
        tg.step(*tg.stage_batch(xb, yb))
        tc.step(*tc.stage_batch(xb, yb))
    torch.cuda.synchronize()
    diff = (tg.model.params.cpu() - tc.model.params).abs().max().item()
    assert diff < 5e-4, diff
    lg, ng = tg.consume_loss()
    lc, nc = tc.consume_loss()
    assert ng == nc == 80
    assert abs(lg - lc) < 1e-2 * max(1.0, lc)


def test_gpu_convergence_bf16(device):
    """The flagship bf16 path must actually learn the structured bands."""
    xtr, ytr = synthetic_mnist(4096, seed=2)
    xte, yte = synthetic_mnist(512, seed=3)
    cfg = TrainConfig(batch_size=32, device="cuda", backend="hip",
                      act_dtype="bf16", log_interval=0)
    t = Trainer(cfg)
    before = t.evaluate(xte, yte)
    for _ in range(3):
        t.train_epoch(xtr, ytr, log=lambda *a: None)
    after = t.evaluate(xte, yte)
    assert after < before
    assert after < 5.0, (before, after)


def test_fused_step_loop_matches_python_loop(device):
    """The C++ multi-step enqueue path must produce the same trajectory as
    per-step Python calls."""
    x, y = synthetic_mnist(64, seed=11)
    cfg = TrainConfig(batch_size=16, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0)
    t1, t2 = Trainer(cfg), Trainer(cfg)
    xp, yp = t1.stage_batch(x, y)
    t1.run_steps_pooled(xp.contiguous(), yp.contiguous(), 4)  # C++ loop
    for s in range(4):                                        # Python loop
        t2.step(xp[s * 16:(s + 1) * 16], yp[s * 16:(s + 1) * 16])
    torch.cuda.synchronize()
    diff = (t1.model.params - t2.model.params).abs().max().item()
    assert diff < 1e-5, diff
    l1, n1 = t1.consume_loss()
    l2, n2 = t2.consume_loss()
    assert n1 == n2 == 64
    assert abs(l1 - l2) < 1e-3 * max(1.0, l2)


def test_wgrad_chunk_sizes_agree(device):
    """The wgrad chunk knob must not change the math."""
    B = 64
    x, labels, params = make_case(B, seed=21)
    base = None
    for chunk in (2, 8, 32, 64):
        xd = x.to(device)
        pd = params.to(device)
        ld = labels.to(device, dtype=torch.int32)
        a1 = torch.empty(B, S.C1_OUT, device=device)
        a2 = torch.empty(B, S.S1_OUT, device=device)
        yv = torch.empty(B, S.FC_OUT, device=device)
        dz = torch.empty(B, S.FC_OUT, device=device)
        dz2 = torch.empty(B, S.S1_OUT, device=device)
        dz1 = torch.empty(B, S.C1_OUT, device=device)
        loss = torch.zeros(1, device=device)
        corr = torch.zeros(1, dtype=torch.int32, device=device)
        grads = torch.zeros(S.N_PARAMS, device=device)
        st = native.current_stream_handle()
        _C.hip_fwdbwd(xd, pd, a1, a2, yv, dz, dz2, dz1, ld, loss, corr, B, 0,
                      st)  # fp32 acts: dz1 fp32 too
        _C.hip_wgrad(xd, a1, a2, dz, dz2, dz1, grads, B, chunk, st)
        torch.cuda.synchronize()
        g = grads.cpu()
        if base is None:
            base = g
        else:
            assert torch.allclose(g, base, atol=1e-3), chunk


def test_fwdbwd_wgrad_fp16_acts_close(device):
    B = 64
    x, labels, params = make_case(B, seed=15)
    got = run_gpu_step_pieces(x, labels, params, torch.float16, device)
    want = ref_step_pieces(x, labels, params)
    for n, a, b in [("y", got[2], want[2]), ("dz", got[3], want[3]),
                    ("dz2", got[4], want[4]), ("dz1", got[5], want[5])]:
        diff = (a.reshape(-1) - b.reshape(-1)).abs().max().item()
        assert diff < 1e-2, f"{n}: max abs diff {diff}"
    gdiff = (got[6] - want[6]).abs().max().item()
    assert gdiff < 1e-2 * max(1.0, want[6].abs().max().item()), gdiff


@pytest.mark.parametrize("overlap", [False, True])
def test_dp2_gloo_single_gpu(overlap, device):
    """Two ranks sharing one GPU over gloo: exercises the full distributed
    engine path (sharding, fused-bucket all-reduce — single or two-bucket
    overlapped — identical update) on device tensors.  Transport is gloo;
    the 8-GPU RCCL run uses the same code with backend nccl."""
    import subprocess
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29617", "bench.py", "--gpus", "2", "--steps",
         "20", "--warmup", "5", "--batch-size", "16"]
        + (["--overlap-comm"] if overlap else ["--no-overlap-comm"]),
        capture_output=True, text=True, timeout=600,
        env={**__import__("os").environ, "PCNN_DIST_BACKEND": "gloo"})
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "dp2"
    assert r["value"] > 0


def test_grad_accumulation_gpu_matches_big_batch(device):
    x, y = synthetic_mnist(32, seed=61)
    cfg_a = TrainConfig(batch_size=16, device="cuda", backend="hip",
                        act_dtype="fp32", log_interval=0, grad_accum=2)
    cfg_b = TrainConfig(batch_size=32, device="cuda", backend="hip",
                        act_dtype="fp32", log_interval=0)
    ta, tb = Trainer(cfg_a), Trainer(cfg_b)
    ta.step(*ta.stage_batch(x[:16], y[:16]))
    ta.step(*ta.stage_batch(x[16:], y[16:]))
    tb.step(*tb.stage_batch(x, y))
    torch.cuda.synchronize()
    diff = (ta.model.params - tb.model.params).abs().max().item()
    assert diff < 1e-5, diff


def test_trainer_fp16_step_close_to_fp32(device):
    """Trainer-level fp16 activation mode: same trajectory to fp16
    tolerance."""
    x, y = synthetic_mnist(32, seed=51)
    cfg16 = TrainConfig(batch_size=32, device="cuda", backend="hip",
                        act_dtype="fp16", log_interval=0)
    cfg32 = TrainConfig(batch_size=32, device="cuda", backend="hip",
                        act_dtype="fp32", log_interval=0)
    t16, t32 = Trainer(cfg16), Trainer(cfg32)
    t16.step(*t16.stage_batch(x, y))
    t32.step(*t32.stage_batch(x, y))
    torch.cuda.synchronize()
    d16 = t16.model.params - LeNet5(device, seed=0).params
    d32 = t32.model.params - LeNet5(device, seed=0).params
    diff = (d16 - d32).abs().max().item()
    scale = d32.abs().max().item()
    assert diff < 0.05 * max(1e-3, scale), (diff, scale)


def test_native_extension_is_loaded_on_gpu(device):
    """The HIP path must be the one that runs (no silent eager fallback)."""
    assert native.available()
    assert _C.HAS_HIP_KERNELS
    cfg = TrainConfig(batch_size=4, device="cuda", log_interval=0)
    t = Trainer(cfg)
    assert t.backend == "hip"


@pytest.mark.parametrize("pool,loss", [("max", "residual"),
                                       ("trainable", "softmax_ce"),
                                       ("max", "softmax_ce")])
def test_gpu_modes_match_oracle(pool, loss, device):
    """Max-pool and softmax-CE kernel modes vs the fp32 torch oracle:
    full step trajectory (fwdbwd + wgrad + update)."""
    B = 16
    cfg_g = TrainConfig(batch_size=B, device="cuda", backend="hip",
                        act_dtype="fp32", log_interval=0, pool=pool,
                        loss=loss)
    cfg_c = TrainConfig(batch_size=B, device="cpu", backend="cpu",
                        log_interval=0, pool=pool, loss=loss)
    tg, tc = Trainer(cfg_g), Trainer(cfg_c)
    x, y = synthetic_mnist(B * 3, seed=23)
    for s in range(3):
        xb, yb = x[s * B:(s + 1) * B], y[s * B:(s + 1) * B]
        tg.step(*tg.stage_batch(xb, yb))
        tc.step(*tc.stage_batch(xb, yb))
    torch.cuda.synchronize()
    diff = (tg.model.params.cpu() - tc.model.params).abs().max().item()
    assert diff < 1e-3, f"{pool}/{loss}: {diff}"
    lg, ng = tg.consume_loss()
    lc, nc = tc.consume_loss()
    assert ng == nc
    assert abs(lg - lc) < 1e-2 * max(1.0, lc), (lg, lc)
    if pool == "max":
        from parallel_cnn_amd.ops import shapes as SS
        s1grad = tg.model.params[SS.OFF_S1W:SS.OFF_FW].cpu()
        s1init = tc.model.params[SS.OFF_S1W:SS.OFF_FW]
        assert torch.equal(s1grad, s1init)  # pool params untouched


def test_step_nondeterminism_bounded(device):
    """Race/nondeterminism detector: two identical steps from identical
    state may differ only by atomic-ordering roundoff (a data race or
    missing sync would blow far past this bound)."""
    x, y = synthetic_mnist(64, seed=41)
    cfg = TrainConfig(batch_size=64, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0)
    t1, t2 = Trainer(cfg), Trainer(cfg)
    xb1, yb1 = t1.stage_batch(x, y)
    xb2, yb2 = t2.stage_batch(x, y)
    for _ in range(3):
        t1.step(xb1, yb1)
        t2.step(xb2, yb2)
    torch.cuda.synchronize()
    diff = (t1.model.params - t2.model.params).abs().max().item()
    assert diff < 1e-5, diff


def test_graph_step_matches_eager(device):
    """hipGraph-captured step replay == eager step trajectory."""
    x, y = synthetic_mnist(64, seed=19)
    cfg = TrainConfig(batch_size=16, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0)
    t1, t2 = Trainer(cfg), Trainer(cfg)
    t1.enable_graph()
    xp1, yp1 = t1.stage_batch(x, y)
    xp2, yp2 = t2.stage_batch(x, y)
    t1.run_steps_pooled(xp1.contiguous(), yp1.contiguous(), 4)  # graph
    t2.run_steps_pooled(xp2.contiguous(), yp2.contiguous(), 4)  # C++ loop
    torch.cuda.synchronize()
    diff = (t1.model.params - t2.model.params).abs().max().item()
    assert diff < 1e-5, diff
    l1, n1 = t1.consume_loss()
    l2, n2 = t2.consume_loss()
    assert n1 == n2 == 64
    assert abs(l1 - l2) < 1e-3 * max(1.0, l2)


def test_prefetcher_epoch_matches_sync_staging(device):
    """train_epoch's double-buffered copy-stream pipeline must produce the
    same trajectory as synchronous staging."""
    x, y = synthetic_mnist(256, seed=31)
    cfg = TrainConfig(batch_size=32, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0)
    t1, t2 = Trainer(cfg), Trainer(cfg)
    t1.train_epoch(x, y, log=lambda *a: None)      # prefetcher path
    for s in range(0, 256, 32):                    # sync path
        t2.step(*t2.stage_batch(x[s:s + 32], y[s:s + 32]))
    torch.cuda.synchronize()
    diff = (t1.model.params - t2.model.params).abs().max().item()
    assert diff < 1e-5, diff


def test_native_cli_trainer(device, tmp_path):
    """tools/pcnn_train: the no-Python native driver must train, report in
    the reference's stdout shape, and write a checkpoint the Python side
    can load (cross-implementation checkpoint compatibility)."""
    import os
    bin_path = os.path.join(os.path.dirname(__file__), "..", "tools",
                            "pcnn_train")
    if not os.path.exists(bin_path):
        pytest.skip("native CLI not built")
    ck = str(tmp_path / "cli.bin")
    out = subprocess.run(
        [bin_path, "--epochs", "2", "--train-count", "4096", "--test-count",
         "1024", "--batch-size", "64", "--ckpt-save", ck],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "Learning" in out.stdout
    assert "error: " in out.stdout
    assert "Error Rate: " in out.stdout
    errs = [float(l.split("error: ")[1].split(",")[0])
            for l in out.stdout.splitlines() if l.startswith("error: ")]
    assert errs[-1] < errs[0], errs  # learning on the structured bands
    from parallel_cnn_amd.models.lenet import LeNet5
    m = LeNet5()
    m.load(ck)  # right size, loadable
    assert torch.isfinite(m.params).all()


def test_bench_contract_single_gpu(device):
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "30",
         "--warmup", "5"],
        capture_output=True, text=True, timeout=600, check=True)
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["n_gpus"] == 1 and r["steps"] == 30
    assert r["unit"] == "images/sec" and r["higher_is_better"]
    assert r["value"] > 0 and r["ms_per_step"] > 0
    assert r["dtype"] == "bf16"
    assert r["config"]["global_batch"] == 64


def _torchrun_ws1_nccl(extra, timeout=600):
    """Launch bench.py under torchrun at world_size=1 with a FORCED nccl
    process group: the RCCL comm init, in-step dist.all_reduce, async
    work handles and (with --use-graph) in-graph collectives all execute
    on hardware — the same code the 8-GPU driver run uses (VERDICT r1 #1:
    de-risk the first SCALE run)."""
    import os
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
         "--master-port", "29651", "bench.py", "--gpus", "1",
         "--steps", "20", "--warmup", "5"] + extra,
        capture_output=True, text=True, timeout=timeout,
        env={**os.environ, "PCNN_DIST_BACKEND": "nccl"})
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    return json.loads(line)


def test_nccl_ws1_bench(device):
    r = _torchrun_ws1_nccl([])
    assert r["value"] > 0 and r["config"]["backend"] == "hip"


def test_nccl_ws1_overlap_comm(device):
    """overlap_comm over a real RCCL group: async all-reduce on the comm
    stream + Work.wait ordering before the update."""
    r = _torchrun_ws1_nccl(["--overlap-comm"])
    assert r["value"] > 0


def test_nccl_ws1_graph_capture(device):
    """hipGraph capture with the RCCL all-reduce inside the graph body."""
    r = _torchrun_ws1_nccl(["--use-graph"])
    assert r["value"] > 0 and r["config"]["hipgraph"] is True


def test_nccl_ws1_deepcnn(device):
    r = _torchrun_ws1_nccl(["--model", "deepcnn", "--batch-size", "64"])
    assert r["value"] > 0


def test_scale_check_preflight(device):
    """tools/scale_check.py must report READY on a GPU box (nccl group,
    bucket all-reduces verified)."""
    out = subprocess.run(
        [sys.executable, "tools/scale_check.py"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "READY" in out.stdout
    assert "backend=nccl" in out.stdout
    assert "FAIL" not in out.stdout


def test_native_cli_mnist_idx(device, tmp_path):
    """Native CLI end-to-end off real IDX files on disk (the reference's
    loaddata path, Sequential/Main.cpp:36-42, in the no-Python driver)."""
    import os
    import struct
    import numpy as np
    bin_path = os.path.join(os.path.dirname(__file__), "..", "tools",
                            "pcnn_train")
    if not os.path.exists(bin_path):
        pytest.skip("native CLI not built")
    rng = np.random.default_rng(7)
    d = tmp_path / "data"
    d.mkdir()
    for iname, n, lname in [("train-images.idx3-ubyte", 256,
                             "train-labels.idx1-ubyte"),
                            ("t10k-images.idx3-ubyte", 128,
                             "t10k-labels.idx1-ubyte")]:
        imgs = rng.integers(0, 256, size=(n, 28, 28), dtype=np.uint8)
        lbls = rng.integers(0, 10, size=n, dtype=np.uint8)
        with open(d / iname, "wb") as f:
            f.write(struct.pack(">iiii", 2051, n, 28, 28))
            f.write(imgs.tobytes())
        with open(d / lname, "wb") as f:
            f.write(struct.pack(">ii", 2049, n))
            f.write(lbls.tobytes())
    out = subprocess.run(
        [bin_path, "--data", "mnist", "--data-dir", str(d),
         "--epochs", "1", "--batch-size", "32"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "Learning" in out.stdout and "Error Rate: " in out.stdout


def test_device_prefetcher_batches_match_slices(device):
    """Regression for the round-1 prefetcher race (pinned staging buffer
    overwritten while its H2D copy was in flight): every prefetched
    batch must equal the corresponding host slice exactly."""
    from parallel_cnn_amd.data.pipeline import DevicePrefetcher
    torch.manual_seed(5)
    N, B, P = 64 * 12, 64, S.IN_PIX
    x = torch.rand(N, P)
    labels = torch.randint(0, 10, (N,))
    pf = DevicePrefetcher(x, labels, B, torch.device("cuda"),
                          torch.bfloat16)
    seen = 0
    for i, (xb, lb) in enumerate(pf):
        # force the host loop far ahead of the device to stress the
        # in-flight-overwrite window the race fix guards
        want_x = x[i * B:(i + 1) * B].to(torch.bfloat16)
        want_l = labels[i * B:(i + 1) * B].to(torch.int32)
        got_x = xb.cpu()
        got_l = lb.cpu()
        assert torch.equal(got_x, want_x), f"batch {i} torn"
        assert torch.equal(got_l, want_l), f"labels {i} torn"
        seen += 1
    assert seen == N // B
