"""GPU tests for the DeepCNN im2col + MFMA GEMM path."""
import json
import subprocess
import sys

import pytest
import torch

from parallel_cnn_amd import _C
from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_images
from parallel_cnn_amd.engine.deep import DeepTrainer
from parallel_cnn_amd.models.deepcnn import DeepCNN, DeepCNNSpec
from parallel_cnn_amd.ops import deep_ref, native

pytestmark = pytest.mark.gpu


def bf16_round(t):
    return t.to(torch.bfloat16).float()


def test_mfma_fragment_layout(device):
    """16x32 @ 32x16 with asymmetric operands — catches any A/B/C layout
    transposition (guide rule: never validate MFMA with symmetric data)."""
    torch.manual_seed(0)
    A = torch.randn(16, 32)
    Bm = torch.randn(32, 16)
    D = torch.zeros(16, 16)
    Ad, Bd, Dd = A.to(device), Bm.to(device), D.to(device)
    _C.deep_mfma_selftest(Ad, Bd, Dd, native.current_stream_handle())
    torch.cuda.synchronize()
    want = bf16_round(A) @ bf16_round(Bm)
    diff = (Dd.cpu() - want).abs().max().item()
    assert diff < 1e-2, diff


@pytest.mark.parametrize("M,K,N,b_kxn,epi", [
    (128, 64, 64, 1, 0), (64, 96, 32, 1, 1), (200, 32, 64, 0, 0),
    (256, 800, 64, 1, 1),
])
def test_deep_gemm_vs_matmul(M, K, N, b_kxn, epi, device):
    torch.manual_seed(M + K + N)
    A = torch.randn(M, K)
    W = torch.randn(K, N) if b_kxn else torch.randn(N, K)
    bias = torch.randn(N)
    Ad = A.to(device, dtype=torch.bfloat16)
    Wd = W.to(device)
    bd = bias.to(device)
    Cd = torch.empty(M, N, dtype=torch.bfloat16, device=device)
    _C.deep_gemm(Ad, Wd, bd, Cd, M, K, N, K, N, b_kxn, epi,
                 native.current_stream_handle())
    torch.cuda.synchronize()
    Wl = W if b_kxn else W.T
    want = bf16_round(A) @ bf16_round(Wl)
    if epi == 1:
        want = torch.sigmoid(want + bias)
    diff = (Cd.float().cpu() - want).abs().max().item()
    scale = want.abs().max().item()
    assert diff < 2e-2 * max(1.0, scale), (diff, scale)


def test_deep_im2col_matches_ref(device):
    spec = DeepCNNSpec()
    st = spec.stages[0]
    B = 2
    torch.manual_seed(1)
    x = torch.rand(B, st.h, st.w, st.cin)
    xd = x.reshape(B, -1).to(device, dtype=torch.bfloat16)
    cols = torch.empty(B * st.h * st.w, st.kcp, dtype=torch.bfloat16,
                       device=device)
    _C.deep_im2col(xd, cols, B, st.h, st.w, st.cin, st.k, st.pad, st.kcp,
                   native.current_stream_handle())
    torch.cuda.synchronize()
    want = deep_ref.im2col_ref(x.permute(0, 3, 1, 2).contiguous(), st)
    diff = (cols.float().cpu() - bf16_round(want)).abs().max().item()
    assert diff < 1e-2, diff


def hip_step_pieces(B, act_dtype, device, seed=7, implicit=True):
    cfg = TrainConfig(batch_size=B, device="cuda", backend="hip",
                      act_dtype=act_dtype, log_interval=0,
                      deep_implicit=implicit)
    t = DeepTrainer(cfg)
    x, labels = synthetic_images(B, 32, 32, 3, seed=seed, structured=False)
    xb, lb = t.stage_batch(x, labels)
    t.step(xb, lb)
    torch.cuda.synchronize()
    return t, x, labels


@pytest.mark.parametrize("implicit", [True, False])
def test_hip_step_matches_oracle_fp32(implicit, device):
    """Full fp32-activation training step vs the torchref oracle: the whole
    im2col/GEMM/pool/fc forward+backward+update chain — both the implicit
    (no cols buffer, dgrad-as-conv) and the materialized path."""
    B = 8
    t, x, labels = hip_step_pieces(B, "fp32", device, implicit=implicit)
    ref = DeepCNN(seed=t.cfg.seed)
    xh = x.view(B, 32, 32, 3)
    acts, pouts, y = deep_ref.forward(xh, ref)
    grads, loss = deep_ref.backward(xh, ref, acts, pouts, y, labels)
    with torch.no_grad():
        ref.params += t.cfg.dt * (1.0 / B) * grads
    diff = (t.model.params.cpu() - ref.params).abs()
    rel = diff.max().item()
    assert rel < 5e-3, rel
    lg, n = t.consume_loss()
    assert n == B
    assert abs(lg - loss) < 1e-2 * max(1.0, loss)


def test_hip_step_matches_oracle_bf16(device):
    B = 16
    t, x, labels = hip_step_pieces(B, "bf16", device, seed=11)
    ref = DeepCNN(seed=t.cfg.seed)
    xh = x.view(B, 32, 32, 3)
    acts, pouts, y = deep_ref.forward(xh, ref)
    grads, loss = deep_ref.backward(xh, ref, acts, pouts, y, labels)
    with torch.no_grad():
        ref.params += t.cfg.dt * (1.0 / B) * grads
    # bf16 activation storage: updates are small (dt*mean-grad), so compare
    # the DELTAS with a loose relative tolerance
    delta_hip = t.model.params.cpu() - DeepCNN(seed=t.cfg.seed).params
    delta_ref = ref.params - DeepCNN(seed=t.cfg.seed).params
    diff = (delta_hip - delta_ref).abs().max().item()
    scale = delta_ref.abs().max().item()
    assert diff < 0.1 * max(1e-3, scale), (diff, scale)
    lg, n = t.consume_loss()
    assert abs(lg - loss) < 5e-2 * max(1.0, loss)


def test_hip_step_fp16_close(device):
    B = 8
    t, x, labels = hip_step_pieces(B, "fp16", device, seed=13)
    ref = DeepCNN(seed=t.cfg.seed)
    xh = x.view(B, 32, 32, 3)
    acts, pouts, y = deep_ref.forward(xh, ref)
    grads, _ = deep_ref.backward(xh, ref, acts, pouts, y, labels)
    with torch.no_grad():
        ref.params += t.cfg.dt * (1.0 / B) * grads
    delta_hip = t.model.params.cpu() - DeepCNN(seed=t.cfg.seed).params
    delta_ref = ref.params - DeepCNN(seed=t.cfg.seed).params
    diff = (delta_hip - delta_ref).abs().max().item()
    scale = delta_ref.abs().max().item()
    assert diff < 0.1 * max(1e-3, scale), (diff, scale)


def test_hip_eval(device):
    B = 32
    cfg = TrainConfig(batch_size=B, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0)
    t = DeepTrainer(cfg)
    x, labels = synthetic_images(B, 32, 32, 3, seed=3, structured=False)
    ref = DeepCNN(seed=cfg.seed)
    _, _, y = deep_ref.forward(x.view(B, 32, 32, 3), ref)
    want_err = 100.0 * (1.0 - (y.argmax(1) == labels).float().mean().item())
    got_err = t.evaluate(x, labels)
    assert abs(got_err - want_err) < 1e-6, (got_err, want_err)


def test_bench_deepcnn_contract(device):
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "deepcnn", "--steps", "10",
         "--warmup", "2", "--batch-size", "256"],
        capture_output=True, text=True, timeout=900, check=True)
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    r = json.loads(line)
    assert r["value"] > 0
    assert "DeepCNN" in r["config"]["model"]
    assert r["config"]["global_batch"] == 256


def test_deep_graph_matches_eager(device):
    """hipGraph-captured deep step == eager step trajectory."""
    B = 16
    cfg = TrainConfig(batch_size=B, device="cuda", backend="hip",
                      act_dtype="bf16", log_interval=0)
    x, labels = synthetic_images(B * 3, 32, 32, 3, seed=21, structured=False)
    te = DeepTrainer(cfg)
    tg = DeepTrainer(cfg)
    tg.enable_graph()
    assert torch.allclose(te.model.params, tg.model.params)
    for s in range(3):
        xb, lb = te.stage_batch(x[s * B:(s + 1) * B],
                                labels[s * B:(s + 1) * B])
        te.step(xb, lb)
        xg, lg = tg.stage_batch(x[s * B:(s + 1) * B],
                                labels[s * B:(s + 1) * B])
        tg.step_graph(xg, lg)
    torch.cuda.synchronize()
    diff = (te.model.params - tg.model.params).abs().max().item()
    assert diff < 1e-5, diff
    le, ne = te.consume_loss()
    lg_, ng = tg.consume_loss()
    assert ne == ng == 3 * B
    assert abs(le - lg_) < 1e-3 * max(1.0, abs(le))


def test_implicit_matches_materialized(device):
    """The implicit-im2col step and the round-1 materialized-cols step are
    the same math: one bf16 step from the same init must land within
    rounding-grouping tolerance (the implicit dgrad skips the bf16 dcols
    rounding, so results are close, not bit-equal)."""
    B = 32
    ti, x, labels = hip_step_pieces(B, "bf16", device, seed=31,
                                    implicit=True)
    tm, _, _ = hip_step_pieces(B, "bf16", device, seed=31, implicit=False)
    base = DeepCNN(seed=ti.cfg.seed).params
    di = ti.model.params.cpu() - base
    dm = tm.model.params.cpu() - base
    diff = (di - dm).abs().max().item()
    scale = dm.abs().max().item()
    assert diff < 3e-2 * max(1e-3, scale), (diff, scale)


def test_implicit_infer_matches_materialized(device):
    """Forward/eval parity between the two engine paths (logits route)."""
    cfg_i = TrainConfig(batch_size=16, device="cuda", backend="hip",
                        act_dtype="bf16", log_interval=0)
    cfg_m = TrainConfig(batch_size=16, device="cuda", backend="hip",
                        act_dtype="bf16", log_interval=0,
                        deep_implicit=False)
    x, _ = synthetic_images(16, 32, 32, 3, seed=41, structured=False)
    yi = DeepTrainer(cfg_i).forward_logits(x)
    ym = DeepTrainer(cfg_m).forward_logits(x)
    diff = (yi - ym).abs().max().item()
    assert diff < 2e-2, diff


def test_deep_serve_oversized_request_gpu(device):
    """ADVICE r1 regression on the REAL backend: a /predict with
    B > cfg.batch_size used to write out of bounds on the hip path.
    The chunked forward_logits must match per-chunk results on GPU."""
    cfg = TrainConfig(batch_size=4, device="cuda", backend="hip",
                      act_dtype="bf16", log_interval=0)
    t = DeepTrainer(cfg)
    x, _ = synthetic_images(11, 32, 32, 3, seed=5)   # 11 > 4, ragged tail
    big = t.forward_logits(x)
    small = torch.cat([t.forward_logits(x[i:i + 4]) for i in range(0, 11, 4)])
    assert big.shape == (11, 10)
    assert torch.allclose(big, small, atol=1e-4), \
        (big - small).abs().max().item()


def test_deep_training_converges_gpu(device):
    """End-to-end: hip DeepCNN training reduces the per-epoch error norm
    on the structured set (same calibration as the torchref test: a 3x
    sigmoid CNN moves slowly — the deep family's benchmark is
    throughput; LeNet covers accuracy)."""
    cfg = TrainConfig(batch_size=32, device="cuda", backend="hip",
                      act_dtype="bf16", log_interval=0, model="deepcnn")
    t = DeepTrainer(cfg)
    x, y = synthetic_images(512, 32, 32, 3, seed=1)  # structured
    losses = []
    for ep in range(3):
        losses.append(t.train_epoch(x, y, log=lambda *a: None))
    torch.cuda.synchronize()
    assert losses[-1] < losses[0] * 0.99, losses


def test_weight_cache_fresh_after_checkpoint_load(device, tmp_path):
    """The fused update+cast keeps bf16 weight images fresh across steps;
    a checkpoint load mutates params OUTSIDE the engine and must
    invalidate them — a stale cache would silently serve the old
    weights' predictions."""
    from parallel_cnn_amd.utils.checkpoint import (load_checkpoint,
                                                   save_checkpoint)
    cfg = TrainConfig(batch_size=8, device="cuda", backend="hip",
                      act_dtype="bf16", log_interval=0)
    t = DeepTrainer(cfg)
    x, y = synthetic_images(8, 32, 32, 3, seed=17)
    t.step(*t.stage_batch(x, y))          # update+cast ran; cache fresh
    torch.cuda.synchronize()
    want = t.forward_logits(x)
    path = str(tmp_path / "w.bin")
    save_checkpoint(t, path)

    t2 = DeepTrainer(TrainConfig(batch_size=8, device="cuda",
                                 backend="hip", act_dtype="bf16",
                                 log_interval=0, seed=99))
    t2.step(*t2.stage_batch(x, y))        # t2's cache is fresh for ITS params
    load_checkpoint(t2, path)             # must invalidate
    got = t2.forward_logits(x)
    assert torch.allclose(got, want, atol=1e-3), \
        (got - want).abs().max().item()


def test_custom_channel_widths_gpu(device):
    """Non-power-of-two widths (48 channels: N%16==0 but the split-epi
    window alignment fails -> must run unsplit, not crash) train on the
    hip path and match the oracle."""
    cfg = TrainConfig(batch_size=8, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0,
                      deep_channels="16,32,48")
    t = DeepTrainer(cfg)
    x, labels = synthetic_images(8, 32, 32, 3, seed=23, structured=False)
    xb, lb = t.stage_batch(x, labels)
    t.step(xb, lb)
    torch.cuda.synchronize()
    from parallel_cnn_amd.models.deepcnn import DeepCNNSpec
    ref = DeepCNN(seed=cfg.seed, spec=DeepCNNSpec(channels=(16, 32, 48)))
    xh = x.view(8, 32, 32, 3)
    acts, pouts, y = deep_ref.forward(xh, ref)
    grads, loss = deep_ref.backward(xh, ref, acts, pouts, y, labels)
    with torch.no_grad():
        ref.params += t.cfg.dt * (1.0 / 8) * grads
    diff = (t.model.params.cpu() - ref.params).abs().max().item()
    assert diff < 5e-3, diff


def test_four_stage_hip_matches_oracle(device):
    """4-stage family on the hip path (last stage h*w = 16 < 64: the
    fused-pool tile alignment fails there and must fall back cleanly)."""
    cfg = TrainConfig(batch_size=8, device="cuda", backend="hip",
                      act_dtype="fp32", log_interval=0,
                      deep_channels="16,32,32,64")
    t = DeepTrainer(cfg)
    x, labels = synthetic_images(8, 32, 32, 3, seed=29, structured=False)
    t.step(*t.stage_batch(x, labels))
    torch.cuda.synchronize()
    from parallel_cnn_amd.models.deepcnn import DeepCNNSpec
    ref = DeepCNN(seed=cfg.seed,
                  spec=DeepCNNSpec(channels=(16, 32, 32, 64)))
    xh = x.view(8, 32, 32, 3)
    acts, pouts, y = deep_ref.forward(xh, ref)
    grads, _ = deep_ref.backward(xh, ref, acts, pouts, y, labels)
    with torch.no_grad():
        ref.params += t.cfg.dt * (1.0 / 8) * grads
    diff = (t.model.params.cpu() - ref.params).abs().max().item()
    assert diff < 5e-3, diff


def test_async_wgrad_mode_matches_multi(device):
    """PCNN_DEEP_WGRAD_MODE=async (side-stream weight grads) must produce
    the same trajectory as the default batched mode (subprocess so the
    env is read at workspace construction)."""
    import os
    code = (
        "import torch;"
        "from parallel_cnn_amd.config import TrainConfig;"
        "from parallel_cnn_amd.engine.deep import DeepTrainer;"
        "from parallel_cnn_amd.data.mnist import synthetic_images;"
        "cfg = TrainConfig(batch_size=16, device='cuda', backend='hip',"
        "act_dtype='fp32', log_interval=0);"
        "t = DeepTrainer(cfg);"
        "x, y = synthetic_images(32, 32, 32, 3, seed=43);"
        "t.step(*t.stage_batch(x[:16], y[:16]));"
        "t.step(*t.stage_batch(x[16:], y[16:]));"
        "torch.cuda.synchronize();"
        "print(float(t.model.params.double().abs().sum()),"
        "float(t.model.params[:100].double().sum()))"
    )
    outs = {}
    for mode in ("multi", "async"):
        r = subprocess.run([sys.executable, "-c", code],
                           capture_output=True, text=True, timeout=300,
                           env={**os.environ,
                                "PCNN_DEEP_WGRAD_MODE": mode})
        assert r.returncode == 0, r.stdout + r.stderr
        outs[mode] = [float(v) for v in r.stdout.split()]
    for a, b in zip(outs["multi"], outs["async"]):
        assert abs(a - b) < 1e-3 * max(1.0, abs(a)), outs


def test_fuzz_corner_configs(device):
    """Four fixed corner configs through tools/fuzz_deep (random channel
    sets, tiny batches, both engine modes) — the broader sweep lives in
    tools/, this keeps a slice of it in the suite."""
    out = subprocess.run(
        [sys.executable, "tools/fuzz_deep.py", "--n", "4", "--seed", "11"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "4/4 configs passed" in out.stdout
