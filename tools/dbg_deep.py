import sys; sys.path.insert(0, ".")
import torch
from parallel_cnn_amd.config import TrainConfig
from parallel_cnn_amd.data.mnist import synthetic_images
from parallel_cnn_amd.engine.deep import DeepTrainer
from parallel_cnn_amd.ops import native

B = 64
cfg = TrainConfig(batch_size=B, device="cuda", backend="hip", act_dtype="bf16", log_interval=0)
t = DeepTrainer(cfg)
x, labels = synthetic_images(B, 32, 32, 3, seed=1, structured=False)
xb, lb = t.stage_batch(x, labels)
m, w, spec = t.model, t.ws, t.model.spec
C = t._C
sh = native.current_stream_handle()

def ck(name):
    torch.cuda.synchronize()
    print("OK:", name, flush=True)

src = xb
for i, st in enumerate(spec.stages):
    C.deep_im2col(src, w.cols[i], B, st.h, st.w, st.cin, st.k, st.pad, st.kcp, sh); ck(f"im2col{i}")
    M = B * st.h * st.w
    C.deep_gemm(w.cols[i], m.view(f"conv{i}_w"), m.view(f"conv{i}_b"), w.acts[i], M, st.kcp, st.cout, st.kcp, st.cout, 1, 1, sh); ck(f"gemm{i}")
    C.deep_pool_fwd(w.acts[i], m.view(f"pool{i}_w"), w.pouts[i], B, st.h, st.w, st.cout, st.pool_k, sh); ck(f"pool{i}")
    src = w.pouts[i]
C.deep_fc_fwd(w.pouts[-1], m.view("fc_w"), m.view("fc_b"), lb, w.y, w.dz, w.loss_accum, w.correct_accum, B, spec.fc_in, spec.n_classes, 0, sh); ck("fc_fwd")
C.deep_fc_bwd(w.dz, w.pouts[-1], m.view("fc_w"), w.dppre[-1], B, spec.fc_in, spec.n_classes, sh); ck("fc_bwd")
fs = max(1, min(32, B // 64))
C.deep_fc_wgrad(w.dz, w.pouts[-1], m.grad_view("fc_w"), m.grad_view("fc_b"), B, spec.fc_in, spec.n_classes, fs, sh); ck("fc_wgrad")
for i in range(len(spec.stages) - 1, -1, -1):
    st = spec.stages[i]
    M = B * st.h * st.w
    G = max(8, min(256, (B * st.oh * st.ow * st.cout) // (256 * 32)))
    C.deep_pool_wgrad(w.dppre[i], w.acts[i], m.grad_view(f"pool{i}_w"), B, st.h, st.w, st.cout, st.pool_k, G, sh); ck(f"pool_wgrad{i}")
    C.deep_pool_bwd(w.dppre[i], w.acts[i], m.view(f"pool{i}_w"), w.acts[i], B, st.h, st.w, st.cout, st.pool_k, sh); ck(f"pool_bwd{i}")
    ktiles = (st.kcp + 63) // 64
    ntiles = (st.cout + 63) // 64
    ms = max(1, min(64, 256 // (ktiles * ntiles)))
    C.deep_wgrad_gemm(w.cols[i], w.acts[i], m.grad_view(f"conv{i}_w"), M, st.kcp, st.cout, ms, sh); ck(f"wgrad_gemm{i}")
    C.deep_colsum(w.acts[i], m.grad_view(f"conv{i}_b"), M, st.cout, 64, sh); ck(f"colsum{i}")
    if i > 0:
        C.deep_gemm(w.acts[i], m.view(f"conv{i}_w"), torch.empty(0), w.cols[i], M, st.cout, st.kcp, st.cout, st.kcp, 0, 0, sh); ck(f"dgrad{i}")
        C.deep_col2im_sigbwd(w.cols[i], w.pouts[i-1], w.dppre[i-1], B, st.h, st.w, st.cin, st.k, st.pad, st.kcp, sh); ck(f"col2im{i}")
C.deep_update(m.params, m.grads, 0.1/B, sh); ck("update")
print("ALL OK")
