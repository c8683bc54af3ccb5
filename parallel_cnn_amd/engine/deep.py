"""Training/eval driver for the DeepCNN family (im2col + MFMA GEMM path).

Backends: "hip" (gfx950 kernels, csrc/hip/conv_kernels.hip) and "torchref"
(the fp32 oracle, also the CPU execution path for this family).

Step sequence (hip, deep_implicit=True — the default), 17 launches:
per stage {implicit-im2col GEMM + bias + sigmoid + fused trainable-pool
epilogue (stage 0 with Cin=3 materializes cols; under-filled grids
split-K with a deterministic combine)}; fc fwd (+ residual loss + fused
fc backward-data); backward {fc wgrad; the dependency chain of fused
pool wgrad+bwd (dapre in place) and implicit dgrad-as-conv GEMMs
(rotated weight image, sigmoid-bwd epilogue, writing dppre[i-1]
directly); then EVERY stage's conv wgrad GEMM in ONE k_wgrad_multi
launch with the bias colsums folded in}; one fused DP all-reduce of the
flat gradient bucket; SGD update fused with the next step's weight cast.
deep_implicit=False keeps the round-1 materialized-cols path.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..config import TrainConfig
from ..models.deepcnn import DeepCNN
from ..ops import deep_ref, native
from ..parallel import dist as pdist

MODE_TRAIN, MODE_EVAL, MODE_INFER = 0, 1, 2


class DeepWorkspace:
    @staticmethod
    def wgrad_ms(st, M: int) -> int:
        """M-slice count for the weight-grad GEMM.  Round 1 measured ~512
        total WGs optimal; after the db fusion + hoisted implicit decode
        the per-WG fixed cost shrank, and under the BATCHED k_wgrad_multi
        launch the optimum is batch-dependent: ~256 per-stage target WGs
        at small batches (341k vs 332k at bs=64 — the combined grid
        already fills), ~512 at bs>=256 (667k vs 617k).
        PCNN_WGRAD_WGS overrides for sweeps."""
        import os
        B = max(1, M // (st.h * st.w))
        default = "256" if B <= 128 else "512"
        target = int(os.environ.get("PCNN_WGRAD_WGS", default))
        ktiles = (st.kcp + 63) // 64
        ntiles = (st.cout + 63) // 64
        return max(1, min(256, target // (ktiles * ntiles), M // 64))

    def __init__(self, model: DeepCNN, max_batch: int, device, act_dtype,
                 implicit: bool = True):
        spec = model.spec
        B = max_batch
        self.max_batch = B
        self.act_dtype = act_dtype
        # a stage can run the implicit-im2col GEMM fast path when its input
        # channel count is a multiple of 8 (16B spans inside one pixel)
        self.stage_implicit = [implicit and st.cin % 8 == 0
                               for st in spec.stages]
        # Cin < 8 stages (the 3-channel input) CAN run implicit too,
        # against a zero-padded 8-channel copy of their input
        # (k_pad_channels) and an 8-padded weight image, grads remapped by
        # k_remap_dw8.  Kills the im2col but grows stage-0 MFMA work
        # K=96 -> 200; measured net-NEGATIVE at bs=256 (re-confirming the
        # round-1 evaluation), so default OFF — PCNN_DEEP_PAD8=1 enables.
        import os
        pad8_on = os.environ.get("PCNN_DEEP_PAD8", "0") == "1"
        fusepool_on = os.environ.get("PCNN_DEEP_FUSEPOOL", "1") == "1"
        self.stage_pad8 = [implicit and pad8_on and 0 < st.cin < 8
                           for st in spec.stages]
        # pool-forward fuses into the conv GEMM epilogue when every pool
        # window lies inside one 64-row M-tile (true for all standard
        # shapes: W in {8,16,32}, pool 2x2) and the stage fits one n-tile
        self.stage_fusepool = [
            implicit and fusepool_on and st.cout <= 64 and st.w <= 64
            and 64 % st.w == 0
            and (64 // st.w) % st.pool_k == 0 and st.h % st.pool_k == 0
            and (st.h * st.w) % 64 == 0
            for st in spec.stages]
        self.cols = []    # [M, KcP] per materialized stage (None when the
                          # stage runs implicit — no cols buffer exists)
        self.acts = []    # [B*H*W, Cout] per stage (conv act == NHWC)
        self.pouts = []   # [B*OH*OW, Cout] per stage
        self.dppre = []   # pool preact grads, same shape as pouts
        self.x8 = [None] * len(spec.stages)   # padded inputs (pad8 stages)
        self.dw8 = [None] * len(spec.stages)  # 8-padded wgrad scratch
        for i, st in enumerate(spec.stages):
            M = B * st.h * st.w
            self.cols.append(
                None if (self.stage_implicit[i] or self.stage_pad8[i]) else
                torch.empty(M, st.kcp, dtype=act_dtype, device=device))
            if self.stage_pad8[i]:
                self.x8[i] = torch.empty(M, 8, dtype=act_dtype,
                                         device=device)
                self.dw8[i] = torch.zeros(st.k * st.k * 8 * st.cout,
                                          dtype=torch.float32,
                                          device=device)
            self.acts.append(torch.empty(M, st.cout, dtype=act_dtype,
                                         device=device))
            mo = B * st.oh * st.ow
            self.pouts.append(torch.empty(mo, st.cout, dtype=act_dtype,
                                          device=device))
            self.dppre.append(torch.empty(mo, st.cout, dtype=act_dtype,
                                          device=device))
        # pre-cast bf16 weight images, refreshed once per step by ONE
        # k_cast_wt_all launch: per stage [KcP][Cout] (wbf, dgrad
        # fallback), [Cout][KcP] (wbfT, forward B), and [Cin][K*K*Cout]
        # (wrot, the rotated/channel-transposed image the implicit
        # dgrad-as-conv consumes).  All slices of one flat buffer.
        self.cast_desc = {"R": [], "C": [], "K": [], "Cin": [], "w_off": [],
                          "bf_off": [], "bfT_off": [], "rot_off": [],
                          "p8_off": []}
        off = 0
        self.wbf, self.wbfT, self.wrot, self.wp8 = [], [], [], []
        offs = []
        for i, st in enumerate(spec.stages):
            d = self.cast_desc
            d["R"].append(st.kcp)
            d["C"].append(st.cout)
            d["K"].append(st.k)
            d["Cin"].append(st.cin)
            d["w_off"].append(spec.offsets[f"conv{i}_w"][0])
            d["bf_off"].append(off)
            offs.append((off, st.kcp * st.cout))
            off += st.kcp * st.cout
            d["bfT_off"].append(off)
            offs.append((off, st.cout * st.kcp))
            off += st.cout * st.kcp
            d["rot_off"].append(off)
            offs.append((off, st.cin * st.k * st.k * st.cout))
            off += st.cin * st.k * st.k * st.cout
            if self.stage_pad8[i]:
                d["p8_off"].append(off)
                offs.append((off, st.cout * st.k * st.k * 8))
                off += st.cout * st.k * st.k * 8
            else:
                d["p8_off"].append(-1)
                offs.append((0, 0))
        self.wbuf = torch.zeros(off, dtype=torch.bfloat16, device=device)
        for i in range(len(spec.stages)):
            o0, n0 = offs[4 * i]
            o1, n1 = offs[4 * i + 1]
            o2, n2 = offs[4 * i + 2]
            o3, n3 = offs[4 * i + 3]
            self.wbf.append(self.wbuf[o0:o0 + n0])
            self.wbfT.append(self.wbuf[o1:o1 + n1])
            self.wrot.append(self.wbuf[o2:o2 + n2])
            self.wp8.append(self.wbuf[o3:o3 + n3] if n3 else None)
        # split-K slab scratch (fp32): sized by replaying the launcher's
        # split policy over every GEMM shape this batch size will launch
        def split_need(M, N, K):
            mn = ((M + 63) // 64) * ((N + 63) // 64)
            ktiles = (K + 63) // 64
            if mn >= 768 or ktiles <= 1 or N % 8:
                return 0
            ks = min(ktiles, 1024 // mn + 1)
            tpc = (ktiles + ks - 1) // ks
            return ((ktiles + tpc - 1) // tpc) * M * N

        need = 0
        for i, st in enumerate(spec.stages):
            M = B * st.h * st.w
            kfwd = st.k * st.k * 8 if self.stage_pad8[i] else st.kcp
            need = max(need, split_need(M, st.cout, kfwd))
            if i > 0:
                need = max(need, split_need(M, st.cin,
                                            st.k * st.k * st.cout))
        self.c32 = (torch.empty(need, dtype=torch.float32, device=device)
                    if need else torch.empty(0, device=device))
        self.y = torch.empty(B, spec.n_classes, dtype=torch.float32,
                             device=device)
        self.dz = torch.empty(B, spec.n_classes, dtype=torch.float32,
                              device=device)
        # colsum per-workgroup partials (up to 512 slices x widest stage)
        maxc = max(st.cout for st in spec.stages)
        self.colsum_part = torch.empty(512 * maxc, dtype=torch.float32,
                                       device=device)

        self.loss_accum = torch.zeros(1, dtype=torch.float32, device=device)
        self.correct_accum = torch.zeros(1, dtype=torch.int32, device=device)


class DeepTrainer:
    def __init__(self, cfg: TrainConfig, model: Optional[DeepCNN] = None,
                 ctx: Optional[pdist.DistContext] = None,
                 max_batch: Optional[int] = None):
        self.cfg = cfg
        self.ctx = ctx or pdist.DistContext()
        self.device = torch.device(cfg.resolved_device())
        backend = cfg.resolved_backend()
        if backend == "cpu":
            backend = "torchref"  # deep family's CPU path is the oracle
        self.backend = backend
        if backend == "hip":
            self._C = native.require()
        if model is None:
            from ..models.deepcnn import DeepCNNSpec
            channels = tuple(
                int(c) for c in str(cfg.deep_channels).split(",") if c)
            spec = DeepCNNSpec(channels=channels)
            model = DeepCNN(self.device, seed=cfg.seed, spec=spec)
        self.model = model
        act_map = {"bf16": torch.bfloat16, "fp16": torch.float16,
                   "fp32": torch.float32}
        self.act_dtype = (act_map[cfg.act_dtype] if backend == "hip"
                          else torch.float32)
        self.ws = DeepWorkspace(self.model, max_batch or cfg.batch_size,
                                self.device, self.act_dtype,
                                implicit=cfg.deep_implicit)
        self._loss_host = 0.0
        self._samples_seen = 0
        self.global_step = 0
        self.epoch = 0              # completed-epoch cursor (exact resume)
        # wbuf freshness: True when the bf16 weight images match params
        # (the fused update+cast keeps them fresh across training steps;
        # any external params mutation must invalidate)
        self._wbuf_fresh = False
        self.timers = None  # set by enable_profiling

    def enable_profiling(self) -> None:
        from ..utils.timers import PhaseTimers
        self.timers = PhaseTimers()

    # ------------------------------------------------------------------ util
    def _scale(self, B: int) -> float:
        if self.cfg.grad_reduction == "mean":
            return 1.0 / float(B * self.ctx.world_size *
                               self.cfg.grad_accum)
        return 1.0

    def stage_batch(self, x: torch.Tensor, labels: torch.Tensor):
        """x: host fp32 [B, H*W*Cin] (NHWC flat)."""
        if self.backend == "hip":
            return (x.to(self.device, dtype=self.act_dtype,
                         non_blocking=True),
                    labels.to(self.device, dtype=torch.int32,
                              non_blocking=True))
        return x.to(torch.float32), labels.to(torch.int64)

    # ------------------------------------------------------------- hip paths
    def invalidate_weight_cache(self) -> None:
        """Call after mutating model.params outside the engine (e.g.
        checkpoint load): the bf16 weight images are re-cast on the next
        forward."""
        self._wbuf_fresh = False

    def _hip_cast_weights(self, force: bool = False):
        # ONE launch casts every stage's weights into all three bf16
        # images (wbf / wbfT / wrot) — skipped entirely when the fused
        # update+cast already refreshed them at the end of the previous
        # step (the common training case)
        if self._wbuf_fresh and not force:
            return
        d = self.ws.cast_desc
        self._C.deep_cast_all(self.model.params, self.ws.wbuf, d["R"],
                              d["C"], d["K"], d["Cin"], d["w_off"],
                              d["bf_off"], d["bfT_off"], d["rot_off"],
                              d["p8_off"], native.current_stream_handle())
        self._wbuf_fresh = True

    def _hip_update(self, scale: float) -> None:
        """SGD update fused with the next step's weight cast."""
        d = self.ws.cast_desc
        self._C.deep_update_cast(self.model.params, self.model.grads,
                                 self.cfg.dt * scale, self.ws.wbuf, d["R"],
                                 d["C"], d["K"], d["Cin"], d["w_off"],
                                 d["bf_off"], d["bfT_off"], d["rot_off"],
                                 d["p8_off"],
                                 native.current_stream_handle())
        self._wbuf_fresh = True

    def _hip_forward(self, x: torch.Tensor, labels: torch.Tensor, B: int,
                     mode: int):
        m, w, spec = self.model, self.ws, self.model.spec
        st_h = native.current_stream_handle()
        self._hip_cast_weights()
        src = x
        for i, st in enumerate(spec.stages):
            # Implicit-im2col staging (clamped unconditional gather +
            # hoisted magic-div decode) for Cin % 8 == 0 stages: no cols
            # buffer is written or re-read.  Round 1's implicit attempt
            # lost because its guarded gather de-pipelined the K-loop;
            # see im2col8f in conv_kernels.hip.  Stage 0 (Cin=3)
            # materializes cols — also reused by its wgrad.
            implicit = w.stage_implicit[i]
            pad8 = w.stage_pad8[i]
            M = B * st.h * st.w
            if pad8:
                # zero-pad Cin -> 8 so this stage rides the implicit fast
                # path too (x8 is ~12x smaller than the cols it replaces)
                self._C.deep_pad_channels(src, w.x8[i], M, st.cin, st_h)
            elif not implicit:
                self._C.deep_im2col(src, w.cols[i], B, st.h, st.w, st.cin,
                                    st.k, st.pad, st.kcp, st_h)
            fuse_pool = w.stage_fusepool[i]
            if pad8:
                a_src, kdim, b_img, xc = (w.x8[i], st.k * st.k * 8,
                                          w.wp8[i], 8)
            elif implicit:
                a_src, kdim, b_img, xc = src, st.kcp, w.wbfT[i], st.cin
            else:
                a_src, kdim, b_img, xc = (w.cols[i], st.kcp, w.wbfT[i],
                                          st.cin)
            # epilogue 3: the trainable-pool forward is computed inside
            # the GEMM from the LDS-staged sigmoid tile (pw is the pool
            # kernel view; its scalar bias follows contiguously in the
            # flat param vector, same trick as deep_pool_fwd)
            self._C.deep_gemm(a_src, m.view(f"conv{i}_w"),
                              m.view(f"conv{i}_b"), w.acts[i], M, kdim,
                              st.cout, kdim, st.cout, 1,
                              3 if fuse_pool else 1, st_h,
                              b_img,
                              a_src if (implicit or pad8)
                              else torch.empty(0),
                              st.h, st.w, xc, st.k, st.pad,
                              pw=m.view(f"pool{i}_w") if fuse_pool
                              else torch.empty(0),
                              pout=w.pouts[i] if fuse_pool
                              else torch.empty(0),
                              PK=st.pool_k if fuse_pool else 0,
                              c32=w.c32)
            if not fuse_pool:
                self._C.deep_pool_fwd(w.acts[i], m.view(f"pool{i}_w"),
                                      w.pouts[i], B, st.h, st.w, st.cout,
                                      st.pool_k, st_h)
            src = w.pouts[i]
        # train mode fuses the fc backward-data (dflat -> dppre[-1]) into
        # the forward block — k_fc_bwd is not launched separately
        self._C.deep_fc_fwd(w.pouts[-1], m.view("fc_w"), m.view("fc_b"),
                            labels, w.y, w.dz, w.loss_accum,
                            w.correct_accum, B, spec.fc_in, spec.n_classes,
                            mode, st_h,
                            dflat=w.dppre[-1] if mode == MODE_TRAIN
                            else torch.empty(0))

    def _wgrad_side(self):
        """Lazy side stream + events for the async-wgrad mode: weight
        grads run on a second HIP stream, co-resident with the main
        stream's pool/dgrad chain (chip fill via concurrency AND the
        chain hides them).  PCNN_DEEP_WGRAD_MODE=multi keeps the
        single-launch batched mode instead."""
        if not hasattr(self, "_wside"):
            self._wside = torch.cuda.Stream(device=self.device)
            self._wev = [torch.cuda.Event() for _ in
                         range(len(self.model.spec.stages) + 1)]
            self._wdone = torch.cuda.Event()
        return self._wside

    def _launch_wgrad_stage(self, x: torch.Tensor, B: int, i: int,
                            st_h: int) -> None:
        """One stage's conv weight-grad GEMM (+fused bias colsum) on the
        given stream (the async-wgrad side stream)."""
        m, w, spec = self.model, self.ws, self.model.spec
        st = spec.stages[i]
        M = B * st.h * st.w
        ms = self.ws.wgrad_ms(st, M)
        x_in = x if i == 0 else w.pouts[i - 1]
        if w.stage_pad8[i]:
            self._C.deep_wgrad_gemm(w.x8[i], w.acts[i], w.dw8[i], M,
                                    st.k * st.k * 8, st.cout, ms, st_h,
                                    w.x8[i], st.h, st.w, 8, st.k, st.pad,
                                    db=m.grad_view(f"conv{i}_b"))
            self._C.deep_remap_dw8(w.dw8[i], m.grad_view(f"conv{i}_w"),
                                   st.k * st.k, st.cin, st.cout, st_h)
        elif w.stage_implicit[i]:
            self._C.deep_wgrad_gemm(x_in, w.acts[i],
                                    m.grad_view(f"conv{i}_w"), M, st.kcp,
                                    st.cout, ms, st_h, x_in, st.h, st.w,
                                    st.cin, st.k, st.pad,
                                    db=m.grad_view(f"conv{i}_b"))
        else:
            self._C.deep_wgrad_gemm(w.cols[i], w.acts[i],
                                    m.grad_view(f"conv{i}_w"), M, st.kcp,
                                    st.cout, ms, st_h,
                                    db=m.grad_view(f"conv{i}_b"))

    def _hip_backward(self, x: torch.Tensor, B: int):
        m, w, spec = self.model, self.ws, self.model.spec
        st_h = native.current_stream_handle()
        nstage = len(spec.stages)
        import os
        # measured: the ONE-launch batched mode beats the side-stream
        # async mode at every batch size (e.g. 323k vs 281k @ bs64 —
        # per-stage launches co-schedule worse than one big grid and the
        # event fork/join costs host time); async stays for A/B
        async_wg = (os.environ.get("PCNN_DEEP_WGRAD_MODE", "multi")
                    == "async")
        side = self._wgrad_side() if async_wg else None
        main_s = torch.cuda.current_stream() if async_wg else None
        if async_wg:
            # fc wgrad moves to the side stream: it only needs dz and the
            # last pool output (both final after the forward)
            self._wev[nstage].record(main_s)
            side.wait_event(self._wev[nstage])
        side_h = side.cuda_stream if async_wg else st_h
        # fc backward-data already produced by the fused forward (dflat)
        # batch-slice count sized to ~4 WGs/CU (the owner-per-(k,m) grid
        # alone is only ~40 WGs for the 10x1024 head)
        bps = (spec.n_classes * spec.fc_in + 255 + spec.n_classes) // 256 + 1
        fs = max(1, min(B, 1024 // bps))
        self._C.deep_fc_wgrad(w.dz, w.pouts[-1], m.grad_view("fc_w"),
                              m.grad_view("fc_b"), B, spec.fc_in,
                              spec.n_classes, fs, side_h)
        # pass 1 — the dependency CHAIN, descending: pool wgrad+bwd of
        # stage i (dapre in place over acts[i]), then dgrad into
        # dppre[i-1].  Conv wgrads move to pass 2: each only needs its
        # stage's dapre (stable once written) and its input activation,
        # so all of them batch into ONE launch afterwards.
        for i in range(nstage - 1, -1, -1):
            st = spec.stages[i]
            M = B * st.h * st.w
            # ~2 pooled items (of 8 channels) per thread: the fused
            # wgrad+bwd kernel writes K*K 16B stores per item, so deeper
            # per-thread loops no longer amortize anything
            G = max(64, min(512, (B * st.oh * st.ow * st.cout) // (256 * 16)))
            if st.cout % 8 == 0 and st.pool_k == 2:
                # fused: one pass computes the pool weight grads AND
                # writes dapre in place over the activation
                self._C.deep_pool_wbwd(w.dppre[i], w.acts[i],
                                       m.view(f"pool{i}_w"), w.acts[i],
                                       m.grad_view(f"pool{i}_w"), B, st.h,
                                       st.w, st.cout, st.pool_k, G, st_h)
            else:
                self._C.deep_pool_wgrad(w.dppre[i], w.acts[i],
                                        m.grad_view(f"pool{i}_w"), B, st.h,
                                        st.w, st.cout, st.pool_k, G, st_h)
                # pool bwd writes the conv preact grad IN PLACE over the
                # conv activation (elementwise same-index, safe)
                self._C.deep_pool_bwd(w.dppre[i], w.acts[i],
                                      m.view(f"pool{i}_w"), w.acts[i], B,
                                      st.h, st.w, st.cout, st.pool_k, st_h)
            dapre = w.acts[i]
            implicit = w.stage_implicit[i]
            if async_wg:
                # this stage's dapre is final: its weight grad can run on
                # the side stream, concurrent with the rest of the chain
                self._wev[i].record(main_s)
                side.wait_event(self._wev[i])
                self._launch_wgrad_stage(x, B, i, side.cuda_stream)
            if i > 0:
                if implicit:
                    # dgrad-as-conv: implicit im2col of dapre against the
                    # rotated/channel-transposed weight image, sigmoid-bwd
                    # fused in the epilogue — writes the previous stage's
                    # pool preact grad DIRECTLY (the round-1 path's
                    # dcols write + k_col2im_sigbwd re-read are gone).
                    kd = st.k * st.k * st.cout
                    self._C.deep_gemm(dapre, m.view(f"conv{i}_w"),
                                      torch.empty(0), w.dppre[i - 1], M,
                                      kd, st.cin, kd, st.cin, 0, 2, st_h,
                                      w.wrot[i], dapre, st.h, st.w,
                                      st.cout, st.k, st.pad,
                                      epi=w.pouts[i - 1], c32=w.c32)
                else:
                    # dgrad into the cols buffer (its forward use is done)
                    self._C.deep_gemm(dapre, m.view(f"conv{i}_w"),
                                      torch.empty(0), w.cols[i], M, st.cout,
                                      st.kcp, st.cout, st.kcp, 0, 0, st_h,
                                      w.wbf[i])
                    self._C.deep_col2im_sigbwd(w.cols[i], w.pouts[i - 1],
                                               w.dppre[i - 1], B, st.h,
                                               st.w, st.cin, st.k, st.pad,
                                               st.kcp, st_h)
        if async_wg:
            # all wgrads are queued on the side stream; rejoin before the
            # all-reduce / update touches the gradient bucket
            self._wdone.record(side)
            main_s.wait_event(self._wdone)
            return
        # pass 2 (multi mode) — every stage's conv wgrad GEMM in ONE
        # launch (the conv BIAS colsum stays folded in).  Three
        # sequential ~500-WG launches each ran at ~2 WGs/CU with the
        # per-iteration stall exposed; the combined grid fills the chip.
        da, dd, dwv, dbv = [], [], [], []
        Ml, Kl, Nl, MSl, Il, XHl, XWl, XCl, XKl, XPl = ([] for _ in range(10))
        for i, st in enumerate(spec.stages):
            M = B * st.h * st.w
            x_in = x if i == 0 else w.pouts[i - 1]
            if w.stage_pad8[i]:
                da.append(w.x8[i])
                dwv.append(w.dw8[i])
                Kl.append(st.k * st.k * 8)
                Il.append(1)
                XCl.append(8)
            elif w.stage_implicit[i]:
                da.append(x_in)
                dwv.append(m.grad_view(f"conv{i}_w"))
                Kl.append(st.kcp)
                Il.append(1)
                XCl.append(st.cin)
            else:
                da.append(w.cols[i])
                dwv.append(m.grad_view(f"conv{i}_w"))
                Kl.append(st.kcp)
                Il.append(0)
                XCl.append(st.cin)
            dd.append(w.acts[i])
            dbv.append(m.grad_view(f"conv{i}_b"))
            Ml.append(M)
            Nl.append(st.cout)
            MSl.append(self.ws.wgrad_ms(st, M))
            XHl.append(st.h)
            XWl.append(st.w)
            XKl.append(st.k)
            XPl.append(st.pad)
        self._C.deep_wgrad_multi(da, dd, dwv, dbv, Ml, Kl, Nl, MSl, Il,
                                 XHl, XWl, XCl, XKl, XPl, st_h)
        for i, st in enumerate(spec.stages):
            if w.stage_pad8[i]:
                self._C.deep_remap_dw8(w.dw8[i], m.grad_view(f"conv{i}_w"),
                                       st.k * st.k, st.cin, st.cout, st_h)

    # ----------------------------------------------------------------- graph
    def enable_graph(self) -> None:
        """Capture one full training step (~30 kernels) into a hipGraph.
        Measured perf-neutral at bs=256 (the host enqueue is already
        hidden; the 667-vs-616 us wall/kernel-sum gap is device-side
        dispatch between dependent kernels, which graphs do not remove) —
        provided for the small-batch regime where Python launch overhead
        binds, and as the capture-compatibility check for the step.
        Same shape as the LeNet trainer's capture: warmup on a side
        stream with the training state snapshotted/restored."""
        if self.backend != "hip":
            raise RuntimeError("graph capture requires the hip backend")
        if self.cfg.grad_accum != 1:
            raise RuntimeError("graph capture assumes grad_accum == 1")
        if pdist.is_distributed() and \
                torch.distributed.get_backend() != "nccl":
            raise RuntimeError("graph capture requires RCCL (nccl backend)")
        B = self.ws.max_batch
        spec = self.model.spec
        in_pix = spec.in_h * spec.in_w * spec.in_ch
        self._gx = torch.zeros(B, in_pix, dtype=self.act_dtype,
                               device=self.device)
        self._gl = torch.zeros(B, dtype=torch.int32, device=self.device)
        params0 = self.model.params.clone()
        grads0 = self.model.grads.clone()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                self._graph_body(B)
        torch.cuda.current_stream().wait_stream(side)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._graph_body(B)
        with torch.no_grad():
            self.model.params.copy_(params0)
            self.model.grads.copy_(grads0)
            self.ws.loss_accum.zero_()
            self.ws.correct_accum.zero_()
        # the captured graph contains NO cast (warmup left wbuf fresh at
        # capture); refresh it for the restored params before any replay
        self._hip_cast_weights(force=True)
        torch.cuda.synchronize()

    def _graph_body(self, B: int) -> None:
        self._hip_forward(self._gx, self._gl, B, MODE_TRAIN)
        self._hip_backward(self._gx, B)
        pdist.allreduce_grads(self.model.grads)
        self._hip_update(self._scale(B))

    def step_graph(self, x: torch.Tensor, labels: torch.Tensor) -> None:
        """Replay the captured step on a staged batch (two D2D copies)."""
        self._gx.copy_(x.view(self._gx.shape), non_blocking=True)
        self._gl.copy_(labels, non_blocking=True)
        self._graph.replay()
        self._samples_seen += x.shape[0] * self.ctx.world_size
        self.global_step += 1

    # ------------------------------------------------------------------ step
    def step(self, x: torch.Tensor, labels: torch.Tensor) -> None:
        """One micro/step.  The wgrad kernels ACCUMULATE into the flat
        gradient bucket (deep_update zeroes it), so cfg.grad_accum > 1
        simply defers the all-reduce + update to every grad_accum-th
        call — same contract as the LeNet Trainer."""
        B = x.shape[0]
        assert B <= self.ws.max_batch
        self._accum = getattr(self, "_accum", 0) + 1
        apply_update = self._accum >= self.cfg.grad_accum
        if apply_update:
            self._accum = 0
        scale = self._scale(B)
        if self.backend == "hip":
            if self.timers is not None:
                with self.timers.phase("forward"):
                    self._hip_forward(x, labels, B, MODE_TRAIN)
                with self.timers.phase("backward"):
                    self._hip_backward(x, B)
                if apply_update:
                    with self.timers.phase("all-reduce"):
                        pdist.allreduce_grads(self.model.grads)
                    with self.timers.phase("update"):
                        self._hip_update(scale)
                self._samples_seen += B * self.ctx.world_size
                self.global_step += 1
                return
            self._hip_forward(x, labels, B, MODE_TRAIN)
            self._hip_backward(x, B)
            if apply_update:
                pdist.allreduce_grads(self.model.grads)
                self._hip_update(scale)
        else:
            spec = self.model.spec
            xh = x.view(B, spec.in_h, spec.in_w, spec.in_ch)
            acts, pouts, y = deep_ref.forward(xh, self.model)
            grads, loss = deep_ref.backward(xh, self.model, acts, pouts, y,
                                            labels)
            self._loss_host += loss
            self.model.grads += grads
            if apply_update:
                pdist.allreduce_grads(self.model.grads)
                with torch.no_grad():
                    self.model.params += (self.cfg.dt * scale *
                                          self.model.grads)
                    self.model.grads.zero_()
        self._samples_seen += B * self.ctx.world_size
        self.global_step += 1

    def consume_loss(self) -> Tuple[float, int]:
        if self.backend == "hip":
            local = float(self.ws.loss_accum.item())
            self.ws.loss_accum.zero_()
        else:
            local = self._loss_host
            self._loss_host = 0.0
        dev = self.device if self.device.type == "cuda" else None
        total = pdist.allreduce_scalar(local, device=dev)
        n = self._samples_seen
        self._samples_seen = 0
        return total, n

    def train_epoch(self, x: torch.Tensor, labels: torch.Tensor,
                    log=print) -> float:
        """One epoch over a host dataset, DP-sharded; returns the mean
        per-sample error norm."""
        Bl = self.ws.max_batch
        Bg = Bl * self.ctx.world_size
        n = (x.shape[0] // Bg) * Bg
        total_loss, total_n = 0.0, 0
        for s in range(0, n, Bg):
            lo = s + self.ctx.rank * Bl
            self.step(*self.stage_batch(x[lo:lo + Bl], labels[lo:lo + Bl]))
            if self.cfg.log_interval and \
                    self.global_step % self.cfg.log_interval == 0:
                loss, cnt = self.consume_loss()
                total_loss += loss
                total_n += cnt
                if self.ctx.is_main and cnt:
                    log(f"step {self.global_step}: error {loss / cnt:e}")
        loss, cnt = self.consume_loss()
        total_loss += loss
        total_n += cnt
        self.epoch += 1
        return total_loss / max(1, total_n)

    @torch.no_grad()
    def forward_logits(self, x: torch.Tensor) -> torch.Tensor:
        """Forward a host fp32 batch [N, H*W*Cin] of ANY size, chunked by
        the workspace max batch (the deep twin of Trainer.classify's
        chunking — serving requests must never exceed the device
        workspace sizing, see serve.py).  Returns host fp32 logits
        [N, n_classes]."""
        w, spec = self.ws, self.model.spec
        outs = []
        bs = w.max_batch
        for i in range(0, x.shape[0], bs):
            nb = min(bs, x.shape[0] - i)
            xb, lb = self.stage_batch(
                x[i:i + nb], torch.zeros(nb, dtype=torch.int64))
            B = xb.shape[0]
            if self.backend == "hip":
                self._hip_forward(xb, lb, B, MODE_INFER)
                outs.append(w.y[:B].cpu().clone())
            else:
                _, _, y = deep_ref.forward(
                    xb.view(B, spec.in_h, spec.in_w, spec.in_ch), self.model)
                outs.append(y)
        return torch.cat(outs)

    @torch.no_grad()
    def classify(self, x: torch.Tensor) -> torch.Tensor:
        """Predicted labels for a host fp32 batch [N, H*W*Cin] (chunked)."""
        return self.forward_logits(x).argmax(dim=1)

    @torch.no_grad()
    def evaluate(self, x: torch.Tensor, labels: torch.Tensor,
                 batch_size: Optional[int] = None) -> float:
        bs = batch_size or self.ws.max_batch
        n = x.shape[0]
        correct = 0
        if self.backend == "hip":
            self.ws.correct_accum.zero_()
        per = (n + self.ctx.world_size - 1) // self.ctx.world_size
        lo, hi = self.ctx.rank * per, min(n, (self.ctx.rank + 1) * per)
        for i in range(lo, hi, bs):
            xb, lb = self.stage_batch(x[i:i + bs], labels[i:i + bs])
            B = xb.shape[0]
            if self.backend == "hip":
                self._hip_forward(xb, lb, B, MODE_EVAL)
            else:
                spec = self.model.spec
                _, _, y = deep_ref.forward(
                    xb.view(B, spec.in_h, spec.in_w, spec.in_ch), self.model)
                correct += int((y.argmax(1) == lb).sum().item())
        if self.backend == "hip":
            correct = int(self.ws.correct_accum.item())
            self.ws.correct_accum.zero_()
        dev = self.device if self.device.type == "cuda" else None
        correct = int(pdist.allreduce_scalar(float(correct), device=dev))
        return 100.0 * (1.0 - correct / float(n))
