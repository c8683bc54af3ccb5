"""Training/eval driver — the framework equivalent of the reference's
Main.cpp `learn()` / `test()` / `classify()`.

Backends:
  * "hip"      — the gfx950 HIP kernels (3 fused launches per step);
  * "cpu"      — the native C++ reference ops (threaded, race-free);
  * "torchref" — the pure-PyTorch fp32 oracle (tests/debugging).

A training step is: fused fwd+bwd-data -> weight-grad -> [RCCL all-reduce of
the flat gradient bucket] -> fused SGD update.  Loss (sum over samples of
||onehot - y||_2, the reference's metric) accumulates device-side and is
only synced on readout.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..config import TrainConfig
from ..models.lenet import LeNet5
from ..ops import native, torch_ref
from ..ops import shapes as S
from ..parallel import dist as pdist
from ..utils.timers import PhaseTimers

MODE_TRAIN, MODE_EVAL, MODE_INFER = 0, 1, 2


class Workspace:
    """Persistent activation / gradient-data buffers for a max batch size."""

    def __init__(self, max_batch: int, device: torch.device,
                 act_dtype: torch.dtype):
        B = max_batch
        dev = device
        self.max_batch = B
        self.act_dtype = act_dtype
        self.a1 = torch.empty(B, S.C1_OUT, dtype=act_dtype, device=dev)
        self.a2 = torch.empty(B, S.S1_OUT, dtype=act_dtype, device=dev)
        self.y = torch.empty(B, S.FC_OUT, dtype=torch.float32, device=dev)
        self.dz = torch.empty(B, S.FC_OUT, dtype=torch.float32, device=dev)
        self.dz2 = torch.empty(B, S.S1_OUT, dtype=torch.float32, device=dev)
        # dz1 is the one large backward tensor: stored in the activation
        # dtype (fp32 math stays inside the kernels)
        self.dz1 = torch.empty(B, S.C1_OUT, dtype=act_dtype, device=dev)
        self.loss_accum = torch.zeros(1, dtype=torch.float32, device=dev)
        self.correct_accum = torch.zeros(1, dtype=torch.int32, device=dev)


class Trainer:
    def __init__(self, cfg: TrainConfig, model: Optional[LeNet5] = None,
                 ctx: Optional[pdist.DistContext] = None,
                 max_batch: Optional[int] = None):
        self.cfg = cfg
        self.ctx = ctx or pdist.DistContext()
        self.device = torch.device(cfg.resolved_device())
        self.backend = cfg.resolved_backend()
        if self.backend == "hip" and self.device.type != "cuda":
            raise RuntimeError("hip backend requires a GPU device")
        if self.backend in ("hip", "cpu"):
            self._C = native.require()
        self.model = model or LeNet5(self.device, seed=cfg.seed)
        act_map = {"bf16": torch.bfloat16, "fp16": torch.float16,
                   "fp32": torch.float32}
        act_dtype = (act_map[cfg.act_dtype] if self.backend == "hip"
                     else torch.float32)
        self.act_dtype = act_dtype
        self.ws = Workspace(max_batch or cfg.batch_size, self.device,
                            act_dtype)
        self._loss_host = 0.0       # cpu/torchref backends accumulate here
        self._samples_seen = 0
        self.global_step = 0
        self.epoch = 0              # completed-epoch cursor (exact resume)
        # cached stream handle for the hot per-step path (the graph-capture
        # body resolves the live stream instead)
        self._sh = native.current_stream_handle() if self.backend == "hip" \
            else 0
        self.timers: Optional[PhaseTimers] = None  # set by enable_profiling
        self._pool_mode = 1 if cfg.pool == "max" else 0
        self._loss_mode = 1 if cfg.loss == "softmax_ce" else 0
        # fuse_wgrad: conv/pool grads accumulate inside fwdbwd and the
        # wgrad kernel covers only the fc role (experimental — measured
        # slower at bs=64: the LDS-atomic combine serializes); default:
        # full wgrad kernel, minus the pool role under max pooling
        self._fuse = 1 if cfg.fuse_wgrad else 0
        self._wroles = 4 if self._fuse else (5 if self._pool_mode == 1
                                             else 7)

    # ------------------------------------------------------------------ util
    def _update_scale(self, local_batch: int) -> float:
        if self.cfg.grad_reduction == "mean":
            return 1.0 / float(local_batch * self.ctx.world_size *
                               self.cfg.grad_accum)
        return 1.0

    def stage_batch(self, x: torch.Tensor, labels: torch.Tensor
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Move a host fp32 batch to the execution device/dtype."""
        if self.backend == "hip":
            xd = x.to(self.device, dtype=self.act_dtype, non_blocking=True)
            ld = labels.to(self.device, dtype=torch.int32, non_blocking=True)
        else:
            xd = x.to(torch.float32)
            ld = labels.to(torch.int64)
        return xd, ld

    def enable_profiling(self) -> None:
        """Per-phase sync-correct timers (the framework analog of the
        reference's per-layer clock() accumulators, but device-synced).
        Adds a device sync per phase — profiling mode only."""
        if self.cfg.grad_accum != 1:
            raise RuntimeError("profiling mode assumes grad_accum == 1")
        self.timers = PhaseTimers()

    # ------------------------------------------------------------------ step
    def step(self, x: torch.Tensor, labels: torch.Tensor) -> None:
        """One micro/step on an already-staged batch.  The weight-grad
        kernels ACCUMULATE into the flat bucket, so gradient accumulation
        (cfg.grad_accum > 1) simply defers the all-reduce + update to every
        grad_accum-th call."""
        if self.timers is not None:
            return self._step_profiled(x, labels)
        B = x.shape[0]
        assert B <= self.ws.max_batch
        self._accum = getattr(self, "_accum", 0) + 1
        apply_update = self._accum >= self.cfg.grad_accum
        if apply_update:
            self._accum = 0
        m, w = self.model, self.ws
        scale = self._update_scale(B)
        if self.backend == "hip":
            stream = self._sh
            self._C.hip_fwdbwd(x, m.params, w.a1, w.a2, w.y, w.dz, w.dz2,
                               w.dz1, labels, w.loss_accum, w.correct_accum,
                               B, MODE_TRAIN, stream, self._pool_mode,
                               self._loss_mode, m.grads, self._fuse)
            if not apply_update:
                self._C.hip_wgrad_roles(x, w.a1, w.a2, w.dz, w.dz2, w.dz1,
                                        m.grads, B, self.cfg.wgrad_chunk,
                                        self._wroles, stream)
            elif self.cfg.overlap_comm and pdist.is_distributed():
                # two-bucket overlap (SURVEY §5.8 / north star): the first
                # bucket all-reduces on the RCCL stream while the rest of
                # the wgrad still computes on the compute stream
                if self._fuse:
                    wk_head = pdist.allreduce_grads_async(
                        m.grads[:S.OFF_FW])
                    self._C.hip_wgrad_roles(x, w.a1, w.a2, w.dz, w.dz2,
                                            w.dz1, m.grads, B,
                                            self.cfg.wgrad_chunk, 4, stream)
                    wk_tail = pdist.allreduce_grads_async(
                        m.grads[S.OFF_FW:])
                else:
                    tail_roles = self._wroles & ~1
                    self._C.hip_wgrad_roles(x, w.a1, w.a2, w.dz, w.dz2,
                                            w.dz1, m.grads, B,
                                            self.cfg.wgrad_chunk,
                                            tail_roles, stream)
                    wk_tail = pdist.allreduce_grads_async(
                        m.grads[S.OFF_S1W:])
                    self._C.hip_wgrad_roles(x, w.a1, w.a2, w.dz, w.dz2,
                                            w.dz1, m.grads, B,
                                            self.cfg.wgrad_chunk, 1, stream)
                    wk_head = pdist.allreduce_grads_async(
                        m.grads[:S.OFF_S1W])
                if wk_head is not None:
                    wk_head.wait()
                if wk_tail is not None:
                    wk_tail.wait()
            else:
                self._C.hip_wgrad_roles(x, w.a1, w.a2, w.dz, w.dz2, w.dz1,
                                        m.grads, B, self.cfg.wgrad_chunk,
                                        self._wroles, stream)
                pdist.allreduce_grads(m.grads)
            if apply_update:
                self._C.hip_update(m.params, m.grads, self.cfg.dt * scale,
                                   stream)
        elif self.backend == "cpu":
            # CPU path keeps fp32 activations in the workspace directly.
            a1 = self._cpu_view(w.a1, B)
            a2 = self._cpu_view(w.a2, B)
            self._C.cpu_forward(x, m.params, a1, a2, w.y[:B],
                                self._pool_mode, self._loss_mode)
            loss = self._C.cpu_backward(x, m.params, a1, a2, w.y[:B], labels,
                                        w.dz[:B], w.dz2[:B], w.dz1[:B],
                                        m.grads, self._pool_mode,
                                        self._loss_mode)
            self._loss_host += loss
            if apply_update:
                pdist.allreduce_grads(m.grads)
                self._C.cpu_update(m.params, m.grads, self.cfg.dt, scale)
        else:  # torchref
            a1, a2, y = torch_ref.forward(x, m.params, self.cfg.pool,
                                          self.cfg.loss)
            dz, dz2, dz1, grads, loss = torch_ref.backward(
                x, m.params, a1, a2, y, labels, self.cfg.pool, self.cfg.loss)
            self._loss_host += loss
            m.grads += grads
            if apply_update:
                pdist.allreduce_grads(m.grads)
                torch_ref.update(m.params, m.grads, self.cfg.dt, scale)
        self._samples_seen += B * self.ctx.world_size
        self.global_step += 1

    def _step_profiled(self, x: torch.Tensor, labels: torch.Tensor) -> None:
        B = x.shape[0]
        m, w, t = self.model, self.ws, self.timers
        scale = self._update_scale(B)
        if self.backend != "hip":
            with t.phase("step"):
                self.timers = None
                try:
                    self.step(x, labels)
                finally:
                    self.timers = t
            return
        stream = self._sh
        with t.phase("fwd+bwd-data"):
            self._C.hip_fwdbwd(x, m.params, w.a1, w.a2, w.y, w.dz, w.dz2,
                               w.dz1, labels, w.loss_accum, w.correct_accum,
                               B, MODE_TRAIN, stream, self._pool_mode,
                               self._loss_mode, m.grads, self._fuse)
        with t.phase("weight-grad"):
            self._C.hip_wgrad_roles(x, w.a1, w.a2, w.dz, w.dz2, w.dz1,
                                    m.grads, B, self.cfg.wgrad_chunk,
                                    self._wroles, stream)
        with t.phase("all-reduce"):
            pdist.allreduce_grads(m.grads)
        with t.phase("update"):
            self._C.hip_update(m.params, m.grads, self.cfg.dt * scale,
                               stream)
        self._samples_seen += B * self.ctx.world_size
        self.global_step += 1

    # ------------------------------------------------------------ hipGraph
    def enable_graph(self) -> None:
        """Capture one full training step (3 kernels + the RCCL all-reduce
        when distributed) into a hipGraph; run_steps_pooled then replays it.
        Requires world_size==1 or the nccl/RCCL backend (gloo is host-side,
        not capturable)."""
        if self.backend != "hip":
            raise RuntimeError("graph capture requires the hip backend")
        if self.cfg.grad_accum != 1:
            raise RuntimeError("graph capture assumes grad_accum == 1")
        if pdist.is_distributed() and \
                torch.distributed.get_backend() != "nccl":
            raise RuntimeError("graph capture requires RCCL (nccl backend)")
        B = self.ws.max_batch
        self._gx = torch.zeros(B, S.IN_PIX, dtype=self.act_dtype,
                               device=self.device)
        self._gl = torch.zeros(B, dtype=torch.int32, device=self.device)
        # warmup on a side stream, then capture; the warmup replays REAL
        # steps, so snapshot and restore the training state around it
        params0 = self.model.params.clone()
        grads0 = self.model.grads.clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                self._graph_body(B)
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._graph_body(B)
        with torch.no_grad():
            self.model.params.copy_(params0)
            self.model.grads.copy_(grads0)
            self.ws.loss_accum.zero_()
        torch.cuda.synchronize()

    def _graph_body(self, B: int) -> None:
        m, w = self.model, self.ws
        stream = native.current_stream_handle()
        self._C.hip_fwdbwd(self._gx, m.params, w.a1, w.a2, w.y, w.dz, w.dz2,
                           w.dz1, self._gl, w.loss_accum, w.correct_accum,
                           B, MODE_TRAIN, stream, self._pool_mode,
                           self._loss_mode, m.grads, self._fuse)
        self._C.hip_wgrad_roles(self._gx, w.a1, w.a2, w.dz, w.dz2, w.dz1,
                                m.grads, B, self.cfg.wgrad_chunk,
                                self._wroles, stream)
        pdist.allreduce_grads(m.grads)
        self._C.hip_update(m.params, m.grads,
                           self.cfg.dt * self._update_scale(B),
                           native.current_stream_handle())

    def step_graph(self, x: torch.Tensor, labels: torch.Tensor) -> None:
        """Replay the captured step on a staged batch (one D2D copy in)."""
        self._gx.copy_(x, non_blocking=True)
        self._gl.copy_(labels, non_blocking=True)
        self._graph.replay()
        self._samples_seen += x.shape[0] * self.ctx.world_size
        self.global_step += 1

    def run_steps_pooled(self, x_pool: torch.Tensor,
                         labels_pool: torch.Tensor, steps: int) -> None:
        """Run `steps` training steps over a device-resident batch pool
        (pool rows = P*B, cycled).  Single-GPU non-distributed runs use the
        C++ fused enqueue loop (no per-step Python overhead); distributed
        runs keep the per-step loop with the gradient all-reduce."""
        B = self.ws.max_batch
        P = x_pool.shape[0] // B
        if getattr(self, "_graph", None) is not None:
            for s in range(steps):
                i = (s % P) * B
                self.step_graph(x_pool[i:i + B], labels_pool[i:i + B])
            return
        if self.backend == "hip" and not pdist.is_distributed():
            w = self.ws
            self._C.hip_train_steps(
                x_pool, labels_pool, self.model.params, self.model.grads,
                w.a1, w.a2, w.y, w.dz, w.dz2, w.dz1, w.loss_accum, B, steps,
                self.cfg.wgrad_chunk, self.cfg.dt * self._update_scale(B),
                native.current_stream_handle(), self._pool_mode,
                self._loss_mode, self._fuse)
            self._samples_seen += B * steps
            self.global_step += steps
        else:
            for s in range(steps):
                i = (s % P) * B
                self.step(x_pool[i:i + B], labels_pool[i:i + B])

    def _cpu_view(self, t: torch.Tensor, B: int) -> torch.Tensor:
        assert t.dtype == torch.float32, \
            "cpu backend requires fp32 activation workspace"
        return t[:B]

    # ------------------------------------------------------------ loss/eval
    def consume_loss(self) -> Tuple[float, int]:
        """Returns (sum of per-sample error norms across ranks, samples seen)
        since the last call.  Syncs the device."""
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

    @torch.no_grad()
    def evaluate(self, x: torch.Tensor, labels: torch.Tensor,
                 batch_size: Optional[int] = None) -> float:
        """Error rate (%) over a dataset, sharded across ranks."""
        bs = batch_size or self.ws.max_batch
        n = x.shape[0]
        correct = 0
        w = self.ws
        if self.backend == "hip":
            w.correct_accum.zero_()
        # contiguous rank shard
        per = (n + self.ctx.world_size - 1) // self.ctx.world_size
        lo, hi = self.ctx.rank * per, min(n, (self.ctx.rank + 1) * per)
        for i in range(lo, hi, bs):
            xb, lb = self.stage_batch(x[i:i + bs], labels[i:i + bs])
            B = xb.shape[0]
            if self.backend == "hip":
                self._C.hip_fwdbwd(xb, self.model.params, w.a1, w.a2, w.y,
                                   w.dz, w.dz2, w.dz1, lb, w.loss_accum,
                                   w.correct_accum, B, MODE_EVAL,
                                   native.current_stream_handle(),
                                   self._pool_mode, self._loss_mode)
            else:
                preds = self._forward_preds(xb, B)
                correct += int((preds == lb).sum().item())
        if self.backend == "hip":
            correct = int(w.correct_accum.item())
            w.correct_accum.zero_()
        dev = self.device if self.device.type == "cuda" else None
        correct = int(pdist.allreduce_scalar(float(correct), device=dev))
        return 100.0 * (1.0 - correct / float(n))

    def _forward_preds(self, xb: torch.Tensor, B: int) -> torch.Tensor:
        w = self.ws
        if self.backend == "cpu":
            a1, a2 = self._cpu_view(w.a1, B), self._cpu_view(w.a2, B)
            self._C.cpu_forward(xb, self.model.params, a1, a2, w.y[:B],
                                self._pool_mode, self._loss_mode)
            y = w.y[:B]
        else:
            _, _, y = torch_ref.forward(xb, self.model.params, self.cfg.pool,
                                        self.cfg.loss)
        return y.argmax(dim=1)

    @torch.no_grad()
    def classify(self, x: torch.Tensor) -> torch.Tensor:
        """Predicted labels for a host fp32 batch [N, 784] (the reference's
        `classify` entry point, Sequential/Main.cpp:186-200)."""
        w = self.ws
        preds = []
        bs = w.max_batch
        for i in range(0, x.shape[0], bs):
            xb = x[i:i + bs]
            B = xb.shape[0]
            if self.backend == "hip":
                xd = xb.to(self.device, dtype=self.act_dtype)
                dummy = torch.zeros(B, dtype=torch.int32, device=self.device)
                self._C.hip_fwdbwd(xd, self.model.params, w.a1, w.a2, w.y,
                                   w.dz, w.dz2, w.dz1, dummy, w.loss_accum,
                                   w.correct_accum, B, MODE_INFER,
                                   native.current_stream_handle(),
                                   self._pool_mode, self._loss_mode)
                preds.append(w.y[:B].argmax(dim=1).cpu())
            else:
                xd = xb.to(torch.float32)
                preds.append(self._forward_preds(xd, B).cpu())
        return torch.cat(preds)

    # ------------------------------------------------------------- training
    def train_epoch(self, x: torch.Tensor, labels: torch.Tensor,
                    log=print) -> float:
        """One epoch over a host dataset, DP-sharded; returns the mean
        per-sample error norm (the reference's per-epoch `error` print)."""
        Bl = self.ws.max_batch
        Bg = Bl * self.ctx.world_size
        n = (x.shape[0] // Bg) * Bg  # drop ragged tail like a fixed-shape bench
        total_loss, total_n = 0.0, 0

        def maybe_log():
            nonlocal total_loss, total_n
            if self.cfg.log_interval and \
                    self.global_step % self.cfg.log_interval == 0:
                loss, cnt = self.consume_loss()
                total_loss += loss
                total_n += cnt
                if self.ctx.is_main and cnt:
                    log(f"step {self.global_step}: error {loss / cnt:e}")

        if self.backend == "hip":
            # double-buffered async H2D staging on a copy stream
            from ..data.pipeline import DevicePrefetcher
            pf = DevicePrefetcher(x, labels, Bl, self.device, self.act_dtype,
                                  lo=self.ctx.rank * Bl, hi=n, stride=Bg)
            for xb, lb in pf:
                self.step(xb, lb)
                maybe_log()
        else:
            for s in range(0, n, Bg):
                lo = s + self.ctx.rank * Bl
                xb, lb = self.stage_batch(x[lo:lo + Bl], labels[lo:lo + Bl])
                self.step(xb, lb)
                maybe_log()
        loss, cnt = self.consume_loss()
        total_loss += loss
        total_n += cnt
        self.epoch += 1
        return total_loss / max(1, total_n)
