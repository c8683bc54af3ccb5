"""Device input pipeline: double-buffered async H2D staging on a dedicated
copy stream (replaces the reference's per-sample synchronous `setOutput`
upload, CUDA/layer.cu:59-63 — SURVEY.md §7 step 2).

The copy of batch i+1 overlaps the compute of batch i; the compute stream
waits on the copy event, and the copy stream waits for the compute that
last used the target buffer before overwriting it.
"""
from __future__ import annotations

from typing import Iterator, Tuple

import torch


class DevicePrefetcher:
    """Iterates (x_device, labels_device) batches of a host dataset.

    x: host fp32 [N, P]; labels: host int64 [N].  Batches are contiguous
    slices [lo + i*stride, +batch) for rank-sharded epochs.
    """

    def __init__(self, x: torch.Tensor, labels: torch.Tensor, batch: int,
                 device: torch.device, act_dtype: torch.dtype,
                 lo: int = 0, hi: int | None = None, stride: int = 0):
        assert device.type == "cuda", "DevicePrefetcher is for GPU staging"
        self.x, self.labels = x, labels
        self.batch = batch
        self.device = device
        self.lo = lo
        self.hi = hi if hi is not None else x.shape[0]
        self.stride = stride if stride > 0 else batch
        self.copy_stream = torch.cuda.Stream(device=device)
        P = x.shape[1]
        # pinned host staging + device double buffers
        self._hx = [torch.empty(batch, P, dtype=torch.float32,
                                pin_memory=True) for _ in range(2)]
        self._hl = [torch.empty(batch, dtype=torch.int32, pin_memory=True)
                    for _ in range(2)]
        self._dx = [torch.empty(batch, P, dtype=act_dtype, device=device)
                    for _ in range(2)]
        self._dl = [torch.empty(batch, dtype=torch.int32, device=device)
                    for _ in range(2)]
        self._copy_done = [torch.cuda.Event(), torch.cuda.Event()]
        self._compute_done = [torch.cuda.Event(), torch.cuda.Event()]

    def _issue_copy(self, slot: int, off: int) -> None:
        b = self.batch
        # the previous async H2D from this pinned buffer may still be in
        # flight (the host loop runs arbitrarily far ahead of the device);
        # wait for it before overwriting the staging buffer
        self._copy_done[slot].synchronize()
        self._hx[slot].copy_(self.x[off:off + b])
        self._hl[slot].copy_(self.labels[off:off + b].to(torch.int32))
        with torch.cuda.stream(self.copy_stream):
            # don't overwrite a buffer the compute stream still reads
            self.copy_stream.wait_event(self._compute_done[slot])
            self._dx[slot].copy_(self._hx[slot], non_blocking=True)
            self._dl[slot].copy_(self._hl[slot], non_blocking=True)
            self._copy_done[slot].record(self.copy_stream)

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        offs = list(range(self.lo, self.hi - self.batch + 1, self.stride))
        if not offs:
            return
        for ev in self._compute_done:
            ev.record()  # buffers initially free
        self._issue_copy(0, offs[0])
        for i, off in enumerate(offs):
            slot = i & 1
            if i + 1 < len(offs):
                self._issue_copy(slot ^ 1, offs[i + 1])
            cur = torch.cuda.current_stream()
            cur.wait_event(self._copy_done[slot])
            yield self._dx[slot], self._dl[slot]
            self._compute_done[slot].record(cur)
