"""CudaPrefetcher — double-buffered H2D staging on a dedicated copy stream.

The reference moves each batch to the device synchronously inside the hot
loop (reference ddp.py:220 ``x.to(device)`` after a ``pin_memory=True``
DataLoader, ddp.py:151).  On MI355X that serializes an HBM3E-bound copy with
compute.  This wrapper keeps one batch in flight ahead of compute:

* the DataLoader yields pinned host tensors (``pin_memory=True``),
* each batch's H2D copy (and optional dtype cast) is issued on a dedicated
  HIP stream while the model computes on the previous batch,
* the compute stream waits on the copy stream (event ordering, no host
  sync) exactly when the batch is handed over.

SURVEY.md §2b "process-boundary data movement" obligation.
"""

from __future__ import annotations

import torch


class CudaPrefetcher:
    """Iterate ``loader`` yielding device-resident (x, y) one batch ahead.

    ``cast_dtype`` (e.g. bf16) is applied on-device to floating tensors —
    the cast costs device bandwidth, not host time.
    """

    def __init__(self, loader, device, cast_dtype=None):
        self.loader = loader
        self.device = torch.device(device)
        self.cast_dtype = cast_dtype
        self.stream = torch.cuda.Stream(device=self.device)

    def __len__(self):
        return len(self.loader)

    def _stage(self, batch):
        x, y = batch
        with torch.cuda.stream(self.stream):
            x = x.to(self.device, non_blocking=True)
            y = y.to(self.device, non_blocking=True)
            if self.cast_dtype is not None:
                if x.dtype.is_floating_point and x.dtype != self.cast_dtype:
                    x = x.to(self.cast_dtype)
                if y.dtype.is_floating_point and y.dtype != self.cast_dtype:
                    y = y.to(self.cast_dtype)
        return x, y

    def __iter__(self):
        it = iter(self.loader)
        try:
            nxt = self._stage(next(it))
        except StopIteration:
            return
        compute = torch.cuda.current_stream(self.device)
        while True:
            cur = nxt
            try:
                nxt_host = next(it)
            except StopIteration:
                nxt_host = None
            # hand-over: compute stream orders after the copy stream, and the
            # tensors (allocated on the copy stream) are marked as used by
            # the compute stream so the caching allocator can't recycle them
            # under an in-flight kernel.
            compute.wait_stream(self.stream)
            for t in cur:
                t.record_stream(compute)
            # next batch's copy goes out BEFORE yielding: it overlaps the
            # caller's forward/backward on the compute stream.
            if nxt_host is not None:
                nxt = self._stage(nxt_host)
            yield cur
            if nxt_host is None:
                return
