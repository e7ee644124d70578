"""Device prefetcher: pinned-host staging + H2D copy on a side HIP stream,
one batch ahead of compute (the reference's tf.data prefetch(AUTOTUNE) +
device feed, done the MI355X way). On CPU it's a passthrough."""

from __future__ import annotations

import torch


class DevicePrefetcher:
    def __init__(self, iterable, device: torch.device, dtype: torch.dtype):
        self._it = iter(iterable)
        self.device = device
        self.dtype = dtype
        self._gpu = device.type == "cuda"
        self._stream = torch.cuda.Stream(device) if self._gpu else None
        self._next = None
        self._preload()

    def _preload(self):
        batch = next(self._it, None)
        if batch is None:
            self._next = None
            return
        if not self._gpu:
            self._next = tuple(t.to(self.dtype) for t in batch)
            return
        with torch.cuda.stream(self._stream):
            out = []
            for t in batch:
                if t.device.type == "cpu":
                    t = t.contiguous().pin_memory()
                out.append(t.to(self.device, self.dtype, non_blocking=True))
            self._next = tuple(out)

    def __iter__(self):
        return self

    def __next__(self):
        if self._next is None:
            raise StopIteration
        if self._gpu:
            torch.cuda.current_stream(self.device).wait_stream(self._stream)
        batch = self._next
        # keep the tensors alive on the compute stream before overwriting
        if self._gpu:
            for t in batch:
                t.record_stream(torch.cuda.current_stream(self.device))
        self._preload()
        return batch
