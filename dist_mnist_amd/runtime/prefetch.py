"""Pinned-host double-buffered H2D prefetcher.

BASELINE.json:5: "the data loader feeds synthetic 1x28x28 tensors from pinned
host memory via hipMemcpyAsync on a side stream overlapped with compute."

Implementation: a dedicated HIP side stream issues the H2D copies of batch
t+1 while the compute stream works on batch t; an event on the copy stream is
waited on by the compute stream before batch t+1 is consumed (event-synced,
never a host sync). Two device-side slots alternate. On CPU (gloo plumbing
config) it degrades to a passthrough.
"""

from __future__ import annotations

import torch


class Prefetcher:
    def __init__(self, source, device, dtype=torch.float32, depth: int = 2):
        """source: iterator of (x_cpu, y_cpu) with pinned tensors preferred."""
        self.source = iter(source)
        self.device = torch.device(device)
        self.dtype = dtype
        self.is_cuda = self.device.type == "cuda"
        if self.is_cuda:
            self.copy_stream = torch.cuda.Stream(device=self.device)
            self.slots = [None] * depth
            self.events = [torch.cuda.Event() for _ in range(depth)]
            self.idx = 0
            self._preload()

    def _preload(self):
        try:
            x, y = next(self.source)
        except StopIteration:
            self.slots[self.idx % len(self.slots)] = None
            return
        slot = self.idx % len(self.slots)
        with torch.cuda.stream(self.copy_stream):
            xd = x.to(self.device, self.dtype, non_blocking=True)
            yd = y.to(self.device, non_blocking=True)
            self.events[slot].record(self.copy_stream)
        self.slots[slot] = (xd, yd)

    def __iter__(self):
        return self

    def __next__(self):
        if not self.is_cuda:
            x, y = next(self.source)
            return x.to(self.dtype), y
        slot = self.idx % len(self.slots)
        item = self.slots[slot]
        if item is None:
            raise StopIteration
        # compute stream waits for this slot's copy; no host sync
        cur = torch.cuda.current_stream(self.device)
        cur.wait_event(self.events[slot])
        # the tensors were allocated on the copy stream: mark them as used by
        # the compute stream so the caching allocator cannot hand their
        # memory to a later H2D while this stream still reads them
        item[0].record_stream(cur)
        item[1].record_stream(cur)
        self.idx += 1
        self._preload()  # kick off the next copy on the side stream
        return item
