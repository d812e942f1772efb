#!/usr/bin/env python3
"""Prove the H2D copy/compute overlap (BASELINE.json:5; VERDICT round-1
item 6): runs the bench step loop with hipEvents bracketing every H2D copy
on the Prefetcher's side stream and every training step on the compute
stream, then reports how much of each copy interval lies inside compute
intervals.

    gpurun -- 'python tools/overlap_probe.py > gpurun_out/overlap.txt'

Events on different streams share the device timebase, so elapsed_time
between a common origin event and each bracket event yields comparable
timestamps. Copies are timed on the copy stream (where Prefetcher issues
them); steps on the current (compute) stream.
"""

import sys

import torch

sys.path.insert(0, ".")

from dist_mnist_amd.data import SyntheticBatchStream  # noqa: E402
from dist_mnist_amd.models import Net  # noqa: E402
from dist_mnist_amd.ops import SGD  # noqa: E402
from dist_mnist_amd.runtime import GraphedStep, Prefetcher  # noqa: E402


class TimedPrefetcher(Prefetcher):
    """Prefetcher that records an event pair around each H2D copy."""

    def __init__(self, *a, **kw):
        self.copy_events = []
        super().__init__(*a, **kw)

    def _preload(self):
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record(self.copy_stream)
        super()._preload()
        e1.record(self.copy_stream)
        self.copy_events.append((e0, e1))


def main():
    assert torch.cuda.is_available(), "run on the GPU box"
    dev = torch.device("cuda:0")
    torch.manual_seed(1234)
    batch = 8192
    net = Net().to(dev).train().bf16_native_()
    opt = SGD(net.parameters(), lr=0.01, momentum=0.5)
    stream = SyntheticBatchStream(batch_size=batch, n_pool=4, seed=1234,
                                  pin=True, layout="nhwc_bf16")
    batches = TimedPrefetcher(stream, dev, dtype=torch.bfloat16)

    x0, t0 = next(batches)
    gs = GraphedStep(net, opt, x0, t0)

    origin = torch.cuda.Event(enable_timing=True)
    origin.record()
    torch.cuda.synchronize()

    steps = 60
    step_events = []
    for i in range(steps):
        slot = i % len(gs.slots)
        x, t = next(batches)
        gs.slots[slot][0].copy_(x, non_blocking=True)
        gs.slots[slot][1].copy_(t, non_blocking=True)
        e0 = torch.cuda.Event(enable_timing=True)
        e1 = torch.cuda.Event(enable_timing=True)
        e0.record()
        gs.step(slot)
        e1.record()
        step_events.append((e0, e1))
    torch.cuda.synchronize()

    comp = [(origin.elapsed_time(a), origin.elapsed_time(b))
            for a, b in step_events]
    # skip warm-up copies issued before the timed loop
    copies = [(origin.elapsed_time(a), origin.elapsed_time(b))
              for a, b in batches.copy_events
              if origin.elapsed_time(a) >= comp[0][0] - 0.05]

    def overlap(iv, ivs):
        s, e = iv
        return sum(max(0.0, min(e, ce) - max(s, cs)) for cs, ce in ivs)

    tot_copy = sum(e - s for s, e in copies)
    tot_olap = sum(overlap(c, comp) for c in copies)
    comp_span = comp[-1][1] - comp[0][0]
    comp_busy = sum(e - s for s, e in comp)
    print(f"steps={steps} batch={batch} copies_timed={len(copies)}")
    print(f"compute span        : {comp_span:8.3f} ms "
          f"({comp_busy:.3f} ms in-step)")
    print(f"H2D copy time       : {tot_copy:8.3f} ms "
          f"({tot_copy / max(1, len(copies)) * 1000:.1f} us/copy)")
    print(f"copy under compute  : {tot_olap:8.3f} ms "
          f"= {100.0 * tot_olap / max(tot_copy, 1e-9):.1f}% of copy time")
    print(f"copy exposed        : {tot_copy - tot_olap:8.3f} ms "
          f"({100.0 * (tot_copy - tot_olap) / comp_span:.2f}% of wall)")
    ok = tot_olap / max(tot_copy, 1e-9) > 0.5
    print("VERDICT:", "OVERLAPPED (copies hide under compute)" if ok
          else "NOT OVERLAPPED")
    return 0 if ok else 1


if __name__ == "__main__":
    raise SystemExit(main())
