"""hipGraph-captured training step (guide: "HIP streams and graphs instead of
a tracing compiler"; SURVEY.md §7: at 21k params LeNet is launch-latency-bound
— the whole fwd+bwd+SGD step captures into one executable graph).

Two static input slots, one captured graph per slot: the Prefetcher's side
stream fills slot t+1 while the compute stream replays slot t's graph (event
synced both ways, never a host sync).

Distributed mode splits the step into [fwd+bwd graph] -> eager bucket
all-reduce (RCCL; kept OUTSIDE capture so multi-GPU replay never depends on
collective capture support) -> [scale+SGD graph]. LeNet's gradient is one
87 KB bucket: the collective is pure xGMI latency, so losing backward overlap
costs nothing at this scale (SURVEY.md §5 comm).
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class CaptureFailed(RuntimeError):
    """hipGraph capture failed — raised on ALL ranks after a collective
    agreement, so a mixed graphed/eager fleet can never form (the eager and
    graphed paths issue bucket collectives in different orders)."""


def resync_after_fallback(opt, ddp):
    """Restore rank-identical state after a capture failure at world>1.

    GraphedStep's warmup steps run without gradient sync, so ranks drift;
    on the graphed path the constructor re-broadcasts params+momentum, but
    on the eager fallback the caller must resync instead. Different ranks
    may have failed at different points (momentum buffers may exist on some
    ranks only), so the momentum state is CLEARED everywhere — zeros are
    trivially rank-identical — rather than broadcast, and the pointer-table
    cache with it (its tables reference the freed buffers)."""
    if ddp is None or not ddp.enabled:
        return
    with torch.no_grad():
        for p in ddp.module.parameters():
            dist.broadcast(p.data, src=0, group=ddp.pg)
    opt.state.clear()
    if hasattr(opt, "_tables"):
        opt._tables.clear()


class GraphedStep:
    def __init__(self, model, optimizer, sample_x, sample_t, ddp=None,
                 n_slots: int = 2, warmup: int = 3):
        """model: Net in bf16-native or fp32 mode (already on device/train).
        ddp: DistributedDataParallel wrapper (enabled) or None.
        sample_x/sample_t: device tensors of the static batch shape."""
        self.model = model
        self.opt = optimizer
        self.ddp = ddp
        if ddp is not None:
            ddp.disable_comm_hooks()
        self.device = sample_x.device
        self.slots = []
        for _ in range(n_slots):
            self.slots.append((sample_x.clone(), sample_t.clone()))

        # The whole warmup+capture region issues NO collectives (comm hooks
        # are disabled above), so a rank that fails anywhere in it can still
        # reach the agreement point below with a matching collective
        # sequence. Constructor collectives (param/momentum re-broadcast)
        # happen only AFTER all ranks agree capture succeeded.
        err = None
        try:
            self._capture(warmup)
        except Exception as e:  # noqa: BLE001 — agreement must still run
            err = e
        if ddp is not None and ddp.enabled:
            dev = (self.device if dist.get_backend(ddp.pg) == "nccl" else "cpu")
            ok = torch.tensor([0.0 if err is not None else 1.0], device=dev)
            dist.all_reduce(ok, op=dist.ReduceOp.MIN, group=ddp.pg)
            if ok.item() < 1.0:
                raise CaptureFailed(
                    f"capture failed on at least one rank "
                    f"(this rank: {err!r})") from err
        elif err is not None:
            raise CaptureFailed(str(err)) from err

        # warmup steps ran WITHOUT gradient sync (hooks disabled, rank-local
        # data) so ranks drifted; re-broadcast params AND optimizer momentum
        # in place — the captured graphs hold raw pointers, so in-place
        # copies are safe and keep ranks bitwise-identical from here on.
        if self.ddp is not None and self.ddp.enabled:
            with torch.no_grad():
                for p in self.ddp.module.parameters():
                    dist.broadcast(p.data, src=0, group=self.ddp.pg)
                for st in self.opt.state.values():
                    buf = st.get("momentum_buffer")
                    if buf is not None:
                        dist.broadcast(buf, src=0, group=self.ddp.pg)
            torch.cuda.synchronize(self.device)

    def _capture(self, warmup: int):
        side = torch.cuda.Stream(device=self.device)
        # warmup on a side stream (capture requires non-default-stream history)
        side.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(side):
            for _ in range(warmup):
                self._eager_fwd_bwd(*self.slots[0])
                self._eager_opt()
        torch.cuda.current_stream(self.device).wait_stream(side)
        torch.cuda.synchronize(self.device)

        # capture one fwd+bwd graph per slot. No grad-zeroing inside capture:
        # - DDP: grads are bucket views, so backward ACCUMULATES; the fused
        #   SGD kernel resets them to zero as it consumes them (ops/sgd.py).
        # - single GPU: grads are set to None before each slot's capture, so
        #   backward ASSIGNS pool-allocated tensors instead (pointer-stable
        #   across replays of that slot) — no accumulate-add kernels at all
        #   (62 of them per step on ResNet-18). Each slot then gets its own
        #   captured optimizer step over its own grad pointers.
        self.fb_graphs = []
        self.losses = []
        self.opt_graphs = []
        pool = None
        slot_grads = []
        for x, t in self.slots:
            if self.ddp is None:
                for p in self.model.parameters():
                    p.grad = None
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=pool):
                loss = self._eager_fwd_bwd(x, t, zero=False)
            pool = g.pool()
            self.fb_graphs.append(g)
            self.losses.append(loss)
            slot_grads.append([p.grad for p in self.model.parameters()])

        if self.ddp is None:
            # one eager step per slot first: it builds/caches the fused-SGD
            # pointer table for that slot's grads (device-tensor creation is
            # not allowed inside capture), then the capture records the
            # table-driven kernel. The extra updates are just more warmup.
            params = list(self.model.parameters())
            for grads, _ in zip(slot_grads, self.slots):
                for p, gr in zip(params, grads):
                    p.grad = gr
                    # capture RECORDED the producing kernels without running
                    # them, so the pool tensors hold uninitialized memory —
                    # zero them so this table-building step is a benign
                    # momentum-decay update, not a garbage-gradient one
                    gr.zero_()
                self.opt.step()
                og = torch.cuda.CUDAGraph()
                with torch.cuda.graph(og, pool=pool):
                    self._eager_opt(scale=False)
                self.opt_graphs.append(og)
        else:
            # optimizer graph (shared; includes the DDP 1/world scaling)
            og = torch.cuda.CUDAGraph()
            with torch.cuda.graph(og, pool=pool):
                self._eager_opt(scale=True)
            self.opt_graphs = [og for _ in self.slots]
            # capture records without executing, so the bucket-view grads
            # still hold warmup values; establish the all-zero invariant
            # once before the first replay (assign-mode slots need none:
            # their backward overwrites)
            self._zero_grads()

    # -- eager pieces (also used during warmup/capture) ---------------------
    def _zero_grads(self):
        if self.ddp is not None:
            self.ddp.zero_grad()
        else:
            for p in self.model.parameters():
                if p.grad is not None:
                    p.grad.zero_()
                else:
                    p.grad = torch.zeros_like(p)

    def _eager_fwd_bwd(self, x, t, zero: bool = True):
        if zero:
            self._zero_grads()
        _, loss = self.model(x, t)
        loss.backward()
        return loss

    def _eager_opt(self, scale: bool = False):
        if scale and self.ddp is not None and self.ddp.enabled:
            inv = 1.0 / self.ddp.world_size
            for b in self.ddp.buckets:
                b.flat.mul_(inv)
        self.opt.step()

    # -- per-step ------------------------------------------------------------
    def step(self, slot: int):
        """Replay the step for `slot` (whose input buffers must be filled and
        event-synced by the caller). Returns the loss tensor (device)."""
        self.fb_graphs[slot].replay()
        if self.ddp is not None and self.ddp.enabled:
            # one eager all-reduce per bucket (grads are bucket views filled
            # by the captured backward)
            for b in self.ddp.buckets:
                if self.ddp.native_comm is not None:
                    self.ddp.native_comm.all_reduce_sum(b.flat)
                else:
                    dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.ddp.pg)
        self.opt_graphs[slot].replay()
        return self.losses[slot]
