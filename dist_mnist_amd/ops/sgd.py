"""SGD with momentum — fused multi-tensor HIP step on device.

Reference semantics (torch.optim.SGD, lr=0.01 momentum=0.5 — SURVEY.md §2a R8):
    buf = momentum * buf + grad ; param -= lr * buf
(no dampening, no nesterov, no weight decay — the reference uses none).

On GPU the whole parameter set updates in ONE kernel launch: a chunk table
(param/grad/momentum pointers + lengths) is built once at init and the kernel
grid-strides across the concatenated element range. On CPU it falls back to
plain torch ops (used by the gloo plumbing config and the no-GPU tests).

step() also RESETS the grads to zero after consuming them (fused into the
same kernel stores on GPU): the next backward then accumulates onto zeros
with no per-param zero_() launches, which is what lets the hipGraph step
(runtime/graphs.py) drop grad-zeroing from the captured region entirely.
An explicit zero_grad() before backward stays a harmless no-op.
"""

from __future__ import annotations

import torch

from . import native


class SGD(torch.optim.Optimizer):
    def __init__(self, params, lr=0.01, momentum=0.0):
        defaults = dict(lr=lr, momentum=momentum)
        super().__init__(params, defaults)
        # Pointer tables keyed by (param_group_index, tuple-of-grad-pointers).
        # Entries are NEVER evicted: hipGraph-captured optimizer steps
        # (runtime/graphs.py) record the raw device addresses of a slot's
        # (ptrs_t, chunks_t) tensors, so freeing a superseded table would
        # leave an earlier slot's captured graph replaying over freed memory.
        # A training run only ever has a handful of distinct grad-pointer
        # sets (one per graph slot), so the cache stays tiny.
        self._tables: dict = {}

    CHUNK = 4096  # elements per block in the fused kernel

    def _build_table(self, params):
        """Device tables for the one-launch fused step:
        ptrs[t] = (param_ptr, grad_ptr, momentum_ptr, numel);
        chunks[b] = (tensor_idx, elem_offset) — one block per chunk.

        `params` is exactly the filtered list step() launches over, so the
        cache key (grad pointers of that list) and the table rows always
        describe the same tensors.
        """
        ptrs = []
        chunks = []
        for ti, p in enumerate(params):
            st = self.state[p]
            if "momentum_buffer" not in st:
                st["momentum_buffer"] = torch.zeros_like(p, memory_format=torch.contiguous_format)
            assert p.is_contiguous() and p.grad.is_contiguous()
            ptrs.append(
                (p.data_ptr(), p.grad.data_ptr(), st["momentum_buffer"].data_ptr(), p.numel())
            )
            for off in range(0, p.numel(), self.CHUNK):
                chunks.append((ti, off))
        dev = params[0].device
        ptrs_t = torch.tensor(ptrs, dtype=torch.int64, device=dev)
        chunks_t = torch.tensor(chunks, dtype=torch.int32, device=dev)
        return ptrs_t, chunks_t

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            params = [p for p in group["params"] if p.requires_grad and p.grad is not None]
            if not params:
                continue
            if params[0].is_cuda:
                key = (gi, tuple(p.grad.data_ptr() for p in params))
                table = self._tables.get(key)
                if table is None:
                    table = self._build_table(params)
                    self._tables[key] = table
                ptrs_t, chunks_t = table
                native.ext().fused_sgd_step(
                    ptrs_t, chunks_t, self.CHUNK, group["lr"], group["momentum"]
                )
            else:
                for p in params:
                    buf = self.state[p].get("momentum_buffer")
                    if group["momentum"] != 0.0:
                        if buf is None:
                            buf = torch.clone(p.grad).detach()
                            self.state[p]["momentum_buffer"] = buf
                        else:
                            buf.mul_(group["momentum"]).add_(p.grad)
                        p.add_(buf, alpha=-group["lr"])
                    else:
                        p.add_(p.grad, alpha=-group["lr"])
                    p.grad.zero_()  # same consume-and-reset as the GPU kernel
        if torch.cuda.is_available() and torch.cuda.is_initialized():
            # one batched launch re-packs every cached conv-weight buffer the
            # update just invalidated (native/ext_pack.inc); no-op when the
            # model has no packed weights (e.g. LeNet's smallK engines)
            native.ext().repack_refresh_all()
        return loss
