#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract — see task brief).

    python bench.py --gpus N --steps K --warmup W

For N > 1 the driver launches this under torch.distributed.run with one rank
per GPU (RCCL over xGMI); RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* come from the
env. Measures the BASELINE.json metric — images/sec (whole node) of the
LeNet-style CNN on synthetic MNIST-shaped data (BASELINE.json:2, configs 2-4):
full training semantics per step (fwd incl. dropout, fused loss, bwd, DDP
gradient averaging, fused SGD), W untimed warmup steps, then exactly K timed
steps bracketed by barrier+synchronize on both sides, MAX step time over
ranks, one JSON line from rank 0.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

from dist_mnist_amd.data import SyntheticBatchStream
from dist_mnist_amd.models import Net
from dist_mnist_amd.ops import SGD
from dist_mnist_amd.parallel import DistributedDataParallel, init_distributed
from dist_mnist_amd.runtime import Prefetcher


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=None,
                   help="per-GPU batch (default 8192 on GPU — BASELINE config 4 shape)")
    p.add_argument("--dtype", type=str, default="bf16", choices=["fp32", "bf16"],
                   help="compute dtype of the hot path (bf16 = NHWC native engine, "
                        "the BASELINE config-4 headline; fp32 = NCHW engine)")
    p.add_argument("--bucket-mb", type=float, default=25.0)
    p.add_argument("--no-graphs", action="store_true",
                   help="disable hipGraph step capture (eager launches)")
    p.add_argument("--comm", choices=["torch", "native"], default="torch",
                   help="gradient collective backend for world>1: torch "
                        "ProcessGroupNCCL, or the framework's C++ RCCL core")
    p.add_argument("--model", choices=["lenet", "resnet18"], default="lenet",
                   help="lenet = BASELINE headline config; resnet18 = config 5 "
                        "(synthetic 3x224x224, bf16 only)")
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", 1))
    use_cuda = torch.cuda.is_available()
    if world > 1:
        rank, world = init_distributed("nccl" if use_cuda else "gloo")
    else:
        rank = 0
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    use_bf16 = args.dtype == "bf16" and use_cuda
    torch.manual_seed(1234)

    if args.model == "resnet18":
        from dist_mnist_amd.models import ResNet18

        assert use_cuda, "resnet18 config runs on the bf16 GPU engine"
        use_bf16 = True
        batch = args.batch or 256
        model = ResNet18().to(device).train()
        layout = "rgb224_bf16"
        n_classes = 1000
    else:
        batch = args.batch or (8192 if use_cuda else 256)
        model = Net().to(device).train()
        if use_bf16:
            model.bf16_native_()
        layout = "nhwc_bf16" if use_bf16 else "nchw_f32"
        n_classes = 10
    if world > 1:
        native_comm = None
        if args.comm == "native" and use_cuda:
            from dist_mnist_amd.parallel import RcclComm

            native_comm = RcclComm.bootstrap()
        model = DistributedDataParallel(model, bucket_bytes=int(args.bucket_mb * 2**20),
                                        native_comm=native_comm)
    opt = SGD(model.parameters(), lr=0.01, momentum=0.5)

    stream = SyntheticBatchStream(batch_size=batch, n_pool=4, seed=1234 + rank,
                                  pin=use_cuda, layout=layout, n_classes=n_classes)
    batches = Prefetcher(stream, device,
                         dtype=torch.bfloat16 if use_bf16 else torch.float32)

    use_graphs = use_cuda and not args.no_graphs
    graphed = None
    step_idx = [0]
    if use_graphs:
        from dist_mnist_amd.runtime import GraphedStep
        from dist_mnist_amd.runtime.graphs import resync_after_fallback

        x0, t0 = next(batches)
        ddp = model if isinstance(model, DistributedDataParallel) else None
        try:
            # GraphedStep raises CaptureFailed on ALL ranks after a collective
            # agreement (before any constructor collective), so the fleet
            # takes one consistent step path.
            graphed = GraphedStep(model, opt, x0, t0, ddp=ddp)
        except Exception as e:  # capture failure must never kill the bench
            print(f"# hipGraph capture failed ({type(e).__name__}: {e}); "
                  "falling back to eager launches", flush=True)
            graphed = None
            # warmup drifted params rank-locally; resync before eager training
            resync_after_fallback(opt, ddp)
        if graphed is None and ddp is not None:
            ddp._install_hooks()  # restore comm-in-backward overlap

    def one_step():
        x, t = next(batches)
        if graphed is not None:
            slot = step_idx[0] % len(graphed.slots)
            step_idx[0] += 1
            graphed.slots[slot][0].copy_(x, non_blocking=True)
            graphed.slots[slot][1].copy_(t, non_blocking=True)
            return graphed.step(slot)
        # Single-GPU eager: ASSIGN-mode grads (p.grad = None) so backward
        # allocates instead of accumulating — no ATen add/Fill launches in
        # the step. The caching allocator hands back the same blocks, so
        # the fused-SGD pointer-table cache still hits. DDP keeps its
        # persistent bucket views (the SGD kernel zeroes them in-kernel).
        if not isinstance(model, DistributedDataParallel):
            for p in model.parameters():
                p.grad = None
        _, loss = model(x, t)
        loss.backward()
        if isinstance(model, DistributedDataParallel):
            model.sync()
        opt.step()
        return loss

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize(device)

    for _ in range(args.warmup):
        one_step()
    barrier_sync()

    t0 = time.perf_counter()
    last_loss = None
    for _ in range(args.steps):
        last_loss = one_step()
    barrier_sync()
    dt = time.perf_counter() - t0

    # MAX step time over ranks
    if world > 1:
        dt_t = torch.tensor([dt], dtype=torch.float64,
                            device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(dt_t, op=dist.ReduceOp.MAX)
        dt = float(dt_t.item())

    n_gpus = world if use_cuda else world  # one rank per GPU
    images_per_sec = world * batch * args.steps / dt

    if rank == 0:
        result = {
            "metric": ("images/sec (whole node), ResNet-18 synthetic 3x224x224"
                       if args.model == "resnet18"
                       else "images/sec (whole node), LeNet-style CNN on MNIST"),
            "value": images_per_sec,
            "unit": "images/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": dt / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "bf16" if use_bf16 else "fp32",
            "data": "synthetic",
            "final_loss": float(last_loss.item()) if last_loss is not None else None,
            "config": {
                "model": ("resnet18-synthetic-3x224x224" if args.model == "resnet18"
                          else "lenet5-style-2conv (SURVEY.md R5)"),
                "global_batch": world * batch,
                "per_gpu_batch": batch,
                "seq_len": None,
                "parallelism": f"dp{world}",
                "device": str(device),
            },
        }
        print(json.dumps(result))

    if world > 1 and dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
