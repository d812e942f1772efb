#!/usr/bin/env python3
"""Reference-compatible CLI entrypoint (BASELINE.json:5
"same CLI entrypoint and checkpoint format as the reference").

Launch exactly like the reference: one OS process per rank with
MASTER_ADDR / MASTER_PORT / RANK / WORLD_SIZE in the environment, e.g.

    MASTER_ADDR=127.0.0.1 MASTER_PORT=29500 WORLD_SIZE=2 RANK=0 \
        python dist_mnist.py --backend gloo &
    MASTER_ADDR=127.0.0.1 MASTER_PORT=29500 WORLD_SIZE=2 RANK=1 \
        python dist_mnist.py --backend gloo &

or via torchrun (env is filled in for you):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 dist_mnist.py --backend nccl

The reference tree at v0 was empty, so the flag surface is the
union recalled in SURVEY.md §5 "Config / flag system", frozen here:
--backend --epochs --lr --momentum --batch-size --seed --modelpath --no-cuda
--log-interval (+ framework extensions that never collide: --bucket-mb,
--resume, --steps-per-epoch, --ddp {bucketed,reference}).

Per-epoch output reproduces the reference's line: Rank <r>, epoch <e>: <loss>.
Checkpoint: torch.save(state_dict) with the SURVEY.md §3.4 keys.
"""

from __future__ import annotations

import argparse
import os

import torch
import torch.distributed as dist

from dist_mnist_amd.data import partition_dataset
from dist_mnist_amd.models import Net
from dist_mnist_amd.ops import SGD
from dist_mnist_amd.parallel import DistributedDataParallel, average_gradients, init_distributed
from dist_mnist_amd.runtime import Prefetcher
from dist_mnist_amd.utils import load_checkpoint, save_checkpoint


def parse_args(argv=None):
    p = argparse.ArgumentParser(description="MI355X-native distributed MNIST training")
    p.add_argument("--backend", type=str, default=None,
                   help="torch.distributed backend (gloo CPU | nccl == RCCL on ROCm)")
    p.add_argument("--epochs", type=int, default=10)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--momentum", type=float, default=0.5)
    p.add_argument("--batch-size", type=int, default=128,
                   help="GLOBAL batch size, split across ranks (reference R4)")
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--modelpath", type=str, default=None,
                   help="where to torch.save the final state_dict (rank 0)")
    p.add_argument("--no-cuda", action="store_true")
    p.add_argument("--log-interval", type=int, default=10)
    # framework extensions (absent in reference, non-colliding):
    p.add_argument("--bucket-mb", type=float, default=25.0,
                   help="DDP gradient bucket size in MiB (xGMI-tuned default)")
    p.add_argument("--ddp", choices=["bucketed", "reference"], default="bucketed",
                   help="bucketed = overlapped engine; reference = per-param blocking allreduce")
    p.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32",
                   help="bf16 = native NHWC engine on GPU (fp32 master weights; "
                        "checkpoints unchanged)")
    p.add_argument("--resume", type=str, default=None)
    p.add_argument("--eval", action="store_true",
                   help="per-epoch accuracy on a held-out synthetic split (rank 0)")
    p.add_argument("--perf-log", action="store_true",
                   help="per-epoch images/sec + gradient-sync time")
    p.add_argument("--steps-per-epoch", type=int, default=None,
                   help="cap steps per epoch (synthetic data is unbounded)")
    p.add_argument("--dataset-size", type=int, default=60000)
    return p.parse_args(argv)


def evaluate(model, device, use_bf16, n=2048, seed=99991):
    """Held-out accuracy on a fixed synthetic split (extension; the reference
    lineage's DDP variant carries a test() loop — SURVEY.md §2a)."""
    from dist_mnist_amd.data import SyntheticMNIST

    ds = SyntheticMNIST(n=n, seed=seed)
    model.eval()
    correct = 0
    with torch.no_grad():
        for i in range(0, n, 256):
            x, t = ds.batch(range(i, min(n, i + 256)), device=device)
            if use_bf16:
                x = x[:, 0].to(torch.bfloat16).unsqueeze(-1).contiguous()
            logp = model(x)
            correct += int((logp.argmax(1) == t).sum())
    model.train()
    return correct / n


def run(rank: int, world_size: int, args):
    use_cuda = torch.cuda.is_available() and not args.no_cuda
    device = torch.device(f"cuda:{rank % max(1, torch.cuda.device_count())}" if use_cuda else "cpu")
    torch.manual_seed(args.seed)

    loader, bsz = partition_dataset(rank, world_size, global_batch=args.batch_size,
                                    n=args.dataset_size, seed=args.seed)
    use_bf16 = args.dtype == "bf16" and use_cuda
    model = Net().to(device)
    if use_bf16:
        model.bf16_native_()
    if args.resume:
        load_checkpoint(model, args.resume, device)

    if world_size > 1 and args.ddp == "bucketed":
        model = DistributedDataParallel(model, bucket_bytes=int(args.bucket_mb * 1024 * 1024))
    optimizer = SGD(model.parameters(), lr=args.lr, momentum=args.momentum)

    import time as _time
    for epoch in range(args.epochs):
        model.train()
        epoch_loss = 0.0
        steps = 0
        sync_s = 0.0
        t_epoch = _time.perf_counter()
        batches = Prefetcher(iter(loader), device) if use_cuda else loader
        for data, target in batches:
            if not use_cuda:
                data, target = data.to(device), target.to(device)
            if use_bf16:
                # native engine input layout: NHWC bf16 [N, 28, 28, 1]
                data = data[:, 0].to(torch.bfloat16).unsqueeze(-1).contiguous()
            optimizer.zero_grad() if not isinstance(model, DistributedDataParallel) \
                else model.zero_grad()
            _, loss = model(data, target)
            loss.backward()
            if world_size > 1:
                t_s = _time.perf_counter()
                if isinstance(model, DistributedDataParallel):
                    model.sync()
                else:
                    average_gradients(model)
                sync_s += _time.perf_counter() - t_s
            optimizer.step()
            epoch_loss += loss.item()
            steps += 1
            if args.steps_per_epoch and steps >= args.steps_per_epoch:
                break
        # the reference's exact per-epoch line (SURVEY.md §5 observability)
        print("Rank ", rank, ", epoch ", epoch, ": ", epoch_loss / max(1, steps))
        if args.perf_log:
            dt = _time.perf_counter() - t_epoch
            print(f"[perf] rank {rank} epoch {epoch}: "
                  f"{steps * bsz / max(dt, 1e-9):.0f} images/s, "
                  f"grad-sync {sync_s * 1000:.1f} ms total")
        if args.eval and rank == 0:
            base = model.module if isinstance(model, DistributedDataParallel) else model
            acc = evaluate(base, device, use_bf16)
            print(f"[eval] epoch {epoch}: held-out accuracy {acc:.4f}")

    if args.modelpath and rank == 0:
        save_checkpoint(model, args.modelpath)


def main(argv=None):
    args = parse_args(argv)
    rank, world_size = init_distributed(args.backend)
    try:
        run(rank, world_size, args)
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


if __name__ == "__main__":
    main()
