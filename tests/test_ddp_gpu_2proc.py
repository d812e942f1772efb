"""Two-process DDP on ONE GPU (gloo transport, bf16 native engine, hipGraph
step) — dress rehearsal for the driver's multi-GPU run: everything except the
RCCL transport is the exact world>1 bench path (DDP bucket views, hook
overlap, GraphedStep with out-of-capture collectives, param re-broadcast)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)

        from dist_mnist_amd.models import Net
        from dist_mnist_amd.ops import SGD
        from dist_mnist_amd.parallel import DistributedDataParallel
        from dist_mnist_amd.runtime import GraphedStep

        torch.manual_seed(7)
        net = Net().to("cuda:0").train().bf16_native_()
        ddp = DistributedDataParallel(net)  # gloo CPU<->GPU copies inside
        opt = SGD(ddp.parameters(), lr=0.05, momentum=0.5)

        g = torch.Generator().manual_seed(100 + rank)
        x = torch.randn(64, 28, 28, 1, generator=g).to(torch.bfloat16).cuda()
        t = torch.randint(0, 10, (64,), generator=g).cuda()

        gs = GraphedStep(ddp, opt, x, t, ddp=ddp)
        losses = []
        for i in range(6):
            slot = i % 2
            gs.slots[slot][0].copy_(x)
            gs.slots[slot][1].copy_(t)
            loss = gs.step(slot)
            losses.append(float(loss))
        torch.cuda.synchronize()
        assert all(v == v and v < 20 for v in losses), losses

        # ranks stay in sync: params must be identical after synced steps
        flat = torch.cat([p.flatten() for p in ddp.module.parameters()]).cpu()
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        torch.testing.assert_close(gathered[0], gathered[1], rtol=0, atol=0)
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {type(e).__name__}: {e}\n{traceback.format_exc()[-800:]}"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_graphed_two_process_one_gpu(free_port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 2, free_port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, status = q.get()
        results[rank] = status
    for p in procs:
        p.join(timeout=120)
    assert all(s == "ok" for s in results.values()), results
