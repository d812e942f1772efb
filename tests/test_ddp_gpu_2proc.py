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


def _run_two(worker, free_port, timeout=240):
    import queue as _queue

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, 2, free_port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(2):
            try:
                rank, status = q.get(timeout=timeout)
            except _queue.Empty:
                results["timeout"] = "a worker never reported (hang?)"
                break
            results[rank] = status
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():  # kill the exact child PID, never by pattern
                p.terminate()
                p.join(timeout=30)
    assert len(results) == 2 and all(s == "ok" for s in results.values()), results


@pytest.mark.timeout(300)
def test_ddp_graphed_two_process_one_gpu(free_port):
    _run_two(_worker, free_port)


def _worker_resnet(rank, world, port, q):
    """ResNet-18 multi-bucket DDP + hipGraph step, 2 procs on one GPU: the
    multi-bucket world>1 branch of graphs.py/ddp.py (LeNet has a single
    bucket; the driver's SCALE run must not be the first multi-bucket run)."""
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)

        from dist_mnist_amd.models import ResNet18
        from dist_mnist_amd.ops import SGD
        from dist_mnist_amd.parallel import DistributedDataParallel
        from dist_mnist_amd.runtime import GraphedStep

        torch.manual_seed(11)
        net = ResNet18(num_classes=100).to("cuda:0").train()
        # 4 MB buckets over ~47 MB of fp32 grads -> genuinely multi-bucket
        ddp = DistributedDataParallel(net, bucket_bytes=4 * 2**20)
        assert len(ddp.buckets) > 4, f"expected multi-bucket, got {len(ddp.buckets)}"
        opt = SGD(ddp.parameters(), lr=0.01, momentum=0.5)

        g = torch.Generator().manual_seed(500 + rank)
        x = torch.zeros(4, 224, 224, 8, dtype=torch.bfloat16)
        x[..., :3] = torch.randn(4, 224, 224, 3, generator=g).to(torch.bfloat16)
        x = x.cuda()
        t = torch.randint(0, 100, (4,), generator=g).cuda()

        gs = GraphedStep(ddp, opt, x, t, ddp=ddp, warmup=2)
        losses = []
        for i in range(4):
            slot = i % 2
            gs.slots[slot][0].copy_(x)
            gs.slots[slot][1].copy_(t)
            losses.append(float(gs.step(slot)))
        torch.cuda.synchronize()
        assert all(v == v and v < 50 for v in losses), losses

        flat = torch.cat([p.flatten().float() for p in ddp.module.parameters()]).cpu()
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


@pytest.mark.timeout(600)
def test_ddp_graphed_resnet_multibucket_two_process(free_port):
    _run_two(_worker_resnet, free_port, timeout=400)


def _worker_capture_fallback(rank, world, port, q):
    """One rank fails hipGraph capture -> BOTH ranks must raise CaptureFailed
    (collective agreement precedes any constructor collective), and
    resync_after_fallback + eager stepping must leave ranks bitwise equal."""
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)

        from dist_mnist_amd.models import Net
        from dist_mnist_amd.ops import SGD
        from dist_mnist_amd.parallel import DistributedDataParallel
        from dist_mnist_amd.runtime import GraphedStep
        from dist_mnist_amd.runtime.graphs import CaptureFailed, resync_after_fallback

        torch.manual_seed(13)
        net = Net().to("cuda:0").train().bf16_native_()
        ddp = DistributedDataParallel(net)
        opt = SGD(ddp.parameters(), lr=0.05, momentum=0.5)

        g = torch.Generator().manual_seed(300 + rank)
        x = torch.randn(32, 28, 28, 1, generator=g).to(torch.bfloat16).cuda()
        t = torch.randint(0, 10, (32,), generator=g).cuda()

        if rank == 1:  # sabotage capture on ONE rank only
            orig = GraphedStep._capture

            def boom(self, warmup):
                orig(self, max(warmup - 1, 1))
                raise RuntimeError("injected capture failure")

            GraphedStep._capture = boom
        try:
            GraphedStep(ddp, opt, x, t, ddp=ddp)
            q.put((rank, "FAIL: CaptureFailed not raised"))
            return
        except CaptureFailed:
            pass  # expected on BOTH ranks

        resync_after_fallback(opt, ddp)
        ddp._install_hooks()
        for _ in range(3):
            ddp.zero_grad()
            _, loss = ddp(x, t)
            loss.backward()
            ddp.sync()
            opt.step()
        torch.cuda.synchronize()

        flat = torch.cat([p.flatten().float() for p in ddp.module.parameters()]).cpu()
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
def test_capture_failure_agreement_and_resync(free_port):
    _run_two(_worker_capture_fallback, free_port)


def _worker_native_comm(rank, world, port, q):
    """RcclComm.bootstrap dress rehearsal at world=2 on ONE GPU: the unique-id
    exchange, comm_set_timeout arming, and ncclCommInitRank ordering all
    execute for real. RCCL may refuse two ranks on one device (duplicate-GPU
    check) — a CLEAN RuntimeError within the timeout is a pass (the watchdog/
    fail-fast contract); an actual communicator is a pass with a verified
    all-reduce."""
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)

        from dist_mnist_amd.parallel import RcclComm

        try:
            comm = RcclComm.bootstrap(timeout_s=60.0)
        except RuntimeError as e:
            # duplicate-GPU refusal is the expected single-box outcome;
            # what matters is that it FAILED LOUDLY instead of hanging
            q.put((rank, "ok"))
            dist.barrier()
            return
        t = torch.full((1024,), float(rank + 1), device="cuda:0")
        comm.all_reduce_sum(t)
        torch.cuda.synchronize()
        expect = float(sum(range(1, world + 1)))
        assert torch.all(t == expect), t[:4]
        dist.barrier()
        comm.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {type(e).__name__}: {e}\n{traceback.format_exc()[-800:]}"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_comm_bootstrap_two_process_one_gpu(free_port):
    _run_two(_worker_native_comm, free_port)
