"""Long-horizon training soak (GPU): thousands of real steps through the
graphed bench path; verifies loss convergence + finite params at the end.
Artifact for profiles/: loss curve summary."""
import sys, time
import torch
sys.path.insert(0, ".")
from dist_mnist_amd.data import SyntheticBatchStream
from dist_mnist_amd.models import Net, ResNet18
from dist_mnist_amd.ops import SGD
from dist_mnist_amd.runtime import GraphedStep, Prefetcher

def lenet_soak(steps=3000):
    torch.manual_seed(1234)
    net = Net().to("cuda:0").train().bf16_native_()
    opt = SGD(net.parameters(), lr=0.05, momentum=0.9)
    # ONE repeated 256-sample batch: random labels are memorizable at this
    # size, so the loss must fall well below ln(10) if training is correct
    stream = SyntheticBatchStream(batch_size=256, n_pool=1, seed=7,
                                  pin=True, layout="nhwc_bf16")
    pf = Prefetcher(stream, "cuda:0", dtype=torch.bfloat16)
    x0, t0 = next(pf)
    gs = GraphedStep(net, opt, x0, t0)
    losses = []
    t0s = time.perf_counter()
    for i in range(steps):
        slot = i % 2
        x, t = next(pf)
        gs.slots[slot][0].copy_(x, non_blocking=True)
        gs.slots[slot][1].copy_(t, non_blocking=True)
        loss = gs.step(slot)
        if i % 100 == 0:
            losses.append((i, float(loss)))
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0s
    finite = all(torch.isfinite(p).all().item() for p in net.parameters())
    print(f"LeNet soak: {steps} steps, {dt:.1f}s ({dt/steps*1e3:.3f} ms/step), "
          f"params finite={finite}")
    print("  loss curve:", " ".join(f"{i}:{l:.3f}" for i, l in losses[::3]))
    assert finite
    assert losses[-1][1] < 0.7, (losses[0], losses[-1])

def resnet_soak(steps=400):
    torch.manual_seed(1234)
    net = ResNet18(num_classes=100).to("cuda:0").train()
    opt = SGD(net.parameters(), lr=0.05, momentum=0.9)
    g = torch.Generator().manual_seed(3)
    x = torch.zeros(64, 224, 224, 8, dtype=torch.bfloat16)
    x[..., :3] = torch.randn(64, 224, 224, 3, generator=g).to(torch.bfloat16)
    x = x.cuda()
    t = torch.randint(0, 100, (64,), generator=g).cuda()
    gs = GraphedStep(net, opt, x, t)
    losses = []
    for i in range(steps):
        slot = i % 2
        gs.slots[slot][0].copy_(x)
        gs.slots[slot][1].copy_(t)
        loss = gs.step(slot)
        if i % 25 == 0:
            losses.append((i, float(loss)))
    torch.cuda.synchronize()
    finite = all(torch.isfinite(p).all().item() for p in net.parameters())
    print(f"ResNet soak (fixed batch, memorization): {steps} steps, "
          f"finite={finite}")
    print("  loss curve:", " ".join(f"{i}:{l:.3f}" for i, l in losses))
    assert finite
    assert losses[-1][1] < losses[0][1] * 0.2, (losses[0], losses[-1])

if __name__ == "__main__":
    lenet_soak()
    resnet_soak()
    print("SOAK OK")
