"""HIP kernel numerics vs plain PyTorch fp32 references (SURVEY.md §4 Unit).
All tests need a real MI355X -> @pytest.mark.gpu (run via gpurun)."""

import pytest
import torch
import torch.nn.functional as F

from dist_mnist_amd import ops
from dist_mnist_amd.ops import native

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def rand(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(DEV)


def test_native_extension_is_loaded():
    # the HIP path must be the one that runs on a GPU box — loud if missing
    assert native.available(), "native _C.so missing on GPU box"


@pytest.mark.parametrize(
    "N,CI,H,W,CO,KH,KW,stride,pad",
    [
        (4, 1, 28, 28, 10, 5, 5, 1, 0),   # LeNet conv1
        (4, 10, 12, 12, 20, 5, 5, 1, 0),  # LeNet conv2
        (2, 3, 17, 19, 7, 3, 3, 1, 1),    # odd shapes + padding
        (2, 8, 16, 16, 16, 3, 3, 2, 1),   # stride 2 (ResNet-style)
        (2, 3, 32, 32, 8, 7, 7, 2, 3),    # 7x7 s2 (ResNet stem)
        (2, 16, 8, 8, 32, 1, 1, 1, 0),    # 1x1
        (129, 1, 28, 28, 10, 5, 5, 1, 0), # M not divisible by tile
    ],
)
def test_conv2d_fwd(N, CI, H, W, CO, KH, KW, stride, pad):
    x = rand(N, CI, H, W, seed=1)
    w = rand(CO, CI, KH, KW, seed=2, scale=0.3)
    b = rand(CO, seed=3)
    y = native.ext().conv2d_fwd(x, w, b, stride, pad)
    ref = F.conv2d(x, w, b, stride=stride, padding=pad)
    torch.testing.assert_close(y, ref, rtol=2e-4, atol=2e-4)


@pytest.mark.parametrize(
    "N,CI,H,W,CO,KH,KW,stride,pad",
    [
        (4, 10, 12, 12, 20, 5, 5, 1, 0),
        (2, 3, 17, 19, 7, 3, 3, 1, 1),
        (2, 8, 16, 16, 16, 3, 3, 2, 1),
        (2, 3, 32, 32, 8, 7, 7, 2, 3),
    ],
)
def test_conv2d_bwd(N, CI, H, W, CO, KH, KW, stride, pad):
    x = rand(N, CI, H, W, seed=4).requires_grad_(True)
    w = rand(CO, CI, KH, KW, seed=5, scale=0.3).requires_grad_(True)
    b = rand(CO, seed=6).requires_grad_(True)
    dy = rand(N, CO, (H + 2 * pad - KH) // stride + 1, (W + 2 * pad - KW) // stride + 1, seed=7)

    y = ops.conv2d(x, w, b, stride=stride, padding=pad)
    y.backward(dy)

    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    F.conv2d(xr, wr, br, stride=stride, padding=pad).backward(dy)

    torch.testing.assert_close(x.grad, xr.grad, rtol=2e-4, atol=2e-4)
    torch.testing.assert_close(w.grad, wr.grad, rtol=2e-4, atol=3e-3)
    torch.testing.assert_close(b.grad, br.grad, rtol=2e-4, atol=3e-3)


def test_maxpool2x2_relu():
    x = rand(3, 5, 12, 16, seed=8).requires_grad_(True)
    dy = rand(3, 5, 6, 8, seed=9)
    y = ops.maxpool2x2_relu(x)
    y.backward(dy)

    xr = x.detach().clone().requires_grad_(True)
    F.relu(F.max_pool2d(xr, 2)).backward(dy)
    torch.testing.assert_close(y, F.relu(F.max_pool2d(x.detach(), 2)))
    torch.testing.assert_close(x.grad, xr.grad)


@pytest.mark.parametrize("M,CI,CO,relu", [
    (64, 320, 50, True),   # fc1
    (64, 50, 10, False),   # fc2
    (257, 320, 50, True),  # non-divisible M
    (128, 100, 130, True), # CO > 2 tiles
])
def test_linear(M, CI, CO, relu):
    x = rand(M, CI, seed=10).requires_grad_(True)
    w = rand(CO, CI, seed=11, scale=0.1).requires_grad_(True)
    b = rand(CO, seed=12).requires_grad_(True)
    dy = rand(M, CO, seed=13)

    y = ops.linear(x, w, b, act_relu=relu)
    y.backward(dy)

    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref = F.linear(xr, wr, br)
    if relu:
        ref = F.relu(ref)
    ref.backward(dy)

    torch.testing.assert_close(y, ref, rtol=2e-4, atol=2e-4)
    torch.testing.assert_close(x.grad, xr.grad, rtol=2e-4, atol=2e-4)
    torch.testing.assert_close(w.grad, wr.grad, rtol=2e-4, atol=2e-3)
    torch.testing.assert_close(b.grad, br.grad, rtol=2e-4, atol=2e-3)


def test_logsoftmax_nll():
    x = rand(130, 10, seed=14).requires_grad_(True)
    t = torch.randint(0, 10, (130,), device=DEV)
    logp, loss = ops.logsoftmax_nll(x, t)
    loss.backward()

    xr = x.detach().clone().requires_grad_(True)
    lr = F.log_softmax(xr, dim=1)
    F.nll_loss(lr, t).backward()
    torch.testing.assert_close(logp, lr, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(loss, F.nll_loss(F.log_softmax(x.detach(), 1), t),
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-5, atol=1e-6)


def test_logsoftmax_standalone():
    x = rand(64, 10, seed=15).requires_grad_(True)
    dy = rand(64, 10, seed=16)
    y = ops.log_softmax(x)
    y.backward(dy)
    xr = x.detach().clone().requires_grad_(True)
    F.log_softmax(xr, dim=1).backward(dy)
    torch.testing.assert_close(y, F.log_softmax(x.detach(), 1), rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-5, atol=1e-6)


def test_relu_standalone():
    x = rand(1000, seed=17).requires_grad_(True)
    dy = rand(1000, seed=18)
    y = ops.relu(x)
    y.backward(dy)
    torch.testing.assert_close(y, F.relu(x.detach()))
    torch.testing.assert_close(x.grad, dy * (x.detach() > 0))


def test_dropout_elementwise():
    x = torch.ones(100000, device=DEV).requires_grad_(True)
    y = ops.dropout(x, p=0.5, training=True)
    keep = (y != 0)
    frac = keep.float().mean().item()
    assert 0.47 < frac < 0.53, frac
    torch.testing.assert_close(y[keep], torch.full_like(y[keep], 2.0))  # 1/(1-p)
    y.backward(torch.ones_like(y))
    torch.testing.assert_close(x.grad[keep], torch.full_like(y[keep], 2.0))
    torch.testing.assert_close(x.grad[~keep], torch.zeros_like(y[~keep]))


def test_dropout2d_channelwise():
    x = torch.ones(64, 20, 8, 8, device=DEV)
    y = ops.dropout(x, p=0.5, training=True, channelwise=True)
    per_chan = y.reshape(64 * 20, 64)
    zero_chan = (per_chan == 0).all(dim=1)
    live_chan = (per_chan == 2.0).all(dim=1)
    assert (zero_chan | live_chan).all(), "Dropout2d must zero whole channels"
    frac = live_chan.float().mean().item()
    assert 0.4 < frac < 0.6, frac


def test_fused_sgd_matches_torch():
    from dist_mnist_amd.ops import SGD

    torch.manual_seed(19)
    shapes = [(10, 1, 5, 5), (10,), (20, 10, 5, 5), (50, 320), (50,), (10, 50)]
    ours = [torch.randn(s, device=DEV) for s in shapes]
    ref = [p.clone() for p in ours]
    ours_p = [torch.nn.Parameter(p) for p in ours]
    ref_p = [torch.nn.Parameter(p) for p in ref]
    opt = SGD(ours_p, lr=0.05, momentum=0.5)
    topt = torch.optim.SGD(ref_p, lr=0.05, momentum=0.5)
    for step in range(4):
        g = [torch.randn_like(p) for p in ours]
        for p, gp in zip(ours_p, g):
            p.grad = gp.clone()
        for p, gp in zip(ref_p, g):
            p.grad = gp.clone()
        opt.step()
        topt.step()
    for a, b in zip(ours_p, ref_p):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-7)


def test_net_forward_backward_matches_torch_gpu():
    from dist_mnist_amd.models import Net
    from tests.test_model_cpu import TorchNet

    torch.manual_seed(20)
    net = Net().to(DEV).eval()
    oracle = TorchNet().to(DEV).eval()
    oracle.load_state_dict(net.state_dict())
    x = rand(16, 1, 28, 28, seed=21)
    t = torch.randint(0, 10, (16,), device=DEV)

    logp, loss = net(x, t)
    loss.backward()
    F.nll_loss(oracle(x), t).backward()

    torch.testing.assert_close(logp, oracle(x), rtol=1e-4, atol=1e-4)
    osd = dict(oracle.named_parameters())
    for name, p in net.named_parameters():
        torch.testing.assert_close(p.grad, osd[name].grad, rtol=1e-3, atol=1e-4,
                                   msg=f"grad mismatch {name}")


def test_training_loss_decreases_gpu():
    from dist_mnist_amd.models import Net
    from dist_mnist_amd.ops import SGD

    torch.manual_seed(22)
    net = Net().to(DEV).eval()  # dropout off: deterministic memorization
    opt = SGD(net.parameters(), lr=0.05, momentum=0.5)
    x = rand(64, 1, 28, 28, seed=23)
    t = torch.randint(0, 10, (64,), device=DEV)
    losses = []
    for _ in range(150):
        opt.zero_grad()
        _, loss = net(x, t)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    tail = sum(losses[-10:]) / 10
    assert tail < losses[0] * 0.7, (losses[0], tail, losses[::30])


@pytest.mark.timeout(120)
def test_native_rccl_comm_single_rank(free_port):
    """C++ RCCL comm core plumbing: bootstrap over a 1-rank gloo group,
    allreduce on device (identity at world=1), destroy."""
    import os

    import torch.distributed as dist

    from dist_mnist_amd.parallel import RcclComm

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(free_port)
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        comm = RcclComm.bootstrap()
        t = torch.arange(1024, dtype=torch.float32, device=DEV)
        ref = t.clone()
        comm.all_reduce_sum(t)
        torch.cuda.synchronize()
        torch.testing.assert_close(t, ref)
        tb = torch.randn(64, dtype=torch.bfloat16, device=DEV)
        refb = tb.clone()
        comm.all_reduce_sum(tb)
        torch.cuda.synchronize()
        torch.testing.assert_close(tb, refb)
        comm.destroy()
    finally:
        dist.destroy_process_group()


def test_conv_pool_stem_f32_matches_torch():
    """Fused fp32 stem (conv1+pool+relu) fwd + its fused wgrad vs torch."""
    torch.manual_seed(91)
    x = rand(16, 1, 28, 28, seed=92)
    w = rand(10, 1, 5, 5, seed=93, scale=0.3).requires_grad_(True)
    b = rand(10, seed=94).requires_grad_(True)

    p = ops.conv_pool_stem(x, w, b)
    dp = rand(16, 10, 12, 12, seed=95)
    p.backward(dp)

    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref = F.relu(F.max_pool2d(F.conv2d(x, wr, br), 2))
    ref.backward(dp)

    torch.testing.assert_close(p, ref, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(b.grad, br.grad, rtol=1e-4, atol=1e-4)


def test_net_f32_gpu_matches_cpu():
    """The fp32 GPU path (incl. the fused stem) reproduces the CPU torch
    reference end to end: loss + grads of the whole Net."""
    from dist_mnist_amd.models import Net

    torch.manual_seed(96)
    net = Net().to(DEV).eval()
    ref = Net().eval()
    ref.load_state_dict({k: v.cpu() for k, v in net.state_dict().items()})
    g = torch.Generator().manual_seed(97)
    x = torch.randn(32, 1, 28, 28, generator=g)
    t = torch.randint(0, 10, (32,), generator=g)

    _, loss = net(x.to(DEV), t.to(DEV))
    loss.backward()
    _, loss_ref = ref(x, t)
    loss_ref.backward()

    torch.testing.assert_close(loss.cpu(), loss_ref, rtol=1e-4, atol=1e-4)
    rsd = dict(ref.named_parameters())
    for name, pm in net.named_parameters():
        torch.testing.assert_close(pm.grad.cpu(), rsd[name].grad,
                                   rtol=2e-3, atol=2e-4,
                                   msg=f"f32 grad mismatch: {name}")
