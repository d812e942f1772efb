"""bf16 NHWC engine numerics vs plain PyTorch fp32 references (GPU only).

Tolerances reflect bf16 storage of activations (rel ~1e-2) while weight-grad
reductions accumulate fp32.
"""

import pytest
import torch
import torch.nn.functional as F

from dist_mnist_amd.ops import functional_bf16 as fb
from dist_mnist_amd.ops import native

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def randn(*s, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*s, generator=g) * scale).to(DEV)


def test_mfma_probe_mapping():
    """Asymmetric A/B catch any transpose/mapping error (guide §3)."""
    a = randn(16, 32, seed=1)
    b = randn(32, 16, seed=2)
    d = native.ext().mfma_probe(a, b)
    ref = (a.to(torch.bfloat16).float() @ b.to(torch.bfloat16).float())
    torch.testing.assert_close(d, ref, rtol=1e-3, atol=1e-3)


@pytest.mark.parametrize(
    "N,CI,H,W,CO,KH,KW,stride,pad",
    [
        (4, 1, 28, 28, 10, 5, 5, 1, 0),   # conv1
        (4, 10, 12, 12, 20, 5, 5, 1, 0),  # conv2
        (2, 3, 18, 18, 7, 3, 3, 1, 1),    # pad
        (2, 8, 16, 16, 16, 3, 3, 2, 1),   # stride 2
        (2, 3, 32, 32, 8, 7, 7, 2, 3),    # 7x7 s2 (gather engine: COp<64)
        (2, 3, 37, 37, 64, 7, 7, 2, 3),   # 7x7 s2 stem engine, ragged tiles
        (4, 5, 41, 37, 128, 7, 7, 2, 3),  # stem engine, 2 co-tiles, non-square
        (2, 3, 224, 224, 64, 7, 7, 2, 3), # the actual ResNet stem shape
        (2, 64, 27, 27, 128, 3, 3, 2, 1), # 3x3 s2 tiled engine (layer2 shape)
        (2, 33, 29, 25, 64, 3, 3, 2, 1),  # s2 tiled, ragged ci chunk, non-sq
        (2, 256, 14, 14, 512, 3, 3, 2, 1),# s2 tiled two-image (layer4 shape)
        (4, 64, 15, 15, 64, 3, 3, 2, 1),  # s2 tiled NIMG2, ragged rows
        (131, 1, 28, 28, 10, 5, 5, 1, 0), # tile remainder
    ],
)
def test_conv2d_bf16_fwd(N, CI, H, W, CO, KH, KW, stride, pad):
    x = randn(N, CI, H, W, seed=3)
    w = randn(CO, CI, KH, KW, seed=4, scale=0.3)
    b = randn(CO, seed=5)
    xb = fb.nchw_to_nhwc_padded(x)
    y = fb.conv2d_bf16(xb, w, b, stride=stride, pad=pad)
    cop = fb.pad8(CO)
    assert y.shape == (N, (H + 2 * pad - KH) // stride + 1,
                       (W + 2 * pad - KW) // stride + 1, cop)
    # oracle sees the same bf16-quantized x and w the kernel stages
    q = lambda t: t.to(torch.bfloat16).float()
    ref = F.conv2d(q(x), q(w), b, stride=stride, padding=pad)
    got = fb.nhwc_to_nchw(y, CO)
    torch.testing.assert_close(got, ref, rtol=2e-2, atol=3e-2)
    # padded channels must be exactly zero
    assert (y[..., CO:].float() == 0).all()


@pytest.mark.parametrize(
    "N,CI,H,W,CO,KH,KW,stride,pad",
    [
        (4, 10, 12, 12, 20, 5, 5, 1, 0),
        (2, 3, 18, 18, 7, 3, 3, 1, 1),
        (2, 8, 16, 16, 16, 3, 3, 2, 1),
        (2, 64, 27, 27, 128, 3, 3, 2, 1),  # tiled s2 dgrad (3x3 pad1)
        (2, 64, 28, 28, 128, 3, 3, 2, 1),  # H==2*OH: region wgrad S=2 path
        (2, 64, 15, 15, 128, 3, 3, 2, 1),  # tiled s2 dgrad, odd H (ragged classes)
        (2, 64, 14, 14, 128, 1, 1, 2, 0),  # tiled s2 dgrad (1x1 downsample)
        (2, 3, 192, 192, 64, 7, 7, 2, 3),  # stem region wgrad (7x7 s2)
        (2, 64, 28, 28, 64, 3, 3, 1, 1),   # region wgrad (layer1-ish)
        (3, 96, 14, 14, 128, 3, 3, 1, 1),  # region wgrad, 3 ci-tiles, odd N
        (2, 64, 7, 7, 64, 3, 3, 1, 1),     # region wgrad, chunks span images
    ],
)
def test_conv2d_bf16_bwd(N, CI, H, W, CO, KH, KW, stride, pad):
    OH = (H + 2 * pad - KH) // stride + 1
    OW = (W + 2 * pad - KW) // stride + 1
    x = randn(N, CI, H, W, seed=6)
    w = randn(CO, CI, KH, KW, seed=7, scale=0.3).requires_grad_(True)
    b = randn(CO, seed=8).requires_grad_(True)
    dy = randn(N, CO, OH, OW, seed=9)

    xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
    y = fb.conv2d_bf16(xb, w, b, stride=stride, pad=pad)
    dyb = fb.nchw_to_nhwc_padded(dy, cp=y.shape[3])
    y.backward(dyb)

    q = lambda t: t.to(torch.bfloat16).float()
    xr = q(x.detach()).requires_grad_(True)
    wr = q(w.detach()).requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    F.conv2d(xr, wr, br, stride=stride, padding=pad).backward(q(dy))

    got_dx = fb.nhwc_to_nchw(xb.grad, CI)
    torch.testing.assert_close(got_dx, xr.grad, rtol=3e-2, atol=4e-2)
    # weight grads are fp32-accumulated over bf16 products
    torch.testing.assert_close(w.grad, wr.grad, rtol=2e-2,
                               atol=2e-2 * max(1.0, wr.grad.abs().max().item()))
    torch.testing.assert_close(b.grad, br.grad, rtol=2e-2,
                               atol=2e-2 * max(1.0, br.grad.abs().max().item()))


def test_maxpool_relu_bf16():
    x = randn(3, 10, 12, 16, seed=10)
    xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
    y = fb.maxpool2x2_relu_bf16(xb)
    dy = randn(3, 10, 6, 8, seed=11)
    dyb = fb.nchw_to_nhwc_padded(dy, cp=16)
    y.backward(dyb)

    xr = x.detach().to(torch.bfloat16).float().requires_grad_(True)
    out = F.relu(F.max_pool2d(xr, 2))
    out.backward(dy.to(torch.bfloat16).float())
    torch.testing.assert_close(fb.nhwc_to_nchw(y, 10), out, rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(fb.nhwc_to_nchw(xb.grad, 10), xr.grad,
                               rtol=1e-2, atol=1e-2)


@pytest.mark.parametrize("M,CI,CO,relu", [(64, 320, 50, True), (64, 50, 10, False),
                                          (257, 320, 50, True)])
def test_linear_bf16(M, CI, CO, relu):
    cik = fb.pad8(CI)
    cop = fb.pad8(CO)
    x = randn(M, CI, seed=12)
    w = randn(CO, CI, seed=13, scale=0.1).requires_grad_(True)
    b = randn(CO, seed=14).requires_grad_(True)
    dy = randn(M, CO, seed=15)

    xb = torch.zeros(M, cik, dtype=torch.bfloat16, device=DEV)
    xb[:, :CI] = x.to(torch.bfloat16)
    xb.requires_grad_(True)
    y = fb.linear_bf16(xb, w, b, cop=cop, relu=relu)
    dyb = torch.zeros(M, cop, dtype=torch.bfloat16, device=DEV)
    dyb[:, :CO] = dy.to(torch.bfloat16)
    y.backward(dyb)

    xr = xb.detach()[:, :CI].float().requires_grad_(True)
    wr = w.detach().to(torch.bfloat16).float().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref = F.linear(xr, wr, br)
    if relu:
        ref = F.relu(ref)
    ref.backward(dyb.detach()[:, :CO].float())

    torch.testing.assert_close(y[:, :CO].float(), ref, rtol=2e-2, atol=3e-2)
    torch.testing.assert_close(xb.grad[:, :CI].float(), xr.grad, rtol=3e-2, atol=4e-2)
    torch.testing.assert_close(w.grad, wr.grad, rtol=2e-2,
                               atol=2e-2 * max(1.0, wr.grad.abs().max().item()))
    torch.testing.assert_close(b.grad, br.grad, rtol=2e-2,
                               atol=2e-2 * max(1.0, br.grad.abs().max().item()))


def test_loss_bf16():
    M = 130
    x = randn(M, 10, seed=16)
    xb = torch.zeros(M, 16, dtype=torch.bfloat16, device=DEV)
    xb[:, :10] = x.to(torch.bfloat16)
    xb.requires_grad_(True)
    t = torch.randint(0, 10, (M,), device=DEV)
    logp, loss = fb.logsoftmax_nll_bf16(xb, t, c=10)
    loss.backward()

    xr = xb.detach()[:, :10].float().requires_grad_(True)
    lr = F.log_softmax(xr, dim=1)
    F.nll_loss(lr, t).backward()
    torch.testing.assert_close(logp, lr, rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(xb.grad[:, :10].float(), xr.grad, rtol=1e-2, atol=1e-3)
    assert (xb.grad[:, 10:].float() == 0).all()


def test_dropout_bf16():
    x = torch.ones(100000 * 8, dtype=torch.bfloat16, device=DEV).requires_grad_(True)
    y = fb.dropout_bf16(x, p=0.5, training=True)
    keep = y.float() != 0
    frac = keep.float().mean().item()
    assert 0.47 < frac < 0.53, frac
    y.backward(torch.ones_like(y))
    assert (x.grad.float()[keep] == 2.0).all()
    assert (x.grad.float()[~keep] == 0.0).all()


def test_dropout2d_bf16_channelwise():
    x = torch.ones(64, 8, 8, 24, dtype=torch.bfloat16, device=DEV)
    y = fb.dropout_bf16(x, p=0.5, training=True, channelwise=True)
    yc = y.float().permute(0, 3, 1, 2).reshape(64 * 24, 64)
    zero = (yc == 0).all(dim=1)
    live = (yc == 2.0).all(dim=1)
    assert (zero | live).all()
    assert 0.4 < live.float().mean().item() < 0.6


def test_dropout2d_bf16_backward_matches_mask():
    """Backward cross-check vs the torch fp32 semantics: dx = dy * mask / (1-p)
    with the SAME per-(sample,channel) mask the forward drew."""
    x = torch.randn(32, 8, 8, 16, device=DEV).to(torch.bfloat16).requires_grad_(True)
    y = fb.dropout_bf16(x, p=0.5, training=True, channelwise=True)
    live = (y.float() != 0).any(dim=1, keepdim=True).any(dim=2, keepdim=True)
    dy = torch.randn_like(y.float()).to(torch.bfloat16)
    y.backward(dy)
    # torch reference of the same op given the drawn mask
    ref = dy.float() * live.float() * 2.0
    torch.testing.assert_close(x.grad.float(), ref.to(torch.bfloat16).float(),
                               rtol=1e-2, atol=1e-3)
    # forward consistency: y == x * mask * 2 for the same mask
    torch.testing.assert_close(
        y.float(), (x.detach().float() * live.float() * 2.0)
        .to(torch.bfloat16).float(), rtol=1e-2, atol=1e-3)


def test_net_bf16_matches_fp32_reference():
    from dist_mnist_amd.models import Net

    torch.manual_seed(30)
    net = Net().to(DEV).eval().bf16_native_()
    x = randn(32, 1, 28, 28, seed=31)
    t = torch.randint(0, 10, (32,), device=DEV)
    xb = fb.nchw_to_nhwc_padded(x)
    logp, loss = net(xb, t)

    ref = Net().to(DEV).eval()
    ref.load_state_dict(net.state_dict())
    logp_ref, loss_ref = ref(x, t)
    torch.testing.assert_close(loss, loss_ref, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(logp, logp_ref, rtol=5e-2, atol=8e-2)


def test_net_bf16_grads_match_fp32():
    from dist_mnist_amd.models import Net

    torch.manual_seed(32)
    net = Net().to(DEV).eval().bf16_native_()
    ref = Net().to(DEV).eval()
    ref.load_state_dict(net.state_dict())
    x = randn(64, 1, 28, 28, seed=33)
    t = torch.randint(0, 10, (64,), device=DEV)

    _, loss = net(fb.nchw_to_nhwc_padded(x), t)
    loss.backward()
    _, loss_ref = ref(x, t)
    loss_ref.backward()

    rsd = dict(ref.named_parameters())
    for name, p in net.named_parameters():
        r = rsd[name].grad
        # the bf16 path quantizes every activation; compare direction and
        # magnitude via relative RMS rather than elementwise bounds. The
        # earliest layer sees 5 backprop quantization stages (the per-op
        # tests in this file pin each kernel against a quantized oracle),
        # so its threshold is looser.
        lim = 0.20 if name.startswith("conv1") else 0.08
        rel_rms = ((p.grad - r).norm() / (r.norm() + 1e-12)).item()
        cos = torch.nn.functional.cosine_similarity(
            p.grad.flatten(), r.flatten(), dim=0).item()
        assert rel_rms < lim, f"bf16 grad rel-RMS {rel_rms:.4f} for {name}"
        assert cos > 0.97, f"bf16 grad cosine {cos:.4f} for {name}"


def test_training_loss_decreases_bf16():
    from dist_mnist_amd.models import Net
    from dist_mnist_amd.ops import SGD

    torch.manual_seed(34)
    net = Net().to(DEV).eval().bf16_native_()
    opt = SGD(net.parameters(), lr=0.05, momentum=0.5)
    x = fb.nchw_to_nhwc_padded(randn(64, 1, 28, 28, seed=35))
    t = torch.randint(0, 10, (64,), device=DEV)
    losses = []
    for _ in range(150):
        opt.zero_grad()
        _, loss = net(x, t)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    tail = sum(losses[-10:]) / 10
    assert tail < losses[0] * 0.75, (losses[0], tail, losses[::30])


def test_conv_pool_stem_fused():
    """Fused conv1+pool+relu vs torch oracle, fwd + wgrad/bias grads."""
    N, CI, H, W, CO = 5, 1, 28, 28, 10
    x = randn(N, CI, H, W, seed=40)
    w = randn(CO, CI, 5, 5, seed=41, scale=0.3).requires_grad_(True)
    b = randn(CO, seed=42).requires_grad_(True)
    xb = fb.nchw_to_nhwc_padded(x)
    p = fb.conv_pool_stem_bf16(xb, w, b, cop=16)
    assert p.shape == (N, 12, 12, 16)

    q = lambda t: t.to(torch.bfloat16).float()
    wr = q(w.detach()).requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref = F.relu(F.max_pool2d(F.conv2d(q(x), wr, br), 2))
    torch.testing.assert_close(fb.nhwc_to_nchw(p, CO), ref, rtol=2e-2, atol=3e-2)
    assert (p[..., CO:].float() == 0).all()

    dy = randn(N, CO, 12, 12, seed=43)
    dyb = fb.nchw_to_nhwc_padded(dy, cp=16)
    p.backward(dyb)

    # oracle for the backward must follow the KERNEL's argmax routing (a
    # near-tie in the fp32 conv can legitimately route the subgradient to a
    # different quadrant than torch's own pool): reconstruct dy_conv from the
    # kernel's (p, idx) and compute the weight grad with torch from there.
    pk, idxk = native.ext().conv_pool_stem_bf16(xb, w.detach(), b.detach(), 16)
    gate = (pk.float() > 0).to(torch.bfloat16)
    dyq = (dyb * gate).float()
    dy_conv = torch.zeros(N, 24, 24, 16, device=DEV)
    for qd in range(4):
        oh_off, ow_off = qd // 2, qd % 2
        sel = (idxk == qd).float() * dyq
        dy_conv[:, oh_off::2, ow_off::2, :] = sel
    dy_conv_nchw = dy_conv[..., :CO].permute(0, 3, 1, 2)
    dw_ref = torch.nn.grad.conv2d_weight(
        q(x), w.shape, q(dy_conv_nchw))
    db_ref = q(dy_conv_nchw).sum(dim=(0, 2, 3))
    torch.testing.assert_close(w.grad, dw_ref, rtol=2e-2,
                               atol=2e-2 * max(1.0, dw_ref.abs().max().item()))
    torch.testing.assert_close(b.grad, db_ref, rtol=2e-2,
                               atol=2e-2 * max(1.0, db_ref.abs().max().item()))


def test_graphed_step_replay():
    """hipGraph-captured step: replays train, inputs swap per slot, dropout
    masks stay fresh across replays (device-side counter)."""
    from dist_mnist_amd.models import Net
    from dist_mnist_amd.ops import SGD
    from dist_mnist_amd.runtime import GraphedStep

    torch.manual_seed(60)
    model = Net().to(DEV).train().bf16_native_()
    opt = SGD(model.parameters(), lr=0.05, momentum=0.5)
    x0 = torch.zeros(64, 28, 28, 1, dtype=torch.bfloat16, device=DEV)
    x0[..., 0] = randn(64, 28, 28, seed=61).to(torch.bfloat16)
    t0 = torch.randint(0, 10, (64,), device=DEV)

    gs = GraphedStep(model, opt, x0, t0)
    p0 = model.fc2.weight.clone()
    losses = []
    for i in range(20):
        slot = i % 2
        gs.slots[slot][0].copy_(x0)
        gs.slots[slot][1].copy_(t0)
        loss = gs.step(slot)
        losses.append(float(loss))
    torch.cuda.synchronize()
    # params actually moved and loss went down on the fixed batch
    assert not torch.equal(p0, model.fc2.weight)
    assert losses[-1] < losses[0], losses
    # dropout freshness: same input, training mode => losses vary replay to
    # replay (a frozen mask would give a monotone deterministic sequence;
    # check successive deltas are not all identical signs/magnitudes)
    diffs = {round(losses[i + 1] - losses[i], 6) for i in range(10, 19)}
    assert len(diffs) > 1


def test_conv_pool_stem_unpadded_input():
    """REGRESSION (bench path): the stem takes CIp=1 input ([N,28,28,1] —
    what the loader actually ships); its wgrad must handle KW*CIp=5 rows
    that are not 8-aligned (row-padded k-space + clamped loads)."""
    N, CO = 5, 10
    x = randn(N, 1, 28, 28, seed=70)
    w = randn(CO, 1, 5, 5, seed=71, scale=0.3).requires_grad_(True)
    b = randn(CO, seed=72).requires_grad_(True)
    xb = x[:, 0].to(torch.bfloat16).unsqueeze(-1).contiguous()  # [N,28,28,1]
    p = fb.conv_pool_stem_bf16(xb, w, b, cop=16)

    q = lambda t: t.to(torch.bfloat16).float()
    ref_fwd = F.relu(F.max_pool2d(F.conv2d(q(x), q(w.detach()), b.detach()), 2))
    torch.testing.assert_close(fb.nhwc_to_nchw(p, CO), ref_fwd, rtol=2e-2, atol=3e-2)

    dy = randn(N, CO, 12, 12, seed=73)
    p.backward(fb.nchw_to_nhwc_padded(dy, cp=16))

    # oracle follows the kernel's argmax routing (see test_conv_pool_stem_fused)
    pk, idxk = native.ext().conv_pool_stem_bf16(xb, w.detach(), b.detach(), 16)
    gate = (pk.float() > 0).to(torch.bfloat16)
    dyq = (fb.nchw_to_nhwc_padded(dy, cp=16) * gate).float()
    dy_conv = torch.zeros(N, 24, 24, 16, device=DEV)
    for qd in range(4):
        sel = (idxk == qd).float() * dyq
        dy_conv[:, qd // 2::2, qd % 2::2, :] = sel
    dy_conv_nchw = dy_conv[..., :CO].permute(0, 3, 1, 2)
    dw_ref = torch.nn.grad.conv2d_weight(q(x), w.shape, q(dy_conv_nchw))
    db_ref = q(dy_conv_nchw).sum(dim=(0, 2, 3))
    torch.testing.assert_close(w.grad, dw_ref, rtol=2e-2,
                               atol=2e-2 * max(1.0, dw_ref.abs().max().item()))
    torch.testing.assert_close(b.grad, db_ref, rtol=2e-2,
                               atol=2e-2 * max(1.0, db_ref.abs().max().item()))
