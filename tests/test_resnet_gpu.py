"""ResNet-18 bf16 engine tests (GPU): op numerics vs torch fp32 references on
bf16-quantized inputs, block/model forward+backward sanity, training step."""

import pytest
import torch
import torch.nn.functional as F

from dist_mnist_amd.ops import functional_bf16 as fb
from dist_mnist_amd.ops import functional_resnet as fr

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def randn(*s, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*s, generator=g) * scale).to(DEV)


def test_batchnorm_bf16_fwd_bwd():
    N, C, H, W = 4, 16, 10, 12
    x = randn(N, C, H, W, seed=1)
    gamma = randn(C, seed=2).abs().add(0.5).requires_grad_(True)
    beta = randn(C, seed=3).requires_grad_(True)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)

    xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
    y = fr.batchnorm_bf16(xb, gamma, beta, rm, rv, training=True)
    dy = randn(N, C, H, W, seed=4)
    y.backward(fb.nchw_to_nhwc_padded(dy, cp=16))

    q = lambda t: t.to(torch.bfloat16).float()
    xr = q(x).requires_grad_(True)
    gr = gamma.detach().clone().requires_grad_(True)
    br = beta.detach().clone().requires_grad_(True)
    rm2 = torch.zeros(C, device=DEV)
    rv2 = torch.ones(C, device=DEV)
    ref = F.batch_norm(xr, rm2, rv2, gr, br, training=True, momentum=0.1, eps=1e-5)
    ref.backward(q(dy))

    torch.testing.assert_close(fb.nhwc_to_nchw(y, C), ref, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(rm, rm2, rtol=1e-2, atol=1e-3)
    torch.testing.assert_close(rv, rv2, rtol=1e-2, atol=1e-3)
    torch.testing.assert_close(fb.nhwc_to_nchw(xb.grad, C), xr.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(gamma.grad, gr.grad, rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(beta.grad, br.grad, rtol=2e-2, atol=2e-2)


def test_batchnorm_relu_fused():
    N, C, H, W = 2, 8, 6, 6
    x = randn(N, C, H, W, seed=5)
    gamma = torch.ones(C, device=DEV, requires_grad=True)
    beta = torch.zeros(C, device=DEV, requires_grad=True)
    rm, rv = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)
    xb = fb.nchw_to_nhwc_padded(x)
    y = fr.batchnorm_bf16(xb, gamma, beta, rm, rv, training=True, relu=True)
    assert (y.float() >= 0).all()


def test_maxpool3x3s2():
    x = randn(2, 8, 16, 16, seed=6)
    xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
    y = fr.maxpool3x3s2_bf16(xb)
    dy = randn(2, 8, 8, 8, seed=7)
    y.backward(fb.nchw_to_nhwc_padded(dy, cp=8))

    xr = x.detach().to(torch.bfloat16).float().requires_grad_(True)
    ref = F.max_pool2d(xr, 3, stride=2, padding=1)
    ref.backward(dy.to(torch.bfloat16).float())
    torch.testing.assert_close(fb.nhwc_to_nchw(y, 8), ref, rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(fb.nhwc_to_nchw(xb.grad, 8), xr.grad, rtol=1e-2, atol=1e-2)


def test_global_avgpool():
    x = randn(3, 16, 7, 7, seed=8)
    xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
    y = fr.global_avgpool_bf16(xb)
    dy = randn(3, 16, seed=9)
    dyb = torch.zeros(3, 16, dtype=torch.bfloat16, device=DEV)
    dyb[:, :16] = dy.to(torch.bfloat16)
    y.backward(dyb)
    ref = x.to(torch.bfloat16).float().mean(dim=(2, 3))
    torch.testing.assert_close(y[:, :16].float(), ref, rtol=1e-2, atol=1e-2)
    expect = (dy.to(torch.bfloat16).float() / 49).to(torch.bfloat16).float()
    torch.testing.assert_close(xb.grad[..., :16].float()[0, 0, 0], expect[0],
                               rtol=2e-2, atol=1e-3)


def test_add_relu():
    a = randn(1024 * 8, seed=10).to(torch.bfloat16).requires_grad_(True)
    b = randn(1024 * 8, seed=11).to(torch.bfloat16).requires_grad_(True)
    out = fr.add_relu_bf16(a, b)
    out.backward(torch.ones_like(out))
    ref = F.relu(a.detach().float() + b.detach().float())
    torch.testing.assert_close(out.float(), ref, rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(a.grad.float(), (ref > 0).float(), rtol=0, atol=0)


def test_resnet18_forward_backward_step():
    from dist_mnist_amd.models import ResNet18
    from dist_mnist_amd.ops import SGD

    torch.manual_seed(50)
    model = ResNet18(num_classes=1000).to(DEV).train()
    opt = SGD(model.parameters(), lr=0.01, momentum=0.9)
    x = torch.zeros(4, 224, 224, 8, dtype=torch.bfloat16, device=DEV)
    x[..., :3] = randn(4, 224, 224, 3, seed=51).to(torch.bfloat16)
    t = torch.randint(0, 1000, (4,), device=DEV)

    logp, loss = model(x, t)
    assert logp.shape == (4, 1000)
    lv = float(loss)
    assert lv == lv and 2.0 < lv < 14.0, lv  # ~ln(1000)=6.9 at init
    loss.backward()
    for n, p in model.named_parameters():
        assert p.grad is not None, n
        assert torch.isfinite(p.grad).all(), n
    opt.step()


def test_resnet18_matches_torch_reference():
    """End-to-end vs a plain-torch NCHW forward using the SAME parameters
    (eval mode: running stats, no dropout anywhere)."""
    from dist_mnist_amd.models import ResNet18

    torch.manual_seed(52)
    model = ResNet18(num_classes=64).to(DEV)
    # initialize running stats with a training pass
    xb = torch.zeros(2, 224, 224, 8, dtype=torch.bfloat16, device=DEV)
    xb[..., :3] = randn(2, 224, 224, 3, seed=53).to(torch.bfloat16)
    model.train()
    model(xb)
    model.eval()
    logp = model(xb)

    # torch reference forward on the same params (NCHW)
    def tconv(x, m, stride=1, pad=0):
        return F.conv2d(x, m.weight.to(torch.bfloat16).float(), None,
                        stride=stride, padding=pad)

    def tbn(x, m):
        return F.batch_norm(x, m.running_mean, m.running_var, m.weight, m.bias,
                            training=False, eps=m.eps)

    x = xb[..., :3].permute(0, 3, 1, 2).float()
    h = F.relu(tbn(tconv(x, model.conv1, 2, 3), model.bn1))
    h = F.max_pool2d(h, 3, 2, 1)
    for layer in (model.layer1, model.layer2, model.layer3, model.layer4):
        for blk in layer:
            idn = h
            o = F.relu(tbn(tconv(h, blk.conv1, blk.stride, 1), blk.bn1))
            o = tbn(tconv(o, blk.conv2, 1, 1), blk.bn2)
            if blk.downsample is not None:
                idn = tbn(tconv(h, blk.downsample[0], blk.downsample[0].stride[0]),
                          blk.downsample[1])
            h = F.relu(o + idn)
    h = h.mean(dim=(2, 3))
    h = F.linear(h, model.fc.weight, model.fc.bias)
    ref_logp = F.log_softmax(h, dim=1)

    # bf16 through 20 layers: compare distributions loosely but directionally
    rel_rms = ((logp - ref_logp).norm() / ref_logp.norm()).item()
    assert rel_rms < 0.12, rel_rms
    agree = (logp.argmax(1) == ref_logp.argmax(1)).float().mean().item()
    assert agree >= 0.5, agree


def test_batchnorm_relu_bwd_fused_gate():
    """The ReLU gate folded into bn backward must match torch's
    relu(batch_norm(x)) gradients."""
    N, C, H, W = 4, 16, 10, 12
    x = randn(N, C, H, W, seed=20)
    gamma = randn(C, seed=21).abs().add(0.5).requires_grad_(True)
    beta = randn(C, seed=22).requires_grad_(True)
    rm, rv = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)

    xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
    y = fr.batchnorm_bf16(xb, gamma, beta, rm, rv, training=True, relu=True)
    dy = randn(N, C, H, W, seed=23)
    y.backward(fb.nchw_to_nhwc_padded(dy, cp=16))

    q = lambda t: t.to(torch.bfloat16).float()
    xr = q(x).requires_grad_(True)
    gr = gamma.detach().clone().requires_grad_(True)
    br = beta.detach().clone().requires_grad_(True)
    rm2, rv2 = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)
    ref = F.relu(F.batch_norm(xr, rm2, rv2, gr, br, training=True,
                              momentum=0.1, eps=1e-5))
    ref.backward(q(dy))

    torch.testing.assert_close(fb.nhwc_to_nchw(xb.grad, C), xr.grad,
                               rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(gamma.grad, gr.grad, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(beta.grad, br.grad, rtol=3e-2, atol=3e-2)


def test_bn_fusion_matches_unfused_block():
    """The BN-apply fusion (bn_virtual + conv aff staging / add_aff) must
    reproduce the unfused batchnorm_bf16 -> conv / add path bit-for-bit on
    the forward (the staging rounds to bf16 exactly where bn_apply did) and
    to fp32-accumulation tolerance on the gradients."""
    import dist_mnist_amd.ops.functional_bf16 as fb
    import dist_mnist_amd.ops.functional_resnet as fr

    torch.manual_seed(77)
    N, C, Hs = 8, 64, 28
    x = torch.randn(N, Hs, Hs, C, device=DEV).to(torch.bfloat16)
    gamma = torch.randn(C, device=DEV).abs() + 0.5
    beta = torch.randn(C, device=DEV)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)
    w = (torch.randn(C, C, 3, 3, device=DEV) * 0.05).requires_grad_(True)

    # fused: stats-only BN -> conv with affine staging
    xa = x.clone().requires_grad_(True)
    t_tok, aff = fr.bn_virtual_bf16(xa, gamma, beta, rm.clone(), rv.clone(),
                                    training=True, relu=True)
    y_fused = fb.conv2d_bf16(t_tok, w, pad=1, cop=C, aff=aff)

    # unfused: materialized bn_apply -> plain conv
    xb = x.clone().requires_grad_(True)
    wb = w.detach().clone().requires_grad_(True)
    ybn = fr.batchnorm_bf16(xb, gamma, beta, rm.clone(), rv.clone(),
                            training=True, relu=True)
    y_ref = fb.conv2d_bf16(ybn, wb, pad=1, cop=C)

    torch.testing.assert_close(y_fused.float(), y_ref.float(), rtol=0, atol=0)

    dy = torch.randn_like(y_fused.float()).to(torch.bfloat16)
    y_fused.backward(dy)
    y_ref.backward(dy)
    torch.testing.assert_close(w.grad, wb.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(xa.grad.float(), xb.grad.float(),
                               rtol=1e-4, atol=1e-4)


def test_add_aff_fusion_matches_unfused():
    import dist_mnist_amd.ops.functional_resnet as fr

    torch.manual_seed(78)
    N, C, Hs = 8, 64, 14
    a = torch.randn(N, Hs, Hs, C, device=DEV).to(torch.bfloat16)
    b = torch.randn(N, Hs, Hs, C, device=DEV).to(torch.bfloat16)
    gamma = torch.randn(C, device=DEV).abs() + 0.5
    beta = torch.randn(C, device=DEV)
    rm = torch.zeros(C, device=DEV)
    rv = torch.ones(C, device=DEV)

    a1 = a.clone().requires_grad_(True)
    tok, aff = fr.bn_virtual_bf16(a1, gamma, beta, rm.clone(), rv.clone(),
                                  training=True, relu=False)
    out_fused = fr.add_aff_relu_bf16(tok, b, aff)

    a2 = a.clone().requires_grad_(True)
    ybn = fr.batchnorm_bf16(a2, gamma, beta, rm.clone(), rv.clone(),
                            training=True, relu=False)
    out_ref = fr.add_relu_bf16(ybn, b)

    torch.testing.assert_close(out_fused.float(), out_ref.float(),
                               rtol=0, atol=0)
    dy = torch.randn_like(out_fused.float()).to(torch.bfloat16)
    out_fused.backward(dy)
    out_ref.backward(dy)
    torch.testing.assert_close(a1.grad.float(), a2.grad.float(),
                               rtol=1e-4, atol=1e-4)
