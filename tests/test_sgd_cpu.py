"""SGD semantics vs torch.optim.SGD on CPU, including
multi-param-group configs (per-group lr/momentum — the GPU path keys its
fused pointer tables per group, the CPU path must match torch exactly)."""

import torch

from dist_mnist_amd.ops import SGD


def _clone_params(n=3, seed=0):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(s, generator=g).requires_grad_(True)
            for s in [(4, 5), (7,), (2, 3, 3)][:n]]


def _step_pair(opt_a, opt_b, params_a, params_b, seed):
    g = torch.Generator().manual_seed(seed)
    for pa, pb in zip(params_a, params_b):
        gr = torch.randn(pa.shape, generator=g)
        pa.grad = gr.clone()
        pb.grad = gr.clone()
    opt_a.step()
    opt_b.step()


def test_sgd_matches_torch_single_group():
    pa = _clone_params()
    pb = [p.detach().clone().requires_grad_(True) for p in pa]
    ours = SGD(pa, lr=0.05, momentum=0.9)
    ref = torch.optim.SGD(pb, lr=0.05, momentum=0.9)
    for step in range(5):
        _step_pair(ours, ref, pa, pb, seed=100 + step)
    for a, b in zip(pa, pb):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-7)


def test_sgd_matches_torch_multi_group():
    pa = _clone_params(3, seed=1)
    pb = [p.detach().clone().requires_grad_(True) for p in pa]
    ga = [{"params": pa[:1], "lr": 0.1, "momentum": 0.5},
          {"params": pa[1:], "lr": 0.01, "momentum": 0.9}]
    gb = [{"params": pb[:1], "lr": 0.1, "momentum": 0.5},
          {"params": pb[1:], "lr": 0.01, "momentum": 0.9}]
    ours = SGD(ga, lr=0.05, momentum=0.0)
    ref = torch.optim.SGD(gb, lr=0.05, momentum=0.0)
    for step in range(5):
        _step_pair(ours, ref, pa, pb, seed=200 + step)
    for a, b in zip(pa, pb):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-7)


def test_sgd_momentumless_group():
    pa = _clone_params(2, seed=2)
    pb = [p.detach().clone().requires_grad_(True) for p in pa]
    ours = SGD(pa, lr=0.2, momentum=0.0)
    ref = torch.optim.SGD(pb, lr=0.2, momentum=0.0)
    for step in range(3):
        _step_pair(ours, ref, pa, pb, seed=300 + step)
    for a, b in zip(pa, pb):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-7)
