"""Model-level CPU tests: topology, state_dict surface, numerics vs a pure
PyTorch implementation of the reference Net."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from dist_mnist_amd.models import Net

# the §3.4 compatibility surface
EXPECTED_KEYS = {
    "conv1.weight": (10, 1, 5, 5),
    "conv1.bias": (10,),
    "conv2.weight": (20, 10, 5, 5),
    "conv2.bias": (20,),
    "fc1.weight": (50, 320),
    "fc1.bias": (50,),
    "fc2.weight": (10, 50),
    "fc2.bias": (10,),
}


class TorchNet(nn.Module):
    """Plain-torch replica of the reference topology, used as numeric oracle."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 10, kernel_size=5)
        self.conv2 = nn.Conv2d(10, 20, kernel_size=5)
        self.fc1 = nn.Linear(320, 50)
        self.fc2 = nn.Linear(50, 10)

    def forward(self, x):
        x = F.relu(F.max_pool2d(self.conv1(x), 2))
        x = F.relu(F.max_pool2d(self.conv2(x), 2))  # dropout eval-mode = identity
        x = x.view(-1, 320)
        x = F.relu(self.fc1(x))
        x = self.fc2(x)
        return F.log_softmax(x, dim=1)


def test_state_dict_surface():
    sd = Net().state_dict()
    assert set(sd.keys()) == set(EXPECTED_KEYS.keys())
    for k, shape in EXPECTED_KEYS.items():
        assert tuple(sd[k].shape) == shape, k
        assert sd[k].dtype == torch.float32


def test_forward_matches_torch_reference():
    torch.manual_seed(0)
    net = Net().eval()
    oracle = TorchNet().eval()
    oracle.load_state_dict(net.state_dict())
    x = torch.randn(8, 1, 28, 28)
    out = net(x)
    ref = oracle(x)
    assert out.shape == (8, 10)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


def test_fused_loss_matches_separate():
    torch.manual_seed(1)
    net = Net().eval()
    x = torch.randn(16, 1, 28, 28)
    t = torch.randint(0, 10, (16,))
    logp, loss = net(x, t)
    ref = F.nll_loss(net(x), t)
    torch.testing.assert_close(loss, ref, rtol=1e-6, atol=1e-7)
    torch.testing.assert_close(logp, net(x), rtol=1e-6, atol=1e-7)


def test_backward_matches_torch_reference():
    torch.manual_seed(2)
    net = Net().eval()  # eval => dropout off so grads comparable
    oracle = TorchNet().eval()
    oracle.load_state_dict(net.state_dict())
    x = torch.randn(8, 1, 28, 28)
    t = torch.randint(0, 10, (8,))

    _, loss = net(x, t)
    loss.backward()
    F.nll_loss(oracle(x), t).backward()

    osd = dict(oracle.named_parameters())
    for name, p in net.named_parameters():
        torch.testing.assert_close(p.grad, osd[name].grad, rtol=1e-4, atol=1e-6,
                                   msg=f"grad mismatch: {name}")


def test_loss_decreases_single_process():
    from dist_mnist_amd.ops import SGD

    torch.manual_seed(3)
    net = Net().eval()  # dropout off: deterministic memorization of the batch
    opt = SGD(net.parameters(), lr=0.1, momentum=0.5)
    x = torch.randn(64, 1, 28, 28)
    t = torch.randint(0, 10, (64,))
    losses = []
    for _ in range(60):
        opt.zero_grad()
        _, loss = net(x, t)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.7, losses[::10]


def test_resnet18_state_dict_surface():
    """ResNet-18 parameter names/shapes interchange with torchvision's
    resnet18 (torchvision is not installed here; the canonical key set is
    pinned explicitly)."""
    from dist_mnist_amd.models import ResNet18

    sd = ResNet18(num_classes=1000).state_dict()
    # spot-pin the canonical surface
    expected = {
        "conv1.weight": (64, 3, 7, 7),
        "bn1.weight": (64,),
        "bn1.running_mean": (64,),
        "layer1.0.conv1.weight": (64, 64, 3, 3),
        "layer2.0.downsample.0.weight": (128, 64, 1, 1),
        "layer2.0.downsample.1.running_var": (128,),
        "layer3.1.conv2.weight": (256, 256, 3, 3),
        "layer4.1.bn2.bias": (512,),
        "fc.weight": (1000, 512),
        "fc.bias": (1000,),
    }
    for k, shape in expected.items():
        assert k in sd, k
        assert tuple(sd[k].shape) == shape, (k, sd[k].shape)
    # no extra parameter families beyond the torchvision naming scheme
    prefixes = ("conv1", "bn1", "layer1", "layer2", "layer3", "layer4", "fc")
    for k in sd:
        assert k.startswith(prefixes) or k.endswith("num_batches_tracked"), k
