"""Checkpoint format tests: plain torch.save
state_dict with the reference's exact keys/shapes/dtypes; load roundtrip."""

import torch

from dist_mnist_amd.models import Net
from dist_mnist_amd.utils import load_checkpoint, save_checkpoint
from tests.test_model_cpu import EXPECTED_KEYS


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(11)
    net = Net()
    path = str(tmp_path / "model.pt")
    save_checkpoint(net, path)

    # the file must be a torch.load-able plain state_dict (reference format)
    sd = torch.load(path, map_location="cpu", weights_only=True)
    assert set(sd.keys()) == set(EXPECTED_KEYS)
    for k, shape in EXPECTED_KEYS.items():
        assert tuple(sd[k].shape) == shape
        assert sd[k].dtype == torch.float32

    net2 = Net()
    load_checkpoint(net2, path)
    for (k1, v1), (k2, v2) in zip(net.state_dict().items(), net2.state_dict().items()):
        assert k1 == k2
        torch.testing.assert_close(v1, v2, rtol=0, atol=0)


def test_checkpoint_loadable_by_plain_torch_module(tmp_path):
    """A user of the reference must be able to torch.load our checkpoint into
    their own nn.Module with the same layer names."""
    from tests.test_model_cpu import TorchNet

    net = Net()
    path = str(tmp_path / "m.pt")
    save_checkpoint(net, path)
    oracle = TorchNet()
    missing, unexpected = oracle.load_state_dict(
        torch.load(path, map_location="cpu", weights_only=True), strict=True
    ), None
    x = torch.randn(2, 1, 28, 28)
    torch.testing.assert_close(oracle.eval()(x), net.eval()(x), rtol=1e-5, atol=1e-6)


def test_checkpoint_preserves_integer_buffers(tmp_path):
    """ResNet-18 BatchNorm num_batches_tracked is int64: the fp32 interchange
    cast must only touch floating-point tensors (ADVICE round-1 item)."""
    from dist_mnist_amd.models import ResNet18

    net = ResNet18()
    with torch.no_grad():
        net.bn1.num_batches_tracked.fill_(7)
    path = str(tmp_path / "rn.pt")
    save_checkpoint(net, path)
    sd = torch.load(path, map_location="cpu", weights_only=True)
    assert sd["bn1.num_batches_tracked"].dtype == torch.int64
    assert int(sd["bn1.num_batches_tracked"]) == 7
    assert sd["conv1.weight"].dtype == torch.float32

    net2 = ResNet18()
    load_checkpoint(net2, path)
    assert int(net2.bn1.num_batches_tracked) == 7
    for (k1, v1), (k2, v2) in zip(net.state_dict().items(),
                                  net2.state_dict().items()):
        assert k1 == k2
        torch.testing.assert_close(v1, v2, rtol=0, atol=0)
