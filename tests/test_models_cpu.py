"""CPU model tests: shapes, convergence, reference checkpoint parity."""

import torch
import torch.nn as nn

from mi355x import optim
from mi355x.models import Net, build_model, resnet18, resnet50
from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState


def test_net_forward_shape():
    net = Net()
    y = net(torch.randn(4, 3, 32, 32))
    assert y.shape == (4, 10)


def test_resnet18_forward_shape():
    net = resnet18()
    y = net(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_resnet50_forward_shape():
    net = resnet50(num_classes=1000)
    y = net(torch.randn(2, 3, 64, 64))  # smaller than 224 for test speed
    assert y.shape == (2, 1000)


def test_net_loss_decreases():
    torch.manual_seed(0)
    net = Net()
    flat = FlatState(net)
    opt = optim.SGD(flat, lr=0.05, momentum=0.9)
    x = torch.randn(16, 3, 32, 32)
    y = torch.randint(0, 10, (16,))
    losses = []
    for _ in range(40):
        opt.zero_grad()
        loss = cross_entropy(net(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.6


def test_resnet18_loss_decreases():
    torch.manual_seed(0)
    net = resnet18()
    flat = FlatState(net)
    opt = optim.SGD(flat, lr=0.05, momentum=0.9)
    x = torch.randn(8, 3, 32, 32)
    y = torch.randint(0, 10, (8,))
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = cross_entropy(net(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]


class _TorchRefNet(nn.Module):
    """The reference Net verbatim in stock torch
    (/root/reference/cifar_example.py:17-34) for state_dict parity."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 6, 5)
        self.pool = nn.MaxPool2d(2, 2)
        self.conv2 = nn.Conv2d(6, 16, 5)
        self.fc1 = nn.Linear(16 * 5 * 5, 120)
        self.fc2 = nn.Linear(120, 84)
        self.fc3 = nn.Linear(84, 10)

    def forward(self, x):
        import torch.nn.functional as F
        x = self.pool(F.relu(self.conv1(x)))
        x = self.pool(F.relu(self.conv2(x)))
        x = x.view(-1, 16 * 5 * 5)
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        return self.fc3(x)


def test_net_state_dict_parity_with_reference():
    """Our Net's state_dict loads into a torch-built reference Net and the
    two produce identical outputs (checkpoint format parity — SURVEY.md L0)."""
    ours = Net()
    ref = _TorchRefNet()
    sd = ours.state_dict()
    ref_sd = ref.state_dict()
    assert set(sd.keys()) == set(ref_sd.keys())
    for k in sd:
        assert sd[k].shape == ref_sd[k].shape, k
    ref.load_state_dict(sd)
    x = torch.randn(3, 3, 32, 32)
    torch.testing.assert_close(ours(x), ref(x), rtol=1e-4, atol=1e-5)


def test_checkpoint_roundtrip(tmp_path):
    net = Net()
    p = tmp_path / "ck.pth"
    torch.save(net.state_dict(), p)
    net2 = Net()
    net2.load_state_dict(torch.load(p, weights_only=True))
    x = torch.randn(2, 3, 32, 32)
    torch.testing.assert_close(net(x), net2(x))


def test_build_model_names():
    for name in ["net", "resnet18", "resnet50"]:
        assert build_model(name) is not None


def test_bn_num_batches_tracked_state_dict_parity():
    """The host-side BN step counter must still round-trip through
    state_dict under the torch key (torch checkpoints interchange)."""
    from mi355x.models.layers import BatchNorm2d

    bn = BatchNorm2d(8)
    x = torch.randn(2, 4, 4, 8)
    bn.train()
    bn(x)
    bn(x)
    sd = bn.state_dict()
    assert "num_batches_tracked" in sd
    assert int(sd["num_batches_tracked"]) == 2

    bn2 = BatchNorm2d(8)
    bn2.load_state_dict(sd)
    assert bn2._nbt == 2
    assert int(bn2.num_batches_tracked) == 2
