"""Flat param/grad/momentum state + fused SGD semantics."""

import torch

from mi355x import optim
from mi355x.models import Net, resnet18
from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState


def test_views_alias_flat():
    net = Net()
    flat = FlatState(net)
    flat.flat_param.zero_()
    for p in net.parameters():
        assert p.abs().sum() == 0  # params are views of the flat buffer
    assert flat.numel == 62006  # Net parameter count


def test_grad_views_accumulate():
    net = Net()
    flat = FlatState(net)
    x = torch.randn(4, 3, 32, 32)
    loss = cross_entropy(net(x), torch.randint(0, 10, (4,)))
    loss.backward()
    assert flat.flat_grad.abs().sum() > 0
    for p in net.parameters():
        assert p.grad is not None
        s, e = flat.offsets[id(p)]
        torch.testing.assert_close(p.grad.reshape(-1), flat.flat_grad[s:e])
    flat.zero_grad()
    assert flat.flat_grad.abs().sum() == 0


def test_bucket_layout_contiguous_and_complete():
    net = resnet18()
    flat = FlatState(net, bucket_bytes=4 << 20, first_bucket_bytes=1 << 20)
    assert flat.buckets[0].start == 0
    assert flat.buckets[-1].end == flat.numel
    for a, b in zip(flat.buckets, flat.buckets[1:]):
        assert a.end == b.start
    # every param maps to exactly one bucket
    assert len(flat.bucket_of) == len(flat.params)


def test_sgd_matches_torch_sgd():
    torch.manual_seed(3)
    net_a = Net()
    net_b = Net()
    net_b.load_state_dict(net_a.state_dict())

    flat = FlatState(net_a)
    opt_a = optim.SGD(flat, lr=0.01, momentum=0.9, weight_decay=1e-4)
    opt_b = torch.optim.SGD(net_b.parameters(), lr=0.01, momentum=0.9,
                            weight_decay=1e-4)
    x = torch.randn(8, 3, 32, 32)
    y = torch.randint(0, 10, (8,))
    for _ in range(3):
        opt_a.zero_grad()
        cross_entropy(net_a(x), y).backward()
        opt_a.step()
        opt_b.zero_grad()
        torch.nn.functional.cross_entropy(net_b(x), y).backward()
        opt_b.step()
    for pa, pb in zip(net_a.parameters(), net_b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-4, atol=1e-6)


def test_grad_scale_folds_average():
    net = Net()
    flat = FlatState(net)
    opt = optim.SGD(flat, lr=1.0, momentum=0.0, grad_scale=0.5)
    flat.flat_grad.fill_(2.0)
    before = flat.flat_param.clone()
    opt.step()
    torch.testing.assert_close(flat.flat_param, before - 1.0)


def test_checkpoint_after_flatten(tmp_path):
    """state_dict of a flat-param model round-trips (params are views of
    one shared flat storage; torch.save dedups the storage)."""
    torch.manual_seed(5)
    net = Net()
    FlatState(net)
    p = tmp_path / "flat_ck.pth"
    torch.save(net.state_dict(), p)
    net2 = Net()  # plain (unflattened) model loads the same checkpoint
    net2.load_state_dict(torch.load(p, weights_only=True))
    x = torch.randn(2, 3, 32, 32)
    torch.testing.assert_close(net(x), net2(x))


def test_grad_scaler_dynamic():
    from mi355x.amp import GradScaler
    sc = GradScaler(init_scale=16.0, growth_interval=2)
    g = torch.ones(10)
    loss = torch.tensor(2.0)
    assert float(sc.scale(loss)) == 32.0
    assert sc.step_ok(g) and sc.step_ok(g)
    assert sc.scale_value == 32.0  # doubled after growth_interval clean steps
    g[3] = float("inf")
    assert not sc.step_ok(g)
    assert sc.scale_value == 16.0  # backed off on overflow
