"""Dynamic loss scaling (BASELINE config 5 / SURVEY.md §7 hard-part 4):
GradScaler semantics over the flat fp32 grad buffer — overflow skips the
step and halves the scale, clean streaks grow it. CPU tests here; the
GPU end-to-end overflow-recovery run is in test_gpu_train.py."""

import torch
from torch import nn

from mi355x import optim
from mi355x.amp import GradScaler
from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState


def test_scaler_halves_on_overflow_and_grows_on_clean_streak():
    s = GradScaler(init_scale=1024.0, growth_interval=3)
    g = torch.zeros(8)
    assert s.step_ok(g) and s.scale_value == 1024.0          # clean 1
    g[3] = float("inf")
    assert not s.step_ok(g) and s.scale_value == 512.0       # halved
    g[3] = 0.0
    for _ in range(2):
        assert s.step_ok(g) and s.scale_value == 512.0       # clean 1,2
    assert s.step_ok(g) and s.scale_value == 1024.0          # clean 3: grow
    g[0] = float("nan")
    assert not s.step_ok(g) and s.scale_value == 512.0       # nan counts too


def test_overflow_skips_step_and_recovers():
    """End-to-end on CPU: an injected inf gradient leaves params untouched;
    the next clean step trains with the halved scale."""
    torch.manual_seed(0)
    net = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 10))
    flat = FlatState(net)
    opt = optim.SGD(flat, lr=0.1, momentum=0.9)
    scaler = GradScaler(init_scale=2.0**10)
    x = torch.randn(4, 8)
    y = torch.randint(0, 10, (4,))

    def step(inject_inf=False):
        opt.zero_grad()
        used = scaler.scale_value
        loss = cross_entropy(net(x), y)
        (loss * used).backward()
        if inject_inf:
            flat.flat_grad[5] = float("inf")
        if scaler.step_ok(flat.flat_grad):
            opt.grad_scale = 1.0 / used
            opt.step()
            return True
        return False

    p_before = flat.flat_param.clone()
    assert not step(inject_inf=True)                  # skipped
    torch.testing.assert_close(flat.flat_param, p_before, rtol=0, atol=0)
    assert scaler.scale_value == 2.0**9
    assert step()                                     # recovers
    assert not torch.equal(flat.flat_param, p_before)
    # the applied update equals the unscaled-gradient update
    torch.manual_seed(0)
    ref = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 10))
    rflat = FlatState(ref)
    ropt = optim.SGD(rflat, lr=0.1, momentum=0.9)
    ropt.zero_grad()
    cross_entropy(ref(x), y).backward()
    ropt.step()
    torch.testing.assert_close(flat.flat_param, rflat.flat_param,
                               rtol=1e-5, atol=1e-7)
