"""Diagnose the r18 tap-parity gap: noise floor (two TAP=0 runs) vs the
TAP=1 delta, worst element and owning parameter."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import os
import torch
from mi355x.models import build_model
from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState

torch.manual_seed(3)
net = build_model("resnet18").cuda()
flat = FlatState(net)
g = torch.Generator().manual_seed(5)
x = torch.randn(8, 3, 32, 32, generator=g).cuda()
yl = torch.randint(0, 10, (8,), generator=g).cuda()


def grads(tap):
    os.environ["MI355X_TAP"] = tap
    flat.zero_grad()
    cross_entropy(net(x), yl).backward()
    torch.cuda.synchronize()
    return flat.flat_grad.clone()


g0 = grads("0")
g0b = grads("0")
g1 = grads("1")
print("noise floor (0 vs 0):", (g0b - g0).abs().max().item())
d = (g1 - g0).abs()
print("tap delta: max abs", d.max().item())
i = int(d.argmax())
print("worst elem", i, "g0", g0[i].item(), "g1", g1[i].item())
off = 0
names = {id(p): n for n, p in net.named_parameters()}
for p in flat.params:
    n = p.numel()
    if off <= i < off + n:
        print("in param", names.get(id(p)), tuple(p.shape), "offset", i - off)
        break
    off += n
# per-param max deltas (top 5)
rows = []
off = 0
for p in flat.params:
    n = p.numel()
    rows.append(((d[off:off + n]).max().item(), names.get(id(p))))
    off += n
for v, nm in sorted(rows, reverse=True)[:6]:
    print(f"{v:.5f}  {nm}")
