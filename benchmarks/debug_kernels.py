"""One-trip GPU kernel debug harness: per-op max-diff breakdown + r50
repro. Run: python benchmarks/debug_kernels.py [probe|conv|r50|all]"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import mi355x.ops as ops
from mi355x.ops import functional as fn


def q(t):
    return t.to(torch.bfloat16).float()


def md(a, b):
    return (a.float().cpu() - b.float()).abs().max().item()


def probe():
    g = torch.Generator().manual_seed(11)
    A = q(torch.randn(32, 16, generator=g))
    B = q(torch.randn(16, 32, generator=g))
    D = ops.ext().mfma_probe32(A.cuda(), B.cuda())
    print("mfma probe maxdiff vs A@B:", md(D, A @ B))
    I = torch.zeros(32, 16)
    I[:16, :16] = torch.eye(16)
    D2 = ops.ext().mfma_probe32(I.cuda(), B.cuda())
    print("mfma identity rows maxdiff:", md(D2[:16], B),
          " lower-rows-zero:", D2[16:].abs().max().item())
    torch.cuda.synchronize()


CASES = [
    ((2, 32, 32, 3), 6, 5, 1, 0),
    ((2, 14, 14, 6), 16, 5, 1, 0),
    ((2, 32, 32, 16), 32, 3, 1, 1),
    ((2, 16, 16, 32), 64, 3, 2, 1),
    ((2, 8, 8, 64), 64, 3, 1, 1),
    ((2, 64, 64, 64), 64, 3, 1, 1),
    ((2, 8, 8, 128), 128, 1, 1, 0),
    ((2, 10, 10, 256), 128, 1, 1, 0),
    ((2, 16, 16, 128), 256, 1, 2, 0),
    ((2, 15, 15, 128), 128, 3, 2, 1),
    ((3, 7, 7, 64), 192, 3, 2, 1),
    ((4, 16, 16, 64), 128, 3, 2, 1),
    ((2, 9, 9, 256), 512, 1, 2, 0),
    ((2, 32, 32, 3), 64, 3, 1, 1),
    ((2, 32, 32, 3), 64, 7, 2, 3),
    ((2, 10, 10, 24), 64, 3, 1, 1),
]


def conv_cases():
    for shape, K, ksz, stride, pad in CASES:
        C = shape[-1]
        g = torch.Generator().manual_seed(0)
        x = q(torch.randn(*shape, generator=g))
        w = q(torch.randn(K, C, ksz, ksz, generator=g) * 0.2)
        xc = x.clone().requires_grad_(True)
        xg = x.cuda().to(torch.bfloat16).requires_grad_(True)
        wc = w.clone().requires_grad_(True)
        wg = w.cuda().requires_grad_(True)
        yc = fn.conv2d(xc, wc, None, stride, pad, None)
        yg = fn.conv2d(xg, wg, None, stride, pad, None)
        dy = q(torch.randn(*yc.shape, generator=g))
        yc.backward(dy)
        yg.backward(dy.cuda().to(torch.bfloat16))
        torch.cuda.synchronize()
        print(f"conv C={C} K={K} k={ksz} s={stride} p={pad}: "
              f"fwd={md(yg, yc):.4f} dx={md(xg.grad, xc.grad):.4f} "
              f"dw={md(wg.grad, wc.grad):.4f}")


def r50():
    from mi355x import optim
    from mi355x.models import build_model
    from mi355x.parallel.flat import FlatState

    torch.manual_seed(0)
    net = build_model("resnet50", num_classes=1000).cuda()
    flat = FlatState(net)
    opt = optim.SGD(flat, lr=0.01, momentum=0.9)
    x = torch.randn(4, 3, 64, 64, device="cuda")
    y = torch.randint(0, 1000, (4,), device="cuda")
    for i in range(3):
        opt.zero_grad()
        loss = fn.cross_entropy(net(x), y)
        print("fwd done", i, flush=True)
        loss.backward()
        torch.cuda.synchronize()
        print("bwd done", i, flush=True)
        opt.step()
        print("r50 step", i, "loss", loss.item(), flush=True)


def r50_fwd_steps():
    """Step r50 forward submodule-by-submodule with syncs to localize a
    device fault."""
    from mi355x.models import build_model
    from mi355x.models.layers import to_model_layout

    def ck(tag, h):
        torch.cuda.synchronize()
        print("ok:", tag, tuple(h.shape), flush=True)
        return h

    torch.manual_seed(0)
    net = build_model("resnet50").cuda()
    net.train()
    x = torch.randn(4, 3, 64, 64, device="cuda")
    h = ck("layout", to_model_layout(x))
    h = ck("conv1", net.conv1(h))
    h = ck("bn1", net.bn1(h))
    h = ck("maxpool", net.maxpool(h))
    for name in ["layer1", "layer2", "layer3", "layer4"]:
        layer = getattr(net, name)
        for i, blk in enumerate(layer):
            ident = h if blk.downsample is None else None
            if blk.downsample is not None:
                ident = ck(f"{name}.{i}.ds.conv", blk.downsample.conv(h))
                ident = ck(f"{name}.{i}.ds.bn", blk.downsample.bn(ident))
            o = ck(f"{name}.{i}.conv1", blk.conv1(h))
            o = ck(f"{name}.{i}.bn1", blk.bn1(o))
            o = ck(f"{name}.{i}.conv2", blk.conv2(o))
            o = ck(f"{name}.{i}.bn2", blk.bn2(o))
            o = ck(f"{name}.{i}.conv3", blk.conv3(o))
            h = ck(f"{name}.{i}.bn3", blk.bn3(o, residual=ident))
    h = ck("gap", ops.global_avg_pool(h))
    h = ck("fc", net.fc(h))
    print("forward complete", flush=True)


def convperf():
    """fwd conv timing per layer shape (glds A/B via MI355X_CONV_GLDS)."""
    import time
    import mi355x.ops as O
    shapes = [  # (N,H,W,C,K,ksz,stride,pad) — r18-CIFAR b1024 layers
        (1024, 32, 32, 64, 64, 3, 1, 1),
        (1024, 16, 16, 128, 128, 3, 1, 1),
        (1024, 8, 8, 256, 256, 3, 1, 1),
        (1024, 4, 4, 512, 512, 3, 1, 1),
    ]
    e = torch.empty(0, device="cuda")
    for (N, H, W, C, K, ksz, st, pad) in shapes:
        x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
        w = (torch.randn(K, ksz, ksz, C, device="cuda") * 0.1).to(torch.bfloat16)
        for _ in range(3):
            O.ext().conv2d_fwd(x, w, e, st, pad, 0, ksz, ksz)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            O.ext().conv2d_fwd(x, w, e, st, pad, 0, ksz, ksz)
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 30 * 1e6
        tf = 2.0 * N * H * W * K * ksz * ksz * C / st / st / 1e12 / (us / 1e6)
        print(f"fwd C={C} K={K} {H}x{W}: {us:.1f} us  ({tf:.0f} TF)")


def membw():
    """HBM bandwidth calibration: what do simple streams actually reach?
    Gives the real 'floor' to price memory-bound kernels against."""
    import time
    n = 256 * 1024 * 1024  # 512 MB bf16
    a = torch.randn(n, device="cuda").to(torch.bfloat16)
    b = torch.empty_like(a)
    idx = torch.zeros(n // 8, dtype=torch.uint8, device="cuda")
    for tag, fn_, bytes_ in [
        ("copy (torch)", lambda: b.copy_(a), 2 * n * 2),
        ("add (torch)", lambda: torch.add(a, a, out=b), 3 * n * 2),
        ("u8 read + bf16 write", lambda: b[: n // 8].copy_(idx.to(torch.bfloat16)),
         idx.numel() * 3),
    ]:
        for _ in range(3):
            fn_()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fn_()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"membw {tag}: {bytes_ / dt / 1e12:.2f} TB/s")


def stemperf():
    """ImageNet 7x7/s2 stem fwd + wgrad timing (stem7 dot2 kernel vs floor)."""
    import time
    import mi355x.ops as O
    N, H, W, C, K = 256, 224, 224, 3, 64
    P = Q = 112
    x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
    w = (torch.randn(K, 7, 7, C, device="cuda") * 0.1).to(torch.bfloat16)
    dy = torch.randn(N, P, Q, K, device="cuda").to(torch.bfloat16)
    e = torch.empty(0, device="cuda")
    for tag, fn_ in [
        ("fwd", lambda: O.ext().conv2d_fwd(x, w, e, 2, 3, 0, 7, 7)),
        ("wgrad", lambda: O.ext().conv2d_wgrad(x, dy, 7, 7, 2, 3)),
    ]:
        for _ in range(3):
            fn_()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(20):
            fn_()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 20 * 1e6
        tf = 2.0 * N * P * Q * K * 49 * C / 1e12 / (us / 1e6)
        print(f"stem {tag}: {us:.1f} us  ({tf:.1f} TF)")


def pmcprobe():
    """Run each hot kernel ~10x in one process for a rocprofv3 --pmc pass
    (counters aggregate per kernel symbol)."""
    import mi355x.ops as O
    e = torch.empty(0, device="cuda")
    # stem shapes (r50-224 b256)
    x7 = torch.randn(256, 224, 224, 3, device="cuda").to(torch.bfloat16)
    w7 = (torch.randn(64, 7, 7, 3, device="cuda") * 0.1).to(torch.bfloat16)
    dy7 = torch.randn(256, 112, 112, 64, device="cuda").to(torch.bfloat16)
    # maxpool (r50 stem pool)
    xp = torch.randn(256, 112, 112, 64, device="cuda").to(torch.bfloat16)
    # bn apply (r50 layer1-ish)
    xb = torch.randn(256, 56, 56, 256, device="cuda").to(torch.bfloat16)
    g1 = torch.randn(256, device="cuda")
    # layer1 wgrad MFMA (r18 shape)
    xw = torch.randn(256, 32, 32, 64, device="cuda").to(torch.bfloat16)
    dyw = torch.randn(256, 32, 32, 64, device="cuda").to(torch.bfloat16)
    # conv gather-GEMM shapes (r18 layer1 fwd + dgrad at b1024)
    xg = torch.randn(1024, 32, 32, 64, device="cuda").to(torch.bfloat16)
    wg = (torch.randn(64, 3, 3, 64, device="cuda") * 0.1).to(torch.bfloat16)
    yp, ip = O.ext().maxpool_fwd(xp, 3, 2, 1)
    st = O.ext().bn_stats(xb)
    mean = st[0] / xb.numel() * 256
    inv = torch.rsqrt((st[1] / (xb.numel() / 256) - mean * mean).clamp_min(1e-5))
    for _ in range(10):
        O.ext().conv2d_fwd(x7, w7, e, 2, 3, 0, 7, 7)
        O.ext().conv2d_wgrad(x7, dy7, 7, 7, 2, 3)
        O.ext().maxpool_fwd(xp, 3, 2, 1)
        O.ext().maxpool_bwd(yp.clone(), ip, 112, 112, 3, 2, 1)
        O.ext().bn_apply(xb, mean, inv, g1, torch.zeros_like(g1), e, 1, True)
        O.ext().conv2d_wgrad(xw, dyw, 3, 3, 1, 1)
        O.ext().conv2d_fwd(xg, wg, e, 1, 1, 0, 3, 3)
    torch.cuda.synchronize()
    print("pmcprobe done")


def bnperf():
    """bn_stats / bn_bwd_reduce timing on the layer1 shape."""
    import time
    import mi355x.ops as O
    x = torch.randn(1024, 32, 32, 64, device="cuda").to(torch.bfloat16)
    dy = torch.randn_like(x)
    y = torch.relu(x)
    s = O.ext().bn_stats(x)
    mean = s[0] / x.numel() * 64
    invstd = torch.rsqrt(torch.ones(64, device="cuda"))
    eb = torch.empty(0, device="cuda", dtype=torch.uint8)
    ef = torch.empty(0, device="cuda")
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        O.ext().bn_stats(x)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    for _ in range(50):
        O.ext().bn_bwd_reduce(x, dy, y, mean, invstd, eb, ef, ef)
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    mb = x.numel() * 2 / 1e6
    print(f"bn_stats {((t1-t0)/50)*1e6:.1f} us ({mb/((t1-t0)/50*1e6)*1e3:.2f} TB/s); "
          f"bn_bwd_reduce {((t2-t1)/50)*1e6:.1f} us ({3*mb/((t2-t1)/50*1e6)*1e3:.2f} TB/s)")


def wgradperf():
    """Per-variant wgrad timing (layer1 shape) with hip events."""
    import mi355x.ops as O
    torch.manual_seed(0)
    N, H, W, C, K = 256, 32, 32, 64, 64
    x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
    dy = torch.randn(N, H, W, K, device="cuda").to(torch.bfloat16)
    for _ in range(3):
        O.ext().conv2d_wgrad(x, dy, 3, 3, 1, 1)
    torch.cuda.synchronize()
    import time
    t0 = time.perf_counter()
    for _ in range(20):
        O.ext().conv2d_wgrad(x, dy, 3, 3, 1, 1)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 20 * 1e6
    gf = 2 * N * H * W * 64 * 64 * 9 / 1e9
    print(f"wgrad layer1: {dt:.1f} us/call  ({gf / dt * 1e3:.0f} TF)")


def pmcs3():
    """Drive the r18 hot kernels (s3 wgrad + patch fwd/dgrad, b8192 layer
    shapes) repeatedly for a rocprofv3 --pmc pass: is s3 LDS-array-bound
    as the issue-cycle model predicts?"""
    import mi355x.ops as O
    e = torch.empty(0, device="cuda")
    shapes = [(8192, 32, 64), (8192, 16, 128)]
    for b, hw, c in shapes:
        x = torch.randn(b, hw, hw, c, device="cuda").to(torch.bfloat16)
        w = (torch.randn(c, 3, 3, c, device="cuda") * 0.1).to(torch.bfloat16)
        wf = w.permute(1, 2, 3, 0).contiguous()
        dy = torch.randn_like(x)
        for _ in range(6):
            O.ext().conv2d_wgrad(x, dy, 3, 3, 1, 1)            # s3
            O.ext().conv2d_fwd(x, w, e, 1, 1, 0, 3, 3)         # patch fwd
            O.ext().conv2d_dgrad(dy, wf, 1, 1, hw, hw,
                                 torch.empty(0, device="cuda",
                                             dtype=x.dtype))   # patch dgrad
    # round-2 kernels: stem strip fwd / merged wgrad, parity-s2 dgrad
    xs = torch.randn(256, 224, 224, 3, device="cuda").to(torch.bfloat16)
    ws = (torch.randn(64, 7, 7, 3, device="cuda") * 0.1).to(torch.bfloat16)
    dys = torch.randn(256, 112, 112, 64, device="cuda").to(torch.bfloat16)
    xd = torch.randn(512, 28, 28, 128, device="cuda").to(torch.bfloat16)
    wd = (torch.randn(256, 3, 3, 128, device="cuda") * 0.1).to(torch.bfloat16)
    wdf = wd.permute(1, 2, 3, 0).contiguous()
    dyd = torch.randn(512, 14, 14, 256, device="cuda").to(torch.bfloat16)
    ea = torch.empty(0, device="cuda", dtype=torch.bfloat16)
    for _ in range(6):
        O.ext().conv2d_fwd(xs, ws, e, 2, 3, 0, 7, 7)      # stem strip
        O.ext().conv2d_wgrad(xs, dys, 7, 7, 2, 3)          # stem wgrad
        O.ext().conv2d_dgrad(dyd, wdf, 2, 1, 28, 28, ea)   # parity-s2
    torch.cuda.synchronize()
    print("pmcs3 done")


if __name__ == "__main__":
    what = sys.argv[1] if len(sys.argv) > 1 else "all"
    if what in ("probe", "all"):
        probe()
    if what in ("conv", "all"):
        conv_cases()
    if what == "wgradperf":
        wgradperf()
    if what == "bnperf":
        bnperf()
    if what == "convperf":
        convperf()
    if what == "stemperf":
        stemperf()
    if what == "membw":
        membw()
    if what == "pmcs3":
        pmcs3()
    if what == "pmcprobe":
        pmcprobe()
    if what == "r50fwd":
        r50_fwd_steps()
    if what in ("r50", "all"):
        r50()
