"""GPU kernel numerics: every HIP kernel vs the plain-torch fp32 CPU
reference on randomized shapes. bf16 tolerances."""

import pytest
import torch

import mi355x.ops as ops
from mi355x.ops import functional as fn

pytestmark = pytest.mark.gpu

RTOL, ATOL = 3e-2, 3e-2


def _pair(shape, requires_grad=False, seed=0):
    g = torch.Generator().manual_seed(seed)
    gpu = torch.randn(*shape, generator=g).to("cuda").to(torch.bfloat16)
    # the fp32 reference runs on the SAME bf16-quantized values the kernel
    # sees, so only accumulation-order rounding separates the two paths
    cpu = gpu.float().cpu()
    cpu.requires_grad_(requires_grad)
    gpu.requires_grad_(requires_grad)
    return cpu, gpu


def _qt(t):
    """bf16-quantize an fp32 reference tensor (what the kernel consumes)."""
    return t.to(torch.bfloat16).float()


def _close(gpu_t, cpu_t, rtol=RTOL, atol=ATOL):
    torch.testing.assert_close(gpu_t.float().cpu(), cpu_t.float(),
                               rtol=rtol, atol=atol)


@pytest.mark.parametrize("shape,K,ksz,stride,pad,bias,act", [
    ((2, 32, 32, 3), 6, 5, 1, 0, True, "relu"),   # Net conv1
    ((2, 14, 14, 6), 16, 5, 1, 0, True, None),    # Net conv2
    ((2, 32, 32, 16), 32, 3, 1, 1, False, None),  # ResNet 3x3
    ((2, 16, 16, 32), 64, 3, 2, 1, False, None),  # stride-2 downsample
    ((2, 16, 16, 32), 64, 1, 2, 0, False, None),  # 1x1 shortcut
    ((2, 33, 33, 8), 16, 7, 2, 3, False, None),   # 7x7/2 stem, odd size
    ((2, 8, 8, 64), 64, 3, 1, 1, False, "relu"),  # MFMA path shape
    ((2, 64, 64, 64), 64, 3, 1, 1, False, None),   # multi-m-chunk wgrad+reduce
    ((3, 7, 7, 64), 64, 3, 1, 1, False, None),     # patch: Wo=7 (odd row span)
    ((2, 14, 14, 128), 128, 3, 1, 1, False, None), # patch: Wo=14
    ((2, 8, 8, 128), 128, 1, 1, 0, False, None),  # MFMA 1x1
    ((2, 10, 10, 256), 128, 1, 1, 0, False, None), # t128 1x1, 2 c-blocks
    ((2, 16, 16, 128), 256, 1, 2, 0, False, None), # t128 1x1 stride-2 (r50 ds)
    ((2, 15, 15, 128), 128, 3, 2, 1, False, None), # t128 3x3/s2/p1 (r50 conv2)
    ((3, 7, 7, 64), 192, 3, 2, 1, False, None),   # MFMA stride-2, odd M
    ((2, 32, 32, 3), 64, 3, 1, 1, False, None),    # stem3 GEMM strip (CIFAR)
    ((2, 32, 32, 3), 64, 3, 1, 1, True, "relu"),   # stem3 GEMM strip, bias+relu
    ((2, 30, 30, 3), 64, 3, 1, 1, False, None),    # stem3 GEMM v2 fallback (900 % 128 != 0)
    ((2, 34, 34, 3), 64, 3, 1, 0, False, None),    # stem3 GEMM pad=0 (32x32 out)
    ((2, 32, 32, 3), 64, 7, 2, 3, True, "relu"),   # stem7 GEMM (strip: Ho*Wo%128==0)
    ((2, 30, 30, 3), 64, 7, 2, 3, False, None),    # stem7 GEMM v2 fallback (15x15 out)
    ((2, 29, 29, 3), 32, 7, 2, 3, False, None),    # stem7 odd W, K<64
    ((2, 10, 10, 24), 64, 3, 1, 1, False, None),   # GENC C=24
])
def test_conv2d_fwd_bwd(shape, K, ksz, stride, pad, bias, act):
    C = shape[-1]
    xc, xg = _pair(shape, requires_grad=True)
    g = torch.Generator().manual_seed(1)
    w = _qt(torch.randn(K, C, ksz, ksz, generator=g) * 0.2)
    b = torch.randn(K, generator=g) * 0.1 if bias else None
    wc = w.clone().requires_grad_(True)
    wg = w.clone().cuda().requires_grad_(True)
    bc = b.clone().requires_grad_(True) if bias else None
    bg = b.clone().cuda().requires_grad_(True) if bias else None

    yc = fn.conv2d(xc, wc, bc, stride, pad, act)
    yg = fn.conv2d(xg, wg, bg, stride, pad, act)
    _close(yg, yc)

    dy = torch.randn(*yc.shape, generator=g)
    dyq = dy.to(torch.bfloat16).float()
    if act == "relu":
        # the GPU masks with its own bf16 forward's y>0; near-zero y can
        # round across 0, so the fp32 reference must use the GPU's mask
        dyq = dyq * (yg.detach().float().cpu() > 0)
        yc_lin = fn.conv2d(xc, wc, bc, stride, pad, None)
        yc_lin.backward(dyq)
    else:
        yc.backward(dyq)
    yg.backward(dy.cuda().to(torch.bfloat16))
    _close(xg.grad, xc.grad)
    _close(wg.grad, wc.grad, rtol=5e-2, atol=5e-2)
    if bias:
        _close(bg.grad, bc.grad, rtol=5e-2, atol=5e-2)


@pytest.mark.parametrize("M,N,K,bias,act", [
    (8, 10, 400, True, None),
    (16, 120, 400, True, "relu"),
    (4, 84, 120, True, "relu"),
    (32, 10, 512, True, None),
    (8, 33, 100, False, None),  # K % 8 != 0 scalar path
    # MFMA tile path (gemm_mfma.hip): exact tiles and M/N/K tails
    (256, 64, 64, True, None),
    (300, 100, 128, True, "relu"),   # M tail, N tail, dgrad falls back
    (512, 1000, 256, False, None),   # ResNet-50-classifier-like, MFMA dgrad
    (256, 120, 400, True, None),     # partial final K-slice (400 = 6*64+16)
])
def test_linear_fwd_bwd(M, N, K, bias, act):
    xc, xg = _pair((M, K), requires_grad=True)
    g = torch.Generator().manual_seed(2)
    w = _qt(torch.randn(N, K, generator=g) * 0.1)
    b = torch.randn(N, generator=g) * 0.1 if bias else None
    wc = w.clone().requires_grad_(True)
    wg = w.clone().cuda().requires_grad_(True)
    bc = b.clone().requires_grad_(True) if bias else None
    bg = b.clone().cuda().requires_grad_(True) if bias else None

    yc = fn.linear(xc, wc, bc, act)
    yg = fn.linear(xg, wg, bg, act)
    _close(yg, yc)

    dy = torch.randn(*yc.shape, generator=g)
    dyq = dy.to(torch.bfloat16).float()
    if act == "relu":
        dyq = dyq * (yg.detach().float().cpu() > 0)
        yc_lin = fn.linear(xc, wc, bc, None)
        yc_lin.backward(dyq)
    else:
        yc.backward(dyq)
    yg.backward(dy.cuda().to(torch.bfloat16))
    _close(xg.grad, xc.grad)
    _close(wg.grad, wc.grad)
    if bias:
        _close(bg.grad, bc.grad)


@pytest.mark.parametrize("shape,res,act", [
    ((4, 8, 8, 16), False, None),
    ((4, 8, 8, 16), True, None),
    ((2, 5, 7, 32), False, None),
    ((4, 8, 8, 16), True, "relu"),
])
def test_batch_norm_train_fwd_bwd(shape, res, act):
    C = shape[-1]
    xc, xg = _pair(shape, requires_grad=True)
    resc = resg = None
    if res:
        resc, resg = _pair(shape, requires_grad=True, seed=9)
    g = torch.Generator().manual_seed(3)
    gamma = torch.rand(C, generator=g) + 0.5
    beta = torch.randn(C, generator=g) * 0.1
    gc, bc = gamma.clone().requires_grad_(True), beta.clone().requires_grad_(True)
    gg, bg = gamma.clone().cuda().requires_grad_(True), beta.clone().cuda().requires_grad_(True)
    rm_c, rv_c = torch.zeros(C), torch.ones(C)
    rm_g, rv_g = rm_c.clone().cuda(), rv_c.clone().cuda()

    yc = fn.batch_norm(xc, gc, bc, rm_c, rv_c, True, 0.1, resc, act)
    yg = fn.batch_norm(xg, gg, bg, rm_g, rv_g, True, 0.1, resg, act)
    _close(yg, yc)
    _close(rm_g, rm_c, rtol=2e-2, atol=2e-2)  # running stats updated alike
    _close(rv_g, rv_c, rtol=2e-2, atol=2e-2)

    dy = torch.randn(*yc.shape, generator=g)
    dyq = dy.to(torch.bfloat16).float()
    if act == "relu":
        dyq = dyq * (yg.detach().float().cpu() > 0)
        yc_lin = fn.batch_norm(xc, gc, bc, rm_c.clone(), rv_c.clone(), True,
                               0.1, resc, None)
        yc_lin.backward(dyq)
    else:
        yc.backward(dyq)
    yg.backward(dy.cuda().to(torch.bfloat16))
    _close(xg.grad, xc.grad, rtol=5e-2, atol=5e-2)
    _close(gg.grad, gc.grad, rtol=5e-2, atol=5e-2)
    _close(bg.grad, bc.grad, rtol=5e-2, atol=5e-2)
    if res:
        _close(resg.grad, resc.grad)


def test_batch_norm_eval():
    C = 8
    xc, xg = _pair((2, 4, 4, C))
    rm = torch.randn(C) * 0.1
    rv = torch.rand(C) + 0.5
    gamma, beta = torch.rand(C) + 0.5, torch.randn(C) * 0.1
    yc = fn.batch_norm(xc, gamma, beta, rm, rv, False)
    yg = fn.batch_norm(xg, gamma.cuda(), beta.cuda(), rm.cuda(), rv.cuda(),
                       False)
    _close(yg, yc)


@pytest.mark.parametrize("shape,k,s,p", [
    ((2, 28, 28, 6), 2, 2, 0),   # Net pool
    ((2, 10, 10, 16), 2, 2, 0),
    ((2, 56, 56, 8), 3, 2, 1),   # ResNet stem pool (overlapping)
])
def test_maxpool_fwd_bwd(shape, k, s, p):
    xc, xg = _pair(shape, requires_grad=True)
    yc = fn.max_pool2d(xc, k, s, p)
    yg = fn.max_pool2d(xg, k, s, p)
    _close(yg, yc)
    g = torch.Generator().manual_seed(4)
    dy = torch.randn(*yc.shape, generator=g)
    dyq = dy.to(torch.bfloat16).float()
    yc.backward(dyq)
    yg.backward(dy.cuda().to(torch.bfloat16))
    if s >= k:
        # non-overlapping: exactly one element per window carries the grad;
        # bf16 ties may pick a different element than fp32, so compare the
        # per-window sums (invariant to the tie choice)
        gc_w = xc.grad.unfold(1, k, s).unfold(2, k, s).sum(dim=(-2, -1))
        gg_w = xg.grad.float().cpu().unfold(1, k, s).unfold(2, k, s).sum(
            dim=(-2, -1))
        torch.testing.assert_close(gg_w, gc_w, rtol=RTOL, atol=ATOL)
    else:
        torch.testing.assert_close(xg.grad.float().sum().cpu(),
                                   xc.grad.sum(), rtol=2e-2, atol=1.0)


def test_global_avg_pool():
    xc, xg = _pair((3, 4, 4, 512), requires_grad=True)
    yc = fn.global_avg_pool(xc)
    yg = fn.global_avg_pool(xg)
    _close(yg, yc)
    g = torch.Generator().manual_seed(5)
    dy = torch.randn(*yc.shape, generator=g)
    yc.backward(dy)
    yg.backward(dy.cuda().to(torch.bfloat16))
    _close(xg.grad, xc.grad)


@pytest.mark.parametrize("B,C", [(8, 10), (64, 1000)])
def test_cross_entropy_fwd_bwd(B, C):
    xc, xg = _pair((B, C), requires_grad=True)
    g = torch.Generator().manual_seed(6)
    t = torch.randint(0, C, (B,), generator=g)
    lc = fn.cross_entropy(xc, t)
    lg = fn.cross_entropy(xg, t.cuda())
    _close(lg, lc, rtol=2e-2, atol=2e-2)
    lc.backward()
    lg.backward()
    _close(xg.grad, xc.grad, rtol=2e-2, atol=1e-3)


def test_relu_bwd_kernel():
    g = torch.Generator().manual_seed(7)
    y = torch.randn(1000, generator=g)
    dy = torch.randn(1000, generator=g)
    out = ops.ext().relu_bwd(dy.cuda().to(torch.bfloat16),
                             y.cuda().to(torch.bfloat16))
    ref = dy * (y > 0)
    _close(out, ref.to(torch.bfloat16).float())


def test_sgd_step_kernel():
    g = torch.Generator().manual_seed(8)
    n = 10001  # odd size: exercises the scalar tail
    p = torch.randn(n, generator=g)
    grad = torch.randn(n, generator=g)
    m = torch.randn(n, generator=g)
    pg, gg, mg = p.clone().cuda(), grad.clone().cuda(), m.clone().cuda()
    fn.sgd_step(pg, gg, mg, 0.1, 0.9, 1e-4, 0.5)
    fn.sgd_step(p, grad, m, 0.1, 0.9, 1e-4, 0.5)
    torch.testing.assert_close(pg.cpu(), p, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(mg.cpu(), m, rtol=1e-5, atol=1e-6)


def test_fp16_dtype_path():
    from mi355x import amp
    with amp.autocast(torch.float16):
        x = torch.randn(4, 8, 8, 16).cuda().half()
        w = torch.randn(16, 16, 3, 3) * 0.2
        y = fn.conv2d(x, w.cuda(), None, 1, 1, None)
        assert y.dtype == torch.float16


def test_mfma_fragment_layout():
    """Asymmetric A/B elementwise check of the 32x32x16 fragment maps the
    conv/gemm kernels assume (guide §3: symmetric inputs hide transposes)."""
    g = torch.Generator().manual_seed(11)
    A = _qt(torch.randn(32, 16, generator=g))
    B = _qt(torch.randn(16, 32, generator=g))
    D = ops.ext().mfma_probe32(A.cuda(), B.cuda())
    torch.testing.assert_close(D.cpu(), A @ B, rtol=1e-2, atol=1e-2)
    # identity checks pin the exact row/col mapping
    I16 = torch.zeros(32, 16)
    I16[:16, :16] = torch.eye(16)
    D2 = ops.ext().mfma_probe32(I16.cuda(), B.cuda())
    torch.testing.assert_close(D2.cpu()[:16], B, rtol=1e-2, atol=1e-2)


def test_conv_bn_fused_stats_matches_separate():
    """conv2d_fwd_stats epilogue statistics == bn_stats over the conv
    output (fp32-accumulator stats vs bf16-tensor stats: loose tol)."""
    g = torch.Generator().manual_seed(21)
    x = _qt(torch.randn(8, 16, 16, 64, generator=g)).cuda().to(torch.bfloat16)
    w = _qt(torch.randn(128, 3, 3, 64, generator=g) * 0.1).cuda().to(torch.bfloat16)
    # w in [K,R,S,C] kernel layout for the raw ext call
    y, stats = ops.ext().conv2d_fwd_stats(x, w.contiguous(), 1, 1, 3, 3)
    ref = ops.ext().bn_stats(y)
    torch.testing.assert_close(stats, ref, rtol=2e-2, atol=2.0)


def test_resnet_block_fused_vs_unfused_training_step():
    """A BasicBlock forward via conv_bn (fused stats) matches the unfused
    composition."""
    from mi355x.models.resnet import BasicBlock
    torch.manual_seed(3)
    blk = BasicBlock(64, 64).cuda().train()
    x = torch.randn(4, 8, 8, 64).cuda().to(torch.bfloat16)
    y_fused = blk(x)
    # unfused reference: same weights, fresh running stats
    blk2 = BasicBlock(64, 64).cuda().train()
    blk2.load_state_dict(blk.state_dict())
    out = blk2.bn1(blk2.conv1(x))
    y_ref = blk2.bn2(blk2.conv2(out), residual=x)
    torch.testing.assert_close(y_fused.float(), y_ref.float(), rtol=5e-2,
                               atol=5e-2)


def test_stem_epilogue_stats_match_bn_stats():
    """The strip stem kernel's epilogue BN statistics equal a separate
    bn_stats pass over its output (conv->BN fusion for the 7x7/s2 stem)."""
    g = torch.Generator().manual_seed(11)
    x = torch.randn(4, 64, 64, 3, generator=g).cuda().to(torch.bfloat16)
    w = (torch.randn(64, 3, 7, 7, generator=g) * 0.1).cuda()
    y, stats = fn.conv2d_with_stats(x, w, stride=2, padding=3)
    assert stats is not None and stats.numel() == 128, "stem stats missing"
    ref = ops.ext().bn_stats(y.detach().contiguous())
    # the epilogue accumulates PRE-bf16-rounding fp32 values; bn_stats
    # reads the rounded tensor — a sqrt(M)-scaled rounding-noise gap
    # (observed ~0.3 abs on 4096-element channel sums), not an error
    torch.testing.assert_close(stats, ref, rtol=3e-2, atol=1.0)


def test_stem3_epilogue_stats_match_bn_stats():
    """Same check for the 3x3/s1 CIFAR stem geometry of the strip kernel
    (conv->BN fusion of ResNet-18/CIFAR's first layer)."""
    g = torch.Generator().manual_seed(12)
    x = torch.randn(4, 32, 32, 3, generator=g).cuda().to(torch.bfloat16)
    w = (torch.randn(64, 3, 3, 3, generator=g) * 0.2).cuda()
    y, stats = fn.conv2d_with_stats(x, w, stride=1, padding=1)
    assert stats is not None and stats.numel() == 128, "stem3 stats missing"
    ref = ops.ext().bn_stats(y.detach().contiguous())
    torch.testing.assert_close(stats, ref, rtol=3e-2, atol=1.0)
