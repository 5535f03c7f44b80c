"""Functional op layer: HIP/CDNA4 kernels on GPU, plain-torch fp32 on CPU.

Activation layout on GPU is NHWC end to end (coalesced channel-minor access
for CDNA4 implicit-GEMM conv; see mi355x/csrc/). Parameters are stored in
the PyTorch-conventional shapes/dtypes ([K,C,R,S] fp32 conv weight, etc.) so
state_dicts interchange with the reference scripts
(/root/reference/cifar_example.py:92-93 checkpoint format); 16-bit KRSC
copies for the kernels are cached per parameter version.

CPU tensors run a plain differentiable fp32 torch path — this is both the
BASELINE config-1 (CPU) execution mode and the numerics reference that the
GPU kernels are unit-tested against. CUDA tensors REQUIRE the mi355x._C
extension (no silent PyTorch fallback on GPU).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ._ext import ext

BN_EPS = 1e-5

_ACT_NONE = 0
_ACT_RELU = 1


def compute_dtype() -> torch.dtype:
    """Current 16-bit compute dtype for GPU kernels (bf16 unless amp says fp16)."""
    from mi355x import amp

    return amp.get_compute_dtype()


# ---------------------------------------------------------------------------
# cached 16-bit weight copies
# ---------------------------------------------------------------------------

# Bumped by the optimizer after each step (flat-param in-place updates do not
# advance the per-parameter ._version, so the cache keys on both).
_cache_epoch = 0


def bump_cache_epoch() -> None:
    global _cache_epoch
    _cache_epoch += 1


def _w16_conv(weight: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    """fp32 [K,C,R,S] parameter -> cached 16-bit [K,R,S,C] kernel copy."""
    key = (weight._version, _cache_epoch, dtype)
    cache = getattr(weight, "_mi355x_w16", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    if weight.is_cuda:
        like = torch.empty(0, dtype=dtype, device=weight.device)
        K, C = weight.shape[0], weight.shape[1]
        if C % 64 != 0 and K % 64 == 0:
            # generic small-C MFMA path wants [KO, KGP] zero-padded
            w16 = ext().cast_permute_krsc_pad(weight.detach(), like)
        else:
            w16 = ext().cast_permute_krsc(weight.detach(), like)
    else:
        w16 = weight.detach().to(dtype).permute(0, 2, 3, 1).contiguous()
    weight._mi355x_w16 = (key, w16)
    return w16


def _w16_conv_flip(weight: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    """fp32 [K,C,R,S] parameter -> cached 16-bit [R,S,C,K] transposed copy
    (wt[r,s,c,k] = w[k,c,r,s]) — the dgrad MFMA kernel's B-operand layout.
    No spatial flip: the kernel's transposed-window gather pairs source row
    (h+pad-r)/stride directly with weight tap r (conv_mfma.hip)."""
    key = (weight._version, _cache_epoch, dtype)
    cache = getattr(weight, "_mi355x_wflip", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    if weight.is_cuda:
        wf = ext().cast_permute_rsck(
            weight.detach(), torch.empty(0, dtype=dtype, device=weight.device))
    else:
        wf = weight.detach().to(dtype).permute(2, 3, 1, 0).contiguous()
    weight._mi355x_wflip = (key, wf)
    return wf


def _w16_linear(weight: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    """fp32 [N,K] parameter -> cached 16-bit [N,K] kernel copy."""
    key = (weight._version, _cache_epoch, dtype)
    cache = getattr(weight, "_mi355x_w16", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    w16 = weight.detach().to(dtype).contiguous()
    weight._mi355x_w16 = (key, w16)
    return w16


def _w16_linear_t(weight: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    """fp32 [N,K] parameter -> cached 16-bit [K,N] transpose (the MFMA
    dgrad's B operand: dx = dy @ w == gemm_nt(dy, wT))."""
    key = (weight._version, _cache_epoch, dtype)
    cache = getattr(weight, "_mi355x_wT", None)
    if cache is not None and cache[0] == key:
        return cache[1]
    wt = weight.detach().to(dtype).t().contiguous()
    weight._mi355x_wT = (key, wt)
    return wt


def _relu_mask_bwd(dy: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """dy masked by y>0, via the extension's elementwise kernel."""
    return ext().relu_bwd(dy.contiguous(), y)


# ---------------------------------------------------------------------------
# conv2d (NHWC implicit GEMM)
# ---------------------------------------------------------------------------


class _ConvFn(torch.autograd.Function):
    """want_stats: the conv epilogue additionally emits BN (sum,sumsq)
    per-channel statistics of its output (conv->BN fusion — skips the
    separate bn_stats pass). The stats output is marked non-differentiable:
    the BN backward formula already carries the full d(stats)/d(x) terms.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding, act, want_stats=False):
        dtype = x.dtype
        w16 = _w16_conv(weight, dtype)
        stats = None
        if want_stats:
            y, stats = ext().conv2d_fwd_stats(x, w16, stride, padding,
                                              weight.shape[2],
                                              weight.shape[3])
        else:
            b32 = bias.detach().float() if bias is not None else torch.empty(0, device=x.device)
            y = ext().conv2d_fwd(x, w16, b32, stride, padding, act,
                                 weight.shape[2], weight.shape[3])
        ctx.save_for_backward(x, w16, y)
        ctx.weight_ref = weight  # for the cached dgrad weight-flip
        ctx.conf = (stride, padding, act, bias is not None, weight.shape)
        if want_stats:
            if stats is None:
                stats = torch.empty(0, device=x.device, dtype=torch.float32)
            ctx.mark_non_differentiable(stats)
            return y, stats
        return y

    @staticmethod
    def backward(ctx, dy, *unused_stats_grad):
        x, w16, y = ctx.saved_tensors
        stride, padding, act, has_bias, wshape = ctx.conf
        dy = dy.contiguous()
        if act == _ACT_RELU:
            dy = _relu_mask_bwd(dy, y)
        dx = None
        if ctx.needs_input_grad[0]:
            wflip = _w16_conv_flip(ctx.weight_ref, dy.dtype)
            dx = ext().conv2d_dgrad(dy, wflip, stride, padding,
                                    x.shape[1], x.shape[2],
                                    torch.empty(0, device=x.device,
                                                dtype=x.dtype))
        dw = None
        if ctx.needs_input_grad[1]:
            # wgrad kernels emit fp32 [K,C,R,S] directly (parameter layout)
            dw = ext().conv2d_wgrad(x, dy, wshape[2], wshape[3], stride,
                                    padding)
        db = None
        if has_bias and ctx.needs_input_grad[2]:
            db = dy.float().sum(dim=(0, 1, 2))
        return dx, dw, db, None, None, None, None


class _ConvTapFn(torch.autograd.Function):
    """conv2d with a residual TAP: forward returns (y, tap) where tap
    aliases the input. A ResNet block feeds `tap` (not x) to its shortcut
    path, so the junction gradient arrives HERE as d_tap in the same
    backward call as dy and is fused into the dgrad epilogue
    (dx = dgrad(dy) + d_tap) — autograd's separate full-tensor add at the
    junction disappears (r50-224 profile: 19 adds/step, 4.7% of step).
    The addin kernel variant is a separate compile-time instantiation, so
    non-tap convs run the exact same code as before."""

    @staticmethod
    def forward(ctx, x, weight, stride, padding):
        w16 = _w16_conv(weight, x.dtype)
        e = torch.empty(0, device=x.device)
        y = ext().conv2d_fwd(x, w16, e, stride, padding, _ACT_NONE,
                             weight.shape[2], weight.shape[3])
        ctx.save_for_backward(x, w16)
        ctx.weight_ref = weight
        ctx.conf = (stride, padding, weight.shape)
        return y, x.view_as(x)

    @staticmethod
    def backward(ctx, dy, dtap):
        x, w16 = ctx.saved_tensors
        stride, padding, wshape = ctx.conf
        dy = dy.contiguous()
        dx = None
        if ctx.needs_input_grad[0]:
            wflip = _w16_conv_flip(ctx.weight_ref, dy.dtype)
            add = (dtap.contiguous() if dtap is not None
                   else torch.empty(0, device=x.device, dtype=x.dtype))
            dx = ext().conv2d_dgrad(dy, wflip, stride, padding,
                                    x.shape[1], x.shape[2], add)
        elif dtap is not None:
            dx = dtap
        dw = None
        if ctx.needs_input_grad[1]:
            dw = ext().conv2d_wgrad(x, dy, wshape[2], wshape[3], stride,
                                    padding)
        return dx, dw, None, None


def conv2d_tap(x, weight, stride=1, padding=0):
    """conv2d returning (y, tap): route the block's shortcut through `tap`
    so its gradient fuses into this conv's dgrad epilogue (GPU training,
    MFMA shapes only; CPU keeps the plain double-use add semantics)."""
    if (x.is_cuda and x.requires_grad and weight.shape[1] % 64 == 0
            and weight.shape[0] % 64 == 0):
        return _ConvTapFn.apply(x, weight, stride, padding)
    return conv2d(x, weight, None, stride, padding, None), x


class _ConvTapStatsFn(torch.autograd.Function):
    """_ConvTapFn + epilogue BN stats: (y, tap, stats[2,K]) — the lazy-BN
    block entry (conv1 with junction-grad tap and stats for bn1)."""

    @staticmethod
    def forward(ctx, x, weight, stride, padding):
        w16 = _w16_conv(weight, x.dtype)
        y, stats = ext().conv2d_fwd_stats(x, w16, stride, padding,
                                          weight.shape[2], weight.shape[3])
        if stats is None:
            stats = torch.empty(0, device=x.device, dtype=torch.float32)
        ctx.save_for_backward(x, w16)
        ctx.weight_ref = weight
        ctx.conf = (stride, padding, weight.shape)
        ctx.mark_non_differentiable(stats)
        return y, x.view_as(x), stats

    @staticmethod
    def backward(ctx, dy, dtap, _unused):
        x, w16 = ctx.saved_tensors
        stride, padding, wshape = ctx.conf
        dy = dy.contiguous()
        dx = None
        if ctx.needs_input_grad[0]:
            wflip = _w16_conv_flip(ctx.weight_ref, dy.dtype)
            add = (dtap.contiguous() if dtap is not None
                   else torch.empty(0, device=x.device, dtype=x.dtype))
            dx = ext().conv2d_dgrad(dy, wflip, stride, padding,
                                    x.shape[1], x.shape[2], add)
        elif dtap is not None:
            dx = dtap
        dw = None
        if ctx.needs_input_grad[1]:
            dw = ext().conv2d_wgrad(x, dy, wshape[2], wshape[3], stride,
                                    padding)
        return dx, dw, None, None


def conv2d_tap_stats(x, weight, stride=1, padding=0):
    """(y, tap, stats_or_None) — see _ConvTapStatsFn."""
    y, tap, stats = _ConvTapStatsFn.apply(x, weight, stride, padding)
    return y, tap, (stats if stats.numel() else None)


class _BNConvFn(torch.autograd.Function):
    """Lazy BN: the BatchNorm apply (normalize + affine + ReLU) is fused
    into the CONSUMING conv's A-side load — the normalized activation z
    is never materialized (saves one full read + write of the tensor per
    internal BN; the bn_apply pass was 6-10% of the r18/r50 step).

    forward(x=conv1 output, bn params, conv2 weight) -> conv2(relu(bn(x)))
    [+ epilogue stats for the NEXT BN]. backward recomputes the ReLU mask
    from x (z > 0  <=>  x*asc+ash > 0), runs conv2's dgrad/wgrad (wgrad
    re-applies the transform on its x loads) and the standard fused BN
    backward."""

    @staticmethod
    def forward(ctx, x, gamma, beta, conv_weight, running_mean, running_var,
                momentum, stats, process_group, stride, padding, want_stats):
        N, H, W, C = x.shape
        m_local = N * H * W
        s = stats if stats is not None else ext().bn_stats(x)
        m_total = m_local
        if process_group is not None:
            import torch.distributed as dist

            dist.all_reduce(s, group=process_group)
            m_total = m_local * dist.get_world_size(process_group)
        empty = torch.empty(0, device=x.device)
        mi = ext().bn_finalize(
            s, running_mean if running_mean is not None else empty,
            running_var if running_var is not None else empty,
            float(m_total), momentum, BN_EPS)
        mean, invstd = mi[0], mi[1]
        g32 = gamma.detach().float()
        asc = g32 * invstd
        ash = beta.detach().float() - mean * asc
        w16 = _w16_conv(conv_weight, x.dtype)
        y, stats2 = ext().conv2d_fwd_scaled(x, w16, asc, ash, stride,
                                            padding, want_stats)
        ctx.save_for_backward(x, w16, mean, invstd, gamma, asc, ash)
        ctx.weight_ref = conv_weight
        ctx.conf = (stride, padding, conv_weight.shape, m_total,
                    process_group)
        if want_stats:
            ctx.mark_non_differentiable(stats2)
            return y, stats2
        return y

    @staticmethod
    def backward(ctx, dy2, *unused_stats_grad):
        x, w16, mean, invstd, gamma, asc, ash = ctx.saved_tensors
        stride, padding, wshape, m_total, pg = ctx.conf
        dy2 = dy2.contiguous()
        efx = torch.empty(0, device=x.device, dtype=x.dtype)
        # conv2 backward w.r.t. the virtual z
        dw = None
        if ctx.needs_input_grad[3]:
            dw = ext().conv2d_wgrad_scaled(x, asc, ash, dy2, wshape[2],
                                           wshape[3], stride, padding)
        wflip = _w16_conv_flip(ctx.weight_ref, dy2.dtype)
        dz = ext().conv2d_dgrad(dy2, wflip, stride, padding, x.shape[1],
                                x.shape[2], efx)
        # fused BN backward with the ReLU mask recomputed from x
        emask = torch.empty(0, device=x.device, dtype=torch.uint8)
        r = ext().bn_bwd_reduce(x, dz, efx, mean, invstd, emask, asc, ash)
        if pg is not None:
            import torch.distributed as dist

            dist.all_reduce(r, group=pg)
        dgamma, dbeta = r[0], r[1]
        dx, _ = ext().bn_bwd_dx(x, dz, efx, mean, invstd,
                                gamma.detach().float(), dgamma, dbeta,
                                float(m_total), False, emask, asc, ash)
        return (dx, dgamma, dbeta, dw, None, None, None, None, None, None,
                None, None)


def bn_conv(x, gamma, beta, conv_weight, running_mean, running_var,
            momentum=0.1, stats=None, process_group=None, stride=1,
            padding=0, want_stats=False):
    """Fused BN(relu) -> conv2d (lazy-BN consumer fusion). Returns y, or
    (y, stats2) when want_stats. GPU/MFMA shapes only; callers gate."""
    return _BNConvFn.apply(x, gamma, beta, conv_weight, running_mean,
                           running_var, momentum, stats, process_group,
                           stride, padding, want_stats)


def conv2d_with_stats(x, weight, stride=1, padding=0):
    """GPU conv returning (y, bn_stats[2,C] or None) — the fused
    conv->BN entry used by the ResNet blocks. stats is None when the
    shape has no stats-emitting kernel path (caller falls back to
    bn_stats)."""
    return _ConvFn.apply(x, weight, None, stride, padding, _ACT_NONE, True)


def conv2d(x, weight, bias=None, stride=1, padding=0, act=None):
    """2-D convolution. x: NHWC; weight: fp32 [K,C,R,S] (torch layout).

    act: None | 'relu' fused into the epilogue (and into the kernel backward
    mask). Reference call sites: /root/reference/cifar_example.py:20-29.
    """
    a = _ACT_RELU if act == "relu" else _ACT_NONE
    if x.is_cuda:
        return _ConvFn.apply(x, weight, bias, stride, padding, a)
    y = F.conv2d(x.permute(0, 3, 1, 2), weight, bias, stride, padding)
    y = y.permute(0, 2, 3, 1)
    if a == _ACT_RELU:
        y = F.relu(y)
    return y


# ---------------------------------------------------------------------------
# linear (MFMA GEMM)
# ---------------------------------------------------------------------------


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, act):
        w16 = _w16_linear(weight, x.dtype)
        b32 = bias.detach().float() if bias is not None else torch.empty(0, device=x.device)
        y = ext().linear_fwd(x, w16, b32, act)
        ctx.save_for_backward(x, w16, y)
        ctx.weight_ref = weight  # for the cached dgrad transpose
        ctx.conf = (act, bias is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w16, y = ctx.saved_tensors
        act, has_bias = ctx.conf
        dy = dy.contiguous()
        if act == _ACT_RELU:
            dy = _relu_mask_bwd(dy, y)
        dx = None
        if ctx.needs_input_grad[0]:
            wt = _w16_linear_t(ctx.weight_ref, dy.dtype)
            dx = ext().linear_dgrad(dy, w16, wt)
        dw = ext().linear_wgrad(x, dy) if ctx.needs_input_grad[1] else None
        db = dy.float().sum(dim=0) if (has_bias and ctx.needs_input_grad[2]) else None
        return dx, dw, db, None


def linear(x, weight, bias=None, act=None):
    """y = x @ W^T + b. x: [M,K]; weight: fp32 [N,K]."""
    a = _ACT_RELU if act == "relu" else _ACT_NONE
    if x.is_cuda:
        return _LinearFn.apply(x, weight, bias, a)
    y = F.linear(x, weight, bias)
    if a == _ACT_RELU:
        y = F.relu(y)
    return y


# ---------------------------------------------------------------------------
# batch norm (+ residual add + relu, fused)
# ---------------------------------------------------------------------------


class _BNFn(torch.autograd.Function):
    """Training-mode batch norm over NHWC with optional fused residual add
    and ReLU. Stats in fp32. Optionally SyncBN: partial (sum,sumsq) and the
    backward (sum_dy, sum_dy_xhat) reductions are all-reduced over the
    process group (BASELINE config 5)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, residual, running_mean, running_var,
                momentum, act, process_group, stats=None):
        N, H, W, C = x.shape
        m_local = N * H * W
        # stats precomputed by the producing conv's epilogue when fused
        s = stats if stats is not None else ext().bn_stats(x)
        m_total = m_local
        if process_group is not None:
            import torch.distributed as dist

            dist.all_reduce(s, group=process_group)
            m_total = m_local * dist.get_world_size(process_group)
        empty = torch.empty(0, device=x.device)
        mi = ext().bn_finalize(
            s, running_mean if running_mean is not None else empty,
            running_var if running_var is not None else empty,
            float(m_total), momentum, BN_EPS)  # [2,C]: mean, invstd
        mean, invstd = mi[0], mi[1]
        res = residual if residual is not None else torch.empty(0, device=x.device, dtype=x.dtype)
        y, mask = ext().bn_apply(x, mean, invstd, gamma.detach().float(),
                                 beta.detach().float(), res, act, True)
        # mask: 1-bit-per-element "y > 0" (fast path + relu) — backward
        # reads it instead of re-reading y (1/16 the bytes)
        if mask is None:
            mask = torch.empty(0, device=x.device, dtype=torch.uint8)
        ctx.save_for_backward(x, y, mean, invstd, gamma, mask)
        ctx.conf = (act, residual is not None, m_total, process_group)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, mean, invstd, gamma, mask = ctx.saved_tensors
        act, has_res, m_total, pg = ctx.conf
        dy = dy.contiguous()
        # ReLU mask fused into both backward kernels (no standalone pass)
        y_arg = y if act == _ACT_RELU else torch.empty(0, device=x.device,
                                                       dtype=x.dtype)
        ef = torch.empty(0, device=x.device)
        r = ext().bn_bwd_reduce(x, dy, y_arg, mean, invstd, mask, ef, ef)
        if pg is not None:
            import torch.distributed as dist

            dist.all_reduce(r, group=pg)
        dgamma, dbeta = r[0], r[1]
        want_dres = has_res and act == _ACT_RELU
        dx, dres = ext().bn_bwd_dx(x, dy, y_arg, mean, invstd,
                                   gamma.detach().float(), dgamma, dbeta,
                                   float(m_total), want_dres, mask, ef, ef)
        if has_res and not want_dres:
            dres = dy  # no activation: residual grad is dy itself
        elif not has_res:
            dres = None
        return dx, dgamma, dbeta, dres, None, None, None, None, None, None


def batch_norm(x, gamma, beta, running_mean, running_var, training,
               momentum=0.1, residual=None, act=None, process_group=None,
               stats=None):
    """BatchNorm over NHWC with optional fused residual-add + ReLU.

    Matches torch BN semantics (momentum convention, unbiased running var).
    process_group != None => SyncBatchNorm (stats all-reduced over ranks).
    stats: precomputed (sum,sumsq) from the producing conv's epilogue.
    """
    a = _ACT_RELU if act == "relu" else _ACT_NONE
    if x.is_cuda:
        if training:
            return _BNFn.apply(x, gamma, beta, residual, running_mean,
                               running_var, momentum, a, process_group,
                               stats)
        mean = running_mean.float()
        invstd = torch.rsqrt(running_var.float() + BN_EPS)
        res = residual if residual is not None else torch.empty(0, device=x.device, dtype=x.dtype)
        # eval: no backward, skip the relu-mask allocation + stores
        return ext().bn_apply(x.contiguous(), mean, invstd,
                              gamma.detach().float(), beta.detach().float(),
                              res, a, False)[0]
    xc = x.permute(0, 3, 1, 2)
    y = F.batch_norm(xc, running_mean, running_var, gamma, beta, training,
                     momentum, BN_EPS).permute(0, 2, 3, 1)
    if residual is not None:
        y = y + residual
    if a == _ACT_RELU:
        y = F.relu(y)
    return y


# ---------------------------------------------------------------------------
# pooling
# ---------------------------------------------------------------------------


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        y, idx = ext().maxpool_fwd(x, kernel, stride, padding)
        ctx.save_for_backward(idx)
        ctx.conf = (x.shape, kernel, stride, padding)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        xshape, kernel, stride, padding = ctx.conf
        dx = ext().maxpool_bwd(dy.contiguous(), idx, xshape[1], xshape[2],
                               kernel, stride, padding)
        return dx, None, None, None


def max_pool2d(x, kernel, stride=None, padding=0):
    """Max pooling over NHWC. Reference: /root/reference/cifar_example.py:21."""
    stride = stride or kernel
    if x.is_cuda:
        return _MaxPoolFn.apply(x, kernel, stride, padding)
    y = F.max_pool2d(x.permute(0, 3, 1, 2), kernel, stride, padding)
    return y.permute(0, 2, 3, 1)


class _GAPFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.conf = x.shape
        return ext().global_avg_pool(x)

    @staticmethod
    def backward(ctx, dy):
        N, H, W, C = ctx.conf
        dx = (dy / (H * W)).reshape(N, 1, 1, C).expand(N, H, W, C).contiguous()
        return dx


def global_avg_pool(x):
    """NHWC [N,H,W,C] -> [N,C] mean over H,W."""
    if x.is_cuda:
        return _GAPFn.apply(x)
    return x.mean(dim=(1, 2))


# ---------------------------------------------------------------------------
# loss
# ---------------------------------------------------------------------------


class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        loss, lse = ext().cross_entropy_fwd(logits, target)
        ctx.save_for_backward(logits, target, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, target, lse = ctx.saved_tensors
        # dloss read on-device (hipGraph-capturable: no host sync)
        if not (torch.is_tensor(dloss) and dloss.is_cuda):
            dloss = torch.as_tensor(dloss, dtype=torch.float32,
                                    device=logits.device)
        dlogits = ext().cross_entropy_bwd(logits, target, lse,
                                          dloss.float().reshape(()))
        return dlogits, None


def cross_entropy(logits, target):
    """Mean cross-entropy. logits [B,C] (16-bit on GPU), target int64 [B].
    Reference call site: /root/reference/cifar_example.py:63,78."""
    if logits.is_cuda:
        return _CEFn.apply(logits, target)
    return F.cross_entropy(logits.float(), target)


# ---------------------------------------------------------------------------
# fused optimizer step
# ---------------------------------------------------------------------------


def sgd_step(flat_param, flat_grad, flat_momentum, lr, momentum,
             weight_decay=0.0, grad_scale=1.0):
    """v = mu*v + (g*gs + wd*p); p -= lr*v — one kernel over the flat fp32
    buffers (reference semantics: optim.SGD(lr,momentum),
    /root/reference/cifar_example.py:64)."""
    if flat_param.is_cuda:
        ext().sgd_step(flat_param, flat_grad, flat_momentum, lr, momentum,
                       weight_decay, grad_scale)
        return
    g = flat_grad.mul(grad_scale)
    if weight_decay != 0.0:
        g = g.add(flat_param, alpha=weight_decay)
    flat_momentum.mul_(momentum).add_(g)
    flat_param.add_(flat_momentum, alpha=-lr)
