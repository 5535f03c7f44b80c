"""nn.Module wrappers over the mi355x ops layer.

Parameters keep the torch-conventional shapes so checkpoints interchange
with the reference scripts' state_dicts; forward runs NHWC through the
mi355x kernels on GPU (bf16/fp16) and the plain-torch fp32 path on CPU.
"""

from __future__ import annotations

import math
import os

import torch
from torch import nn

from mi355x import ops


def to_model_layout(x: torch.Tensor) -> torch.Tensor:
    """NCHW input (reference contract) -> NHWC activation, 16-bit on GPU."""
    x = x.permute(0, 2, 3, 1)
    if x.is_cuda:
        x = x.to(ops.compute_dtype())
    return x.contiguous()


class Conv2d(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, bias=True, act=None):
        super().__init__()
        self.stride, self.padding, self.act = stride, padding, act
        self.weight = nn.Parameter(
            torch.empty(out_channels, in_channels, kernel_size, kernel_size))
        self.bias = nn.Parameter(torch.empty(out_channels)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            fan_in = self.weight.shape[1] * self.weight.shape[2] * self.weight.shape[3]
            bound = 1 / math.sqrt(fan_in)
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return ops.conv2d(x, self.weight, self.bias, self.stride,
                          self.padding, self.act)


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True, act=None):
        super().__init__()
        self.act = act
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.empty(out_features)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if self.bias is not None:
            bound = 1 / math.sqrt(self.weight.shape[1])
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x):
        return ops.linear(x, self.weight, self.bias, self.act)


class BatchNorm2d(nn.Module):
    """BN over NHWC with optional fused residual-add + ReLU epilogue.

    `sync` marks the layer for SyncBatchNorm: when a process group is
    installed (mi355x.parallel.sync_bn.convert / enable), batch statistics
    are all-reduced across ranks (BASELINE config 5).
    """

    def __init__(self, num_features, momentum=0.1, act=None):
        super().__init__()
        self.momentum, self.act = momentum, act
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        # a host-side int, not a device buffer: the torch-style device
        # tensor cost one captured GPU kernel per BN layer per step (20
        # launches/step on ResNet-18) just to count. state_dict key parity
        # is kept via the save/load hooks below.
        self._nbt = 0
        self.process_group = None  # set by sync_bn.enable()

    @property
    def num_batches_tracked(self):
        return torch.tensor(self._nbt, dtype=torch.long)

    @num_batches_tracked.setter
    def num_batches_tracked(self, v):
        self._nbt = int(v)

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        super()._save_to_state_dict(destination, prefix, keep_vars)
        destination[prefix + "num_batches_tracked"] = torch.tensor(
            self._nbt, dtype=torch.long)

    def _load_from_state_dict(self, state_dict, prefix, local_metadata,
                              strict, missing_keys, unexpected_keys,
                              error_msgs):
        # read (never pop — the caller's dict must stay loadable twice) and
        # clear the hook's unexpected-key report for the host-side counter
        key = prefix + "num_batches_tracked"
        if key in state_dict:
            self._nbt = int(state_dict[key])
        super()._load_from_state_dict(state_dict, prefix, local_metadata,
                                      strict, missing_keys, unexpected_keys,
                                      error_msgs)
        if key in unexpected_keys:
            unexpected_keys.remove(key)

    def forward(self, x, residual=None, stats=None):
        if self.training:
            self._nbt += 1
        return ops.batch_norm(x, self.weight, self.bias, self.running_mean,
                              self.running_var, self.training, self.momentum,
                              residual, self.act,
                              self.process_group if self.training else None,
                              stats)


def _fuse_stats():
    # default ON since round 2: with the shuffle-based stats fold, the
    # two-level slab reduce and the patch-kernel stats path it measures
    # +0.9% r18 / +2.8% r50 (was negative in round 1 when it knocked the
    # 3x3 convs off the patch kernel and LDS atomics serialized)
    return os.environ.get("MI355X_FUSE_BN", "1") == "1"


def conv_bn(conv: "Conv2d", bn: "BatchNorm2d", x, residual=None):
    """conv -> BN(+residual+ReLU). With MI355X_FUSE_BN=1 (default) the BN
    batch statistics come from the conv epilogue, skipping the separate
    full-tensor bn_stats pass (+0.9% r18 / +2.8% r50 after the round-2
    shuffle fold + two-level slab reduce; see _fuse_stats)."""
    if (_fuse_stats() and x.is_cuda and bn.training and conv.bias is None
            and conv.act is None
            # shapes with a stats-emitting MFMA path: C % 64 == 0 convs
            # and the C=3 stems (7x7/s2 ImageNet, 3x3/s1 CIFAR — strip
            # kernel epilogue stats, mi355x/csrc/stem_mfma.hip)
            and (conv.weight.shape[1] % 64 == 0
                 or (conv.weight.shape[1] == 3
                     and ((conv.weight.shape[2] == 7 and conv.stride == 2)
                          or (conv.weight.shape[2] == 3
                              and conv.stride == 1))))):
        from mi355x.ops import functional as F_

        bn._nbt += 1

        y, stats = F_.conv2d_with_stats(x, conv.weight, conv.stride,
                                        conv.padding)
        if stats.numel() == 0:
            stats = None
        return ops.batch_norm(y, bn.weight, bn.bias, bn.running_mean,
                              bn.running_var, bn.training, bn.momentum,
                              residual, bn.act, bn.process_group, stats)
    return bn(conv(x), residual=residual)


def _tap_on():
    return os.environ.get("MI355X_TAP", "1") != "0"


def _lazy_mode():
    # "1": all eligible consumers; "g": only gather-path consumers (1x1 /
    # strided convs); "0": off. DEFAULT OFF — a measured negative result:
    # even with vectorized Sc8 loads and pinned occupancy the transform
    # work on the conv staging paths costs more than the removed apply
    # pass saves (r50-224: off 9507, g 9198, all 8704 img/s; r18: off
    # 111.7k, all 98.7k). Kept as a tested opt-in for future work.
    return os.environ.get("MI355X_LAZY_BN", "0")


def _lazy_ok(bn: "BatchNorm2d", conv: "Conv2d", x) -> bool:
    """Lazy-BN (apply fused into the consuming conv) eligibility: GPU
    training, relu BN with no residual, MFMA conv shapes, fast-BN channel
    layout (C % 8 == 0 and 2048 % C == 0 for the mask-recompute backward)."""
    mode = _lazy_mode()
    if mode == "0":
        return False
    if mode == "g" and conv.weight.shape[2] != 1 and conv.stride == 1:
        return False  # 3x3/s1 consumer = patch kernel: keep materialized
    C = conv.weight.shape[1]
    return (x.is_cuda and bn.training and bn.act == "relu"
            and conv.bias is None and conv.act is None
            and C % 64 == 0 and conv.weight.shape[0] % 64 == 0
            and 2048 % C == 0)


def bn_then_conv(bn: "BatchNorm2d", conv: "Conv2d", x, stats=None,
                 want_stats=False):
    """bn(relu) -> conv with per-pair dispatch: lazy fusion when eligible,
    else materialized apply followed by a (possibly stats-emitting) conv.
    Returns (y, stats_out_or_None)."""
    if _lazy_ok(bn, conv, x) and x.requires_grad:
        return bn_conv_lazy(bn, conv, x, stats, want_stats)
    z = bn(x, stats=stats)
    if (want_stats and _fuse_stats() and z.is_cuda and bn.training
            and conv.bias is None and conv.act is None
            and conv.weight.shape[1] % 64 == 0):
        from mi355x.ops import functional as F_

        y, s2 = F_.conv2d_with_stats(z, conv.weight, conv.stride,
                                     conv.padding)
        if s2 is not None and s2.numel() == 0:
            s2 = None
        return y, s2
    return conv(z), None


def bn_conv_lazy(bn: "BatchNorm2d", conv: "Conv2d", x, stats=None,
                 want_stats=False):
    """relu(BN(x)) -> conv, with the BN apply fused into the conv's
    A-side loads (ops.functional.bn_conv): the normalized activation is
    never materialized. stats: precomputed (sum,sumsq) from the conv that
    PRODUCED x; want_stats: emit epilogue stats of this conv's output for
    the next BN. Returns (y, stats2_or_None)."""
    from mi355x.ops import functional as F_

    if bn.training:
        bn._nbt += 1
    out = F_.bn_conv(x, bn.weight, bn.bias, conv.weight, bn.running_mean,
                     bn.running_var, bn.momentum, stats,
                     bn.process_group if bn.training else None,
                     conv.stride, conv.padding, want_stats)
    if want_stats:
        y, s2 = out
        return y, (s2 if s2 is not None and s2.numel() else None)
    return out, None


def conv_bn_tap(conv: "Conv2d", bn: "BatchNorm2d", x, residual=None):
    """conv -> BN returning (out, tap). The block's shortcut consumes
    `tap` instead of x directly, so the residual-junction gradient fuses
    into this conv's dgrad epilogue (ops.functional._ConvTapFn) instead
    of a separate full-tensor add; the conv epilogue also emits the BN
    statistics (no separate bn_stats pass). Falls back to plain conv_bn
    semantics (tap = x) on CPU / non-MFMA shapes / MI355X_TAP=0."""
    if not (_tap_on() and x.is_cuda and conv.bias is None
            and conv.act is None):
        return conv_bn(conv, bn, x, residual=residual), x
    from mi355x.ops import functional as F_

    if _fuse_stats() and bn.training and conv.weight.shape[1] % 64 == 0:
        y, tap, stats = F_.conv2d_tap_stats(x, conv.weight, conv.stride,
                                            conv.padding)
        return bn(y, residual=residual, stats=stats), tap
    y, tap = F_.conv2d_tap(x, conv.weight, conv.stride, conv.padding)
    return bn(y, residual=residual), tap


class MaxPool2d(nn.Module):
    def __init__(self, kernel_size, stride=None, padding=0):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride or kernel_size
        self.padding = padding

    def forward(self, x):
        return ops.max_pool2d(x, self.kernel_size, self.stride, self.padding)
