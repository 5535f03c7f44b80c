"""SyncBatchNorm: cross-rank batch statistics (BASELINE config 5).

Our BatchNorm2d already computes per-channel (sum, sumsq) partials in one
kernel; SyncBN is the same layer with those partials (and the backward's
(sum_dy_xhat, sum_dy)) all-reduced over the process group — a single
2xC-float RCCL all-reduce per BN layer per direction.
"""

from __future__ import annotations

import torch.distributed as dist
from torch import nn

from mi355x.models.layers import BatchNorm2d


def enable(module: nn.Module, process_group=None) -> nn.Module:
    """Turn every mi355x BatchNorm2d in `module` into a SyncBatchNorm by
    installing the process group (None -> the default group)."""
    if not dist.is_initialized() or dist.get_world_size(process_group) <= 1:
        return module
    pg = process_group if process_group is not None else dist.group.WORLD
    for m in module.modules():
        if isinstance(m, BatchNorm2d):
            m.process_group = pg
    return module


def disable(module: nn.Module) -> nn.Module:
    for m in module.modules():
        if isinstance(m, BatchNorm2d):
            m.process_group = None
    return module
