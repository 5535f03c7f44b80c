"""Flat parameter / gradient / momentum buffers.

All trainable parameters of a model are re-pointed into ONE fp32 flat
buffer, with a matching flat gradient buffer (param.grad pre-assigned as
views, so autograd accumulates straight into it) and a flat momentum
buffer. Consequences, all MI355X-motivated:

- the optimizer step is ONE fused kernel over three flat buffers
  ("single fused multi-tensor HIP kernel over the flat grad
  buffer") instead of ~60 per-tensor launches;
- zero_grad is one memset;
- DDP gradient buckets are contiguous SLICES of the flat grad buffer —
  the all-reduce payload needs no copy in or out (the reference's
  torch-DDP reducer copies grads into bucket flats; ours IS the flat).

Parameters are laid out in REVERSE registration order, which approximates
autograd's backward completion order, so bucket 0 (first slice) becomes
ready earliest and its all-reduce overlaps the rest of backward.
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch
from torch import nn


@dataclass
class Bucket:
    index: int
    start: int          # element offset into the flat buffers
    end: int
    params: list = field(default_factory=list)


class FlatState:
    def __init__(self, module: nn.Module, bucket_bytes: int = 25 * 1 << 20,
                 first_bucket_bytes: int | None = 1 << 20):
        params = [p for p in module.parameters() if p.requires_grad]
        if not params:
            raise ValueError("module has no trainable parameters")
        dev = params[0].device
        # reverse registration order ~ backward completion order
        ordered = list(reversed(params))
        total = sum(p.numel() for p in ordered)
        self.flat_param = torch.empty(total, dtype=torch.float32, device=dev)
        self.flat_grad = torch.zeros(total, dtype=torch.float32, device=dev)
        self.flat_momentum = torch.zeros(total, dtype=torch.float32, device=dev)
        self.params = ordered
        self.offsets: dict[int, tuple[int, int]] = {}

        off = 0
        for p in ordered:
            n = p.numel()
            self.flat_param[off:off + n].copy_(p.detach().reshape(-1))
            p.data = self.flat_param[off:off + n].view(p.shape)
            p.grad = self.flat_grad[off:off + n].view(p.shape)
            self.offsets[id(p)] = (off, off + n)
            off += n

        # contiguous buckets over the flat layout
        self.buckets: list[Bucket] = []
        limit = (first_bucket_bytes or bucket_bytes) // 4
        b = Bucket(0, 0, 0)
        for p in ordered:
            s, e = self.offsets[id(p)]
            if b.params and (e - b.start) > limit:
                b.end = s
                self.buckets.append(b)
                b = Bucket(len(self.buckets), s, s)
                limit = bucket_bytes // 4
            b.params.append(p)
        b.end = total
        self.buckets.append(b)
        self.bucket_of: dict[int, Bucket] = {}
        for bk in self.buckets:
            for p in bk.params:
                self.bucket_of[id(p)] = bk

    @property
    def numel(self) -> int:
        return self.flat_param.numel()

    def zero_grad(self):
        self.flat_grad.zero_()

    def grad_slice(self, bucket: Bucket) -> torch.Tensor:
        return self.flat_grad[bucket.start:bucket.end]
