"""Fused SGD with momentum over the flat parameter/grad/momentum buffers.

Semantics of the reference's optim.SGD(lr=0.001, momentum=0.9)
(/root/reference/cifar_example.py:64): v = mu*v + g; p -= lr*v — but as
ONE HIP kernel over the whole flat buffer per step instead of per-tensor
launches. grad_scale folds the DDP 1/world_size average (and the AMP
1/loss_scale) into the same kernel.
"""

from __future__ import annotations

import torch

from mi355x import ops
from mi355x.ops import functional as _fn
from mi355x.parallel.flat import FlatState


class SGD:
    def __init__(self, flat: FlatState, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, grad_scale: float = 1.0):
        self.flat = flat
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.grad_scale = grad_scale

    def step(self):
        ops.sgd_step(self.flat.flat_param, self.flat.flat_grad,
                     self.flat.flat_momentum, self.lr, self.momentum,
                     self.weight_decay, self.grad_scale)
        _fn.bump_cache_epoch()  # invalidate cached 16-bit weight copies

    def zero_grad(self, set_to_none: bool = False):
        self.flat.zero_grad()

    def state_dict(self):
        return {"momentum_buffer": self.flat.flat_momentum,
                "lr": self.lr, "momentum": self.momentum,
                "weight_decay": self.weight_decay}

    def load_state_dict(self, sd):
        self.flat.flat_momentum.copy_(sd["momentum_buffer"])
        self.lr = sd["lr"]
        self.momentum = sd["momentum"]
        self.weight_decay = sd["weight_decay"]
