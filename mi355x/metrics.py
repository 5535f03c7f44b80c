"""Distributed metrics (SURVEY.md N14 — torchmetrics is not installed).

DistAccuracy mirrors the reference's
torchmetrics.Accuracy(dist_sync_on_step=True) usage
(/root/reference/cifar_example_ddp.py:124-136): accumulates
(correct, total) and all-reduces the two int64 counters across ranks —
either on every update (dist_sync_on_step) or once at compute().
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class DistAccuracy:
    def __init__(self, dist_sync_on_step: bool = False, device=None):
        self.dist_sync_on_step = dist_sync_on_step
        self.device = device or torch.device("cpu")
        self.reset()

    def reset(self):
        self.state = torch.zeros(2, dtype=torch.long, device=self.device)

    def _sync(self, t):
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(t)
        return t

    def update(self, preds: torch.Tensor, target: torch.Tensor):
        if preds.dim() > 1:
            preds = preds.argmax(dim=-1)
        delta = torch.stack([
            (preds == target).sum().to(self.state.device),
            torch.tensor(target.numel(), device=self.state.device),
        ])
        if self.dist_sync_on_step:
            delta = self._sync(delta.clone())
        self.state += delta

    def compute(self) -> float:
        s = self.state if self.dist_sync_on_step else self._sync(self.state.clone())
        return (s[0].float() / s[1].clamp(min=1).float()).item()
