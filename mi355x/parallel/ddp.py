"""From-scratch DDP engine: flat-bucket gradient reducer over RCCL/xGMI.

Re-implements what torch.nn.parallel.DistributedDataParallel's C++
c10d::Reducer does for the reference script
(/root/reference/cifar_example_ddp.py:83), MI355X-first:

- at construction, ONE broadcast of the flat fp32 parameter buffer from
  rank 0 (the reference broadcasts per-bucket), plus the module buffers
  (BN running stats);
- per-parameter post-accumulate-grad hooks mark bucket readiness during
  backward; complete buckets launch async all-reduces on contiguous
  flat-grad slices, overlapped with the remaining backward. Buckets are
  launched in FIXED index order on every rank (a bucket launches only
  after all lower-index buckets have) so collectives can never interleave
  differently across ranks — the same invariant torch's reducer enforces;
- xGMI is 7 point-to-point links/GPU, so bucket size is a first-class
  tunable (MI355X_BUCKET_MB env or ctor arg);
- gradients are reduced as SUMs; the 1/world_size averaging is folded
  into the fused SGD step (grad_scale) instead of a separate pass.

Comm backends: on GPU the default is our native RCCL communicator
(mi355x/parallel/rccl.py + csrc/rccl_comm.cpp) on its own high-priority
HIP stream; MI355X_COMM=torch selects torch.distributed collectives
instead, and CPU/gloo always uses torch.distributed (multi-process CPU
tests run the identical reducer logic).
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist
from torch import nn

from .flat import FlatState


class _TorchComm:
    def __init__(self, process_group):
        self.pg = process_group

    def broadcast(self, t, root=0):
        dist.broadcast(t, src=root, group=self.pg)

    def all_reduce_async(self, t):
        return dist.all_reduce(t, async_op=True, group=self.pg)

    def finish(self, works):
        for w in works:
            w.wait()


class _NativeComm:
    def __init__(self):
        from .rccl import native_comm

        self.c = native_comm()

    def broadcast(self, t, root=0):
        if t.is_cuda:
            self.c.broadcast(t.contiguous(), root)
            self.c.wait()
        else:  # module buffers may live on CPU in odd setups
            dist.broadcast(t, src=root)

    def all_reduce_async(self, t):
        self.c.all_reduce(t)
        return None

    def finish(self, works):
        self.c.wait()


class DistributedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, flat: FlatState | None = None,
                 bucket_mb: float | None = None,
                 first_bucket_mb: float = 1.0,
                 process_group=None, comm: str | None = None):
        super().__init__()
        if bucket_mb is None:
            bucket_mb = float(os.environ.get("MI355X_BUCKET_MB", "25"))
        self.module = module
        self.process_group = process_group
        self.flat = flat or FlatState(
            module,
            bucket_bytes=int(bucket_mb * (1 << 20)),
            first_bucket_bytes=int(first_bucket_mb * (1 << 20)),
        )
        self.world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.grad_scale = 1.0 / self.world_size
        comm = comm or os.environ.get("MI355X_COMM", "rccl")
        use_native = (comm == "rccl"
                      and self.flat.flat_param.is_cuda
                      and self.world_size > 1)
        if use_native:
            try:
                self.comm = _NativeComm()
            except Exception as e:  # construction-time failure only; fall
                # back to torch.distributed (also RCCL underneath on ROCm)
                import warnings
                warnings.warn(f"native RCCL comm init failed ({e}); "
                              "falling back to torch.distributed")
                self.comm = _TorchComm(process_group)
        else:
            self.comm = _TorchComm(process_group)
        self._works: list = []
        self._ready: dict[int, int] = {}
        self._launched = 0
        self._hooks = []
        if self.world_size > 1:
            self._broadcast_initial_state()
            for p in self.flat.params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._mark_ready))
        self._reset_bucket_state()

    # -- init-time sync ----------------------------------------------------
    def _broadcast_initial_state(self):
        self.comm.broadcast(self.flat.flat_param, 0)
        for buf in self.module.buffers():
            self.comm.broadcast(buf, 0)

    # -- per-iteration machinery -------------------------------------------
    def _reset_bucket_state(self):
        self._ready = {b.index: 0 for b in self.flat.buckets}
        self._launched = 0
        self._works = []

    def _mark_ready(self, param: torch.Tensor):
        b = self.flat.bucket_of[id(param)]
        self._ready[b.index] += 1
        # launch complete buckets in fixed index order (cross-rank safety)
        buckets = self.flat.buckets
        while (self._launched < len(buckets)
               and self._ready[self._launched]
               == len(buckets[self._launched].params)):
            w = self.comm.all_reduce_async(
                self.flat.grad_slice(buckets[self._launched]))
            if w is not None:
                self._works.append(w)
            self._launched += 1

    def forward(self, *args, **kwargs):
        if self.world_size > 1:
            self._reset_bucket_state()
        return self.module(*args, **kwargs)

    def finish_grad_sync(self):
        """Wait for all in-flight bucket all-reduces (call before the
        optimizer step). Gradients are left as SUMS over ranks; consume
        self.grad_scale in the optimizer (or scale explicitly)."""
        if self.world_size > 1:
            assert self._launched == len(self.flat.buckets), \
                "backward did not produce grads for every bucket"
            self.comm.finish(self._works)
        self._works = []

    # -- passthroughs: checkpoint key parity with torch DDP ("module." prefix,
    # /root/reference/cifar_example_ddp.py:118-119 saves the wrapper) --------
    def train(self, mode: bool = True):
        self.module.train(mode)
        return super().train(mode)
