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

import contextlib
import os

import torch
import torch.distributed as dist
from torch import nn

from .flat import FlatState


def _device_slot(device: torch.device) -> int:
    """Globally-unique device slot for duplicate-device detection:
    (host, ordinal) for GPUs; CPU ranks get -(rank+1), which never
    collides. Two ranks reporting the same GPU slot (oversubscribed
    rehearsal layouts — more ranks than GPUs) cannot build a native RCCL
    communicator: ncclCommInitRank refuses duplicate devices, and the
    rank-0 uid server would be left hanging."""
    import socket
    import zlib

    if device.type == "cuda":
        host = zlib.crc32(socket.gethostname().encode()) & 0x3FFFFF
        ordinal = (device.index if device.index is not None
                   else torch.cuda.current_device())
        return host * 512 + ordinal
    return -(dist.get_rank() + 1)


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
                 process_group=None, comm: str | None = None,
                 find_unused_parameters: bool | None = None):
        super().__init__()
        if find_unused_parameters is None:
            find_unused_parameters = (
                os.environ.get("MI355X_FIND_UNUSED", "0") == "1")
        self.find_unused_parameters = find_unused_parameters
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
        if use_native and not self._native_feasible_everywhere():
            # the fallback decision must be COLLECTIVE: one rank silently
            # dropping to torch.distributed while the others construct the
            # native communicator would deadlock in ncclCommInitRank
            import warnings
            warnings.warn("native RCCL comm unavailable on at least one "
                          "rank; all ranks falling back to torch.distributed")
            use_native = False
        if use_native:
            # past the agreement point a construction failure raises loudly
            # (a crash torchrun/our launcher tears down beats a silent
            # divergent fallback that hangs the job)
            self.comm = _NativeComm()
        else:
            self.comm = _TorchComm(process_group)
        self._works: list = []
        self._ready: dict[int, int] = {}
        self._ready_ids: set[int] = set()
        self._launched = 0
        self._hooks = []
        self._sync_enabled = True
        self._names = {id(p): n for n, p in module.named_parameters()}
        if self.world_size > 1:
            self._broadcast_initial_state()
            for p in self.flat.params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._mark_ready))
        self._reset_bucket_state()

    def _native_feasible_everywhere(self) -> bool:
        """Collective agreement on native-comm feasibility over the
        already-initialized torch.distributed group, so every rank takes
        the same comm path. Feasible iff on EVERY rank the extension is
        importable with the RCCL symbols present, AND no two ranks sit on
        the same physical GPU (all-gathered _device_slot values must be
        unique — RCCL cannot build a communicator for an oversubscribed
        rehearsal layout; those fall back to torch.distributed/gloo)."""
        ok = 1
        try:
            from mi355x.ops import ext
            e = ext()
            ok = int(hasattr(e, "RcclComm") and hasattr(e, "rccl_get_unique_id"))
        except Exception:
            ok = 0
        backend = dist.get_backend(self.process_group)
        dev = (self.flat.flat_param.device if backend == "nccl" else
               torch.device("cpu"))
        me = torch.tensor([ok, _device_slot(self.flat.flat_param.device)],
                          dtype=torch.int64, device=dev)
        world = dist.get_world_size(self.process_group)
        gathered = [torch.zeros_like(me) for _ in range(world)]
        dist.all_gather(gathered, me, group=self.process_group)
        ok_all = all(int(g[0]) for g in gathered)
        slots = [int(g[1]) for g in gathered]
        return ok_all and len(set(slots)) == len(slots)

    # -- init-time sync ----------------------------------------------------
    def _broadcast_initial_state(self):
        self.comm.broadcast(self.flat.flat_param, 0)
        # buffers (BN running stats): batched into one flat broadcast per
        # (device, dtype) group instead of one tiny collective per buffer
        # (ResNet-50 has 106 of them)
        groups: dict[tuple, list[torch.Tensor]] = {}
        for buf in self.module.buffers():
            if buf.numel() == 0:
                continue
            groups.setdefault((buf.device, buf.dtype), []).append(buf)
        for bufs in groups.values():
            if len(bufs) == 1:
                self.comm.broadcast(bufs[0], 0)
                continue
            flatb = torch.cat([b.detach().reshape(-1) for b in bufs])
            self.comm.broadcast(flatb, 0)
            off = 0
            for b in bufs:
                n = b.numel()
                b.detach().copy_(flatb[off:off + n].view_as(b))
                off += n

    # -- per-iteration machinery -------------------------------------------
    def _reset_bucket_state(self):
        self._ready = {b.index: 0 for b in self.flat.buckets}
        self._ready_ids = set()
        self._launched = 0
        self._works = []

    def _mark_ready(self, param: torch.Tensor):
        if not self._sync_enabled:
            return
        self._ready_ids.add(id(param))
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
        if self.world_size > 1 and self._sync_enabled:
            self._reset_bucket_state()
        return self.module(*args, **kwargs)

    @contextlib.contextmanager
    def no_sync(self):
        """Skip gradient all-reduce inside this context (gradient
        accumulation, torch-DDP semantics): grads accumulate locally into
        the flat buffer; the first backward outside the context reduces
        the accumulated sums."""
        old = self._sync_enabled
        self._sync_enabled = False
        try:
            yield
        finally:
            self._sync_enabled = old

    def finish_grad_sync(self):
        """Wait for all in-flight bucket all-reduces (call before the
        optimizer step). Gradients are left as SUMS over ranks; consume
        self.grad_scale in the optimizer (or scale explicitly)."""
        if self.world_size > 1 and self._sync_enabled:
            if self._launched < len(self.flat.buckets):
                if not self.find_unused_parameters:
                    missing = [self._names.get(id(p), "<unnamed>")
                               for b in self.flat.buckets
                               for p in b.params
                               if id(p) not in self._ready_ids]
                    raise RuntimeError(
                        "backward produced no gradient for parameter(s) "
                        f"{missing}; pass find_unused_parameters=True (or "
                        "MI355X_FIND_UNUSED=1) if the model has frozen or "
                        "conditionally-used parameters")
                # flush: launch the remaining buckets in the same fixed
                # index order every rank uses. Params that got no grad
                # contribute their current flat-grad content (zeros after
                # zero_grad) — summing zeros matches torch DDP's
                # find_unused_parameters behavior.
                for b in self.flat.buckets[self._launched:]:
                    w = self.comm.all_reduce_async(self.flat.grad_slice(b))
                    if w is not None:
                        self._works.append(w)
                self._launched = len(self.flat.buckets)
            self.comm.finish(self._works)
        self._works = []

    # -- passthroughs: checkpoint key parity with torch DDP ("module." prefix,
    # /root/reference/cifar_example_ddp.py:118-119 saves the wrapper) --------
    def train(self, mode: bool = True):
        self.module.train(mode)
        return super().train(mode)
