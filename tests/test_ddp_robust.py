"""DDP robustness semantics (CPU/gloo, world 2): no_sync gradient
accumulation, unused-parameter handling, batched buffer broadcast —
the torch-DDP behaviors implied at /root/reference/cifar_example_ddp.py:83
(torch.nn.parallel.DistributedDataParallel tolerates frozen/conditional
params and offers no_sync())."""

import torch
from torch import nn

from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState
from tests.test_ddp_cpu import WORLD, run_distributed


class _Branchy(nn.Module):
    """fc_b only runs when use_b=True -> its params may get no grad."""

    def __init__(self):
        super().__init__()
        self.fc_a = nn.Linear(8, 10)
        self.fc_b = nn.Linear(8, 10)

    def forward(self, x, use_b=False):
        return self.fc_b(x) if use_b else self.fc_a(x)


def _no_sync_accumulation(rank):
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    torch.manual_seed(0)
    net = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 10))
    ddp = DistributedDataParallel(net, bucket_mb=0.001, first_bucket_mb=0.001)
    g = torch.Generator().manual_seed(5)
    xs = [torch.randn(4, 8, generator=g) for _ in range(3 * WORLD)]
    ys = [torch.randint(0, 10, (4,), generator=g) for _ in range(3 * WORLD)]
    # rank r takes micro-batches r, r+W, r+2W; two under no_sync, one synced
    mine = list(range(rank, 3 * WORLD, WORLD))
    with ddp.no_sync():
        for i in mine[:2]:
            cross_entropy(ddp(xs[i]), ys[i]).backward()
    cross_entropy(ddp(xs[mine[2]]), ys[mine[2]]).backward()
    ddp.finish_grad_sync()
    grad = ddp.flat.flat_grad * ddp.grad_scale

    # single-process reference: mean over ranks of per-rank accumulated sums
    torch.manual_seed(0)
    ref = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 10))
    ref_flat = FlatState(ref)
    for i in range(3 * WORLD):
        cross_entropy(ref(xs[i]), ys[i]).backward()
    return grad, ref_flat.flat_grad / WORLD


def test_no_sync_gradient_accumulation():
    results = run_distributed(_no_sync_accumulation, 29621)
    for rank, (grad, ref) in results.items():
        torch.testing.assert_close(grad, ref, rtol=1e-4, atol=1e-6)


def _unused_param_raises(rank):
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    torch.manual_seed(0)
    ddp = DistributedDataParallel(_Branchy(), bucket_mb=0.001,
                                  first_bucket_mb=0.001)
    x = torch.randn(4, 8)
    y = torch.randint(0, 10, (4,))
    cross_entropy(ddp(x, use_b=False), y).backward()
    try:
        ddp.finish_grad_sync()
        return "no error"
    except RuntimeError as e:
        return str(e)


def test_unused_param_default_raises_naming_params():
    results = run_distributed(_unused_param_raises, 29623)
    for msg in results.values():
        assert "fc_b" in msg, msg
        assert "find_unused_parameters" in msg, msg


def _unused_param_tolerated(rank):
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    torch.manual_seed(0)
    net = _Branchy()
    ddp = DistributedDataParallel(net, bucket_mb=0.001,
                                  first_bucket_mb=0.001,
                                  find_unused_parameters=True)
    g = torch.Generator().manual_seed(11)
    x_all = torch.randn(2 * WORLD, 8, generator=g)
    y_all = torch.randint(0, 10, (2 * WORLD,), generator=g)
    x = x_all[rank * 2:(rank + 1) * 2]
    y = y_all[rank * 2:(rank + 1) * 2]
    ddp.flat.zero_grad()
    cross_entropy(ddp(x, use_b=False), y).backward()
    ddp.finish_grad_sync()
    grad = ddp.flat.flat_grad * ddp.grad_scale

    torch.manual_seed(0)
    ref = _Branchy()
    ref_flat = FlatState(ref)
    cross_entropy(ref(x_all, use_b=False), y_all).backward()
    return grad, ref_flat.flat_grad.clone()


def test_unused_param_flush_matches_reference():
    results = run_distributed(_unused_param_tolerated, 29625)
    for rank, (grad, ref) in results.items():
        torch.testing.assert_close(grad, ref, rtol=1e-4, atol=1e-6)


def _frozen_layer_trains(rank):
    """A requires_grad=False layer (fine-tune pattern) trains under DDP."""
    from mi355x import optim
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    torch.manual_seed(0)
    net = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 10))
    for p in net[0].parameters():
        p.requires_grad = False
    ddp = DistributedDataParallel(net, bucket_mb=0.001)
    opt = optim.SGD(ddp.flat, lr=0.1, grad_scale=ddp.grad_scale)
    g = torch.Generator().manual_seed(3 + rank)
    for _ in range(2):
        x = torch.randn(4, 8, generator=g)
        y = torch.randint(0, 10, (4,), generator=g)
        opt.zero_grad()
        cross_entropy(ddp(x), y).backward()
        ddp.finish_grad_sync()
        opt.step()
    return (net[0].weight.clone(), ddp.flat.flat_param.clone())


def test_frozen_layer_under_ddp():
    results = run_distributed(_frozen_layer_trains, 29627)
    w0, p0 = results[0]
    w1, p1 = results[1]
    torch.testing.assert_close(p0, p1, rtol=0, atol=0)  # ranks in lockstep
    torch.testing.assert_close(w0, w1, rtol=0, atol=0)  # frozen stays equal


class _WithBuffers(nn.Module):
    def __init__(self, seed):
        super().__init__()
        self.fc = nn.Linear(4, 4)
        g = torch.Generator().manual_seed(seed)
        self.register_buffer("stat_f", torch.randn(16, generator=g))
        self.register_buffer("stat_g", torch.randn(8, generator=g))
        self.register_buffer("count", torch.randint(0, 100, (1,), generator=g))

    def forward(self, x):
        return self.fc(x)


def _buffer_broadcast(rank):
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    net = _WithBuffers(seed=1000 + rank)  # DIFFERENT per rank
    DistributedDataParallel(net, bucket_mb=0.001)
    return net.stat_f.clone(), net.stat_g.clone(), net.count.clone()


def test_batched_buffer_broadcast_syncs_all_dtypes():
    results = run_distributed(_buffer_broadcast, 29629)
    ref = _WithBuffers(seed=1000)  # what rank 0 had
    for f, g, c in results.values():
        torch.testing.assert_close(f, ref.stat_f, rtol=0, atol=0)
        torch.testing.assert_close(g, ref.stat_g, rtol=0, atol=0)
        assert torch.equal(c, ref.count)


def _dup_device_fallback(rank):
    """Feasibility agreement: unique device slots -> feasible (on CPU the
    extension-symbol check decides); a forced duplicate slot -> all ranks
    agree native RCCL is infeasible (the oversubscribed-rehearsal case:
    two ranks on one GPU would hang ncclCommInitRank)."""
    from mi355x.parallel import DistributedDataParallel, comm
    from mi355x.parallel import ddp as ddp_mod

    comm.init_process_group(backend="gloo")
    torch.manual_seed(0)
    net = nn.Linear(8, 10)
    ddp = DistributedDataParallel(net)
    unique = ddp._native_feasible_everywhere()

    orig = ddp_mod._device_slot
    ddp_mod._device_slot = lambda device: 7  # every rank reports one GPU
    try:
        dup = ddp._native_feasible_everywhere()
    finally:
        ddp_mod._device_slot = orig
    return unique, dup


def test_duplicate_device_blocks_native_comm():
    results = run_distributed(_dup_device_fallback)
    for rank, (unique, dup) in results.items():
        # CPU ranks always present distinct slots; with the extension
        # built its RCCL symbols exist, so the unique case is feasible
        assert unique is True, f"rank {rank}: unique slots deemed infeasible"
        assert dup is False, f"rank {rank}: duplicate slots not detected"
