"""Multi-process CPU tests of the DDP engine over gloo (world_size 2).

Golden invariant: W-rank DDP gradients equal the
single-process gradients on the concatenated batch; parameters stay
bit-identical across ranks after steps.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

from mi355x.models import Net
from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState

WORLD = 2


def _run_worker(rank, fn, store_path, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(WORLD), "MASTER_ADDR": "127.0.0.1",
        # file-store rendezvous (comm.init_process_group): no TCP port to
        # race over — probing a free port and rebinding it flaked ~1/10
        # suite runs when the OS re-issued the port in between
        "MI355X_RDZV_FILE": store_path,
    })
    try:
        res = fn(rank)
        # serialize to BYTES: putting live tensors on a SimpleQueue shares
        # them by fd-passing over an AF_UNIX socket, and a worker that
        # exits before the parent has received the fds resets the socket
        # (observed as a ~1-in-10 ConnectionResetError in parent q.get())
        import io
        buf = io.BytesIO()
        torch.save(res, buf)
        q.put((rank, "ok", buf.getvalue()))
    except Exception as e:  # surface tracebacks
        import traceback
        q.put((rank, "err", traceback.format_exc() + str(e)))
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, port=None):
    # a fresh FileStore path per call (the `port` arg is kept for caller
    # compatibility but unused — TCP rendezvous raced on ephemeral ports)
    import tempfile

    d = tempfile.mkdtemp(prefix="mi355x_rdzv_")
    store_path = os.path.join(d, "store")
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_run_worker, args=(r, fn, store_path, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, status, res = q.get()
        assert status == "ok", f"rank {rank} failed:\n{res}"
        import io
        results[rank] = torch.load(io.BytesIO(res), weights_only=False)
    for p in procs:
        p.join(timeout=60)
    import shutil

    shutil.rmtree(d, ignore_errors=True)
    return results


def _golden_grads(rank):
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    torch.manual_seed(0)  # same init on every rank (broadcast also enforces)
    net = Net()
    ddp = DistributedDataParallel(net, bucket_mb=0.05, first_bucket_mb=0.02)
    g = torch.Generator().manual_seed(42)
    x_all = torch.randn(8, 3, 32, 32, generator=g)
    y_all = torch.randint(0, 10, (8,), generator=g)
    # each rank takes its shard
    x = x_all[rank * 4:(rank + 1) * 4]
    y = y_all[rank * 4:(rank + 1) * 4]
    loss = cross_entropy(ddp(x), y)
    loss.backward()
    ddp.finish_grad_sync()
    grad = ddp.flat.flat_grad * ddp.grad_scale

    # single-process reference on the concatenated batch
    torch.manual_seed(0)
    ref = Net()
    ref_flat = FlatState(ref)
    cross_entropy(ref(x_all), y_all).backward()
    return grad, ref_flat.flat_grad.clone()


def test_ddp_grads_match_single_process():
    results = run_distributed(_golden_grads, 29611)
    for rank, (grad, ref_grad) in results.items():
        torch.testing.assert_close(grad, ref_grad, rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(results[0][0], results[1][0])  # ranks agree


def _train_steps(rank):
    from mi355x import optim
    from mi355x.parallel import DistributedDataParallel, comm

    comm.init_process_group(backend="gloo")
    torch.manual_seed(100 + rank)  # DIFFERENT init; broadcast must fix it
    net = Net()
    ddp = DistributedDataParallel(net, bucket_mb=0.1)
    opt = optim.SGD(ddp.flat, lr=0.05, momentum=0.9,
                    grad_scale=ddp.grad_scale)
    g = torch.Generator().manual_seed(7 + rank)
    losses = []
    for _ in range(4):
        x = torch.randn(4, 3, 32, 32, generator=g)
        y = torch.randint(0, 10, (4,), generator=g)
        opt.zero_grad()
        loss = cross_entropy(ddp(x), y)
        loss.backward()
        ddp.finish_grad_sync()
        opt.step()
        losses.append(loss.item())
    return ddp.flat.flat_param.clone(), losses


def test_ddp_params_stay_identical_across_ranks():
    results = run_distributed(_train_steps, 29613)
    p0, _ = results[0]
    p1, _ = results[1]
    torch.testing.assert_close(p0, p1, rtol=0, atol=0)


def _metric_sync(rank):
    from mi355x.metrics import DistAccuracy
    from mi355x.parallel import comm

    comm.init_process_group(backend="gloo")
    acc = DistAccuracy(dist_sync_on_step=True)
    preds = torch.tensor([0, 1, 2, 3]) if rank == 0 else torch.tensor([0, 0, 0, 0])
    target = torch.tensor([0, 1, 0, 0])  # rank0: 2 correct; rank1: 3 correct
    acc.update(preds, target)
    return acc.compute()


def test_dist_accuracy_sync():
    results = run_distributed(_metric_sync, 29615)
    for r in results.values():
        assert r == pytest.approx(5 / 8)


def _sync_bn_stats(rank):
    from mi355x.models.layers import BatchNorm2d
    from mi355x.parallel import comm
    import torch.distributed as dist

    comm.init_process_group(backend="gloo")
    bn = BatchNorm2d(4)
    bn.process_group = dist.group.WORLD
    bn.train()
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(2, 3, 3, 4, generator=g)
    # CPU path of ops.batch_norm ignores process_group (plain-torch path) —
    # validate the running-stat math by calling the GPU-path formula by hand
    # on CPU via the stats all-reduce contract:
    s = torch.stack([x.sum(dim=(0, 1, 2)), (x * x).sum(dim=(0, 1, 2))])
    dist.all_reduce(s)
    m_total = 2 * 3 * 3 * WORLD
    mean = s[0] / m_total
    var = s[1] / m_total - mean * mean
    return mean, var, x


def test_sync_bn_global_stats():
    results = run_distributed(_sync_bn_stats, 29617)
    m0, v0, x0 = results[0]
    m1, v1, x1 = results[1]
    torch.testing.assert_close(m0, m1)
    torch.testing.assert_close(v0, v1)
    xall = torch.cat([x0.reshape(-1, 4), x1.reshape(-1, 4)])
    torch.testing.assert_close(m0, xall.mean(0), rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(v0, xall.var(0, unbiased=False), rtol=1e-4,
                               atol=1e-5)
