"""Native RCCL communicator smoke (world_size 1 on the single-GPU box;
the multi-rank path is exercised by the driver's 8-GPU scaling run and
shares all code but the rendezvous fan-in)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_native_comm_world1():
    from mi355x.parallel.rccl import NativeComm

    c = NativeComm(rank=0, world=1)
    t = torch.arange(8, dtype=torch.float32, device="cuda")
    c.all_reduce(t)  # world 1: identity
    c.wait()
    torch.testing.assert_close(t.cpu(), torch.arange(8, dtype=torch.float32))
    b = torch.full((4,), 3.0, device="cuda")
    c.broadcast(b, 0)
    c.wait()
    torch.testing.assert_close(b.cpu(), torch.full((4,), 3.0))
    c.barrier()
    out = torch.empty(8, device="cuda")
    c.all_gather(out, t)
    c.wait()
    torch.testing.assert_close(out.cpu(), t.cpu())


def test_native_comm_bf16():
    from mi355x.parallel.rccl import native_comm

    c = native_comm()
    t = torch.randn(1000, device="cuda").to(torch.bfloat16)
    ref = t.clone()
    c.all_reduce(t)
    c.wait()
    torch.cuda.synchronize()
    torch.testing.assert_close(t, ref)
