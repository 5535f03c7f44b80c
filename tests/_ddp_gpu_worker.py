"""Worker for test_gpu_train.py::test_ddp_world2_single_gpu_fallback:
two ranks SHARING one GPU over gloo. The DDP feasibility agreement must
detect the duplicate device (native RCCL would hang ncclCommInitRank),
fall back to torch.distributed collectives on every rank, and still
train. Launched under torch.distributed.run --nproc-per-node 2."""

import os
import sys

# launched by script path, so sys.path[0] is tests/ — make the repo root
# (the mi355x package location) importable first
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from mi355x.models import build_model
from mi355x.ops import cross_entropy
from mi355x.parallel import comm
from mi355x.parallel.ddp import DistributedDataParallel, _TorchComm


def main():
    rank, world, local = comm.init_process_group(backend="gloo")
    dev = torch.device("cuda", local % torch.cuda.device_count())
    torch.manual_seed(0)
    net = build_model("net").to(dev)
    ddp = DistributedDataParallel(net)
    native = not isinstance(ddp.comm, _TorchComm)
    g = torch.Generator().manual_seed(rank)
    loss = None
    for _ in range(2):
        x = torch.randn(8, 3, 32, 32, generator=g).to(dev)
        y = torch.randint(0, 10, (8,), generator=g).to(dev)
        loss = cross_entropy(ddp(x), y)
        loss.backward()
        ddp.finish_grad_sync()
        ddp.flat.flat_grad.zero_()
    assert torch.isfinite(loss.detach()).all(), "non-finite loss"
    if rank == 0:
        print(f"DDP_GPU_OK native={native}", flush=True)


if __name__ == "__main__":
    main()
