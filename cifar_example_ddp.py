"""Distributed CIFAR-10 training — the reference's DDP script surface
(/root/reference/cifar_example_ddp.py) on the mi355x framework.

Same launcher env contract (RANK/LOCAL_RANK/WORLD_SIZE read from the
environment, MASTER_ADDR/PORT pinned to 127.0.0.1:29500 — single node,
like the reference's init_distributed at cifar_example_ddp.py:42-58), the
same --world_size CLI flag (env wins, as in the reference), per-rank batch
4, DistributedSampler with per-epoch set_epoch, our DDP wrap (flat-bucket
RCCL reducer), 'module.'-prefixed checkpoint keys. Conscious fixes of
reference bugs: eval tensors are moved to the device
(quirk 6), the checkpoint write is rank-0-gated (quirk 7), and the dead
`dataiter.next()` call is dropped (quirk 1), and a launcher-less run
works as world-1 instead of crashing in DistributedSampler (quirk 2 —
comm.init_process_group builds a 1-rank group).

Launch:  python -m mi355x.launcher --nproc-per-node 8 cifar_example_ddp.py
   (or)  torchrun --nproc-per-node 8 cifar_example_ddp.py

Env overrides (benchmark configs; CLI shape unchanged): MI355X_MODEL,
MI355X_SYNTHETIC, MI355X_BATCH, MI355X_EPOCHS, MI355X_STEPS, MI355X_LR,
MI355X_SYNC_BN=1, MI355X_FP16=1 (fp16 compute + dynamic loss scaling),
MI355X_CKPT (checkpoint path, default ./cifar_net.pth).
"""

import argparse
import os
import time

import torch

from mi355x import optim
from mi355x.data import CIFAR10, DataLoader, DistributedSampler, SyntheticImageDataset
from mi355x.metrics import DistAccuracy
from mi355x.models import build_model
from mi355x.ops import cross_entropy
from mi355x.parallel import DistributedDataParallel, comm, sync_bn


def init_distributed(args):
    # env contract, reference cifar_example_ddp.py:43-45; single-node
    # MASTER hard-wire, :55-56
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29500")
    args.rank, args.world_size, args.gpu = comm.init_process_group()
    args.distributed = args.world_size > 1
    comm.barrier()


def get_datasets(args):
    if os.environ.get("MI355X_SYNTHETIC", "0") == "1":
        return (SyntheticImageDataset(50000, seed=1),
                SyntheticImageDataset(10000, seed=2))
    # rank-0-gated download + barrier — a conscious fix of reference quirk
    # 11, where every rank races on the download because dataset setup
    # precedes init_distributed (/root/reference/cifar_example_ddp.py:67-69)
    if args.rank == 0:
        from mi355x.data import download_cifar10
        download_cifar10("./data")
    comm.barrier()
    return CIFAR10("./data", train=True), CIFAR10("./data", train=False)


def main(args):
    init_distributed(args)
    trainset, testset = get_datasets(args)
    use_cuda = torch.cuda.is_available()
    device = (torch.device("cuda", args.gpu % torch.cuda.device_count())
              if use_cuda else torch.device("cpu"))

    batch = int(os.environ.get("MI355X_BATCH", "4"))
    epochs = int(os.environ.get("MI355X_EPOCHS", "2"))
    lr = float(os.environ.get("MI355X_LR", "0.001"))
    max_steps = int(os.environ.get("MI355X_STEPS", "0"))

    sampler_train = DistributedSampler(trainset, num_replicas=args.world_size,
                                       rank=args.rank, shuffle=True)
    sampler_test = DistributedSampler(testset, num_replicas=args.world_size,
                                      rank=args.rank, shuffle=False)
    dev_for_loader = device if use_cuda else None
    trainloader = DataLoader(trainset, batch_size=batch, sampler=sampler_train,
                             device=dev_for_loader)
    testloader = DataLoader(testset, batch_size=batch, sampler=sampler_test,
                            device=dev_for_loader)

    net = build_model(os.environ.get("MI355X_MODEL", "net")).to(device)
    if os.environ.get("MI355X_SYNC_BN", "0") == "1":
        sync_bn.enable(net)
    net = DistributedDataParallel(net)
    # BASELINE config 5: fp16 compute + dynamic loss scaling (overflow
    # skips the step and halves the scale; clean streaks grow it)
    scaler = None
    if os.environ.get("MI355X_FP16", "0") == "1":
        from mi355x import amp
        amp.set_compute_dtype(torch.float16)
        scaler = amp.GradScaler()
    optimizer = optim.SGD(net.flat, lr=lr, momentum=0.9,
                          grad_scale=net.grad_scale)

    meter = None
    if os.environ.get("MI355X_LOG_SPEED", "0") == "1":
        from mi355x.utils import SpeedMeter
        meter = SpeedMeter(print_every=100)

    t0 = time.time()
    steps = 0
    for epoch in range(epochs):
        sampler_train.set_epoch(epoch)
        running_loss = 0.0
        for i, (inputs, labels) in enumerate(trainloader):
            inputs, labels = inputs.to(device), labels.to(device)
            optimizer.zero_grad()
            outputs = net(inputs)
            loss = cross_entropy(outputs, labels)
            if scaler is not None:
                used = scaler.scale_value
                (loss * used).backward()
                net.finish_grad_sync()
                if scaler.step_ok(net.flat.flat_grad):
                    optimizer.grad_scale = net.grad_scale / used
                    optimizer.step()
            else:
                loss.backward()
                net.finish_grad_sync()
                optimizer.step()

            if meter is not None:
                meter.step(inputs.shape[0])
            running_loss += loss.item()
            if i % 2000 == 1999 and args.rank == 0:
                print("[%d, %5d] loss: %.3f" %
                      (epoch + 1, i + 1, running_loss / 2000))
                running_loss = 0.0
            steps += 1
            if max_steps and steps >= max_steps:
                break
        if max_steps and steps >= max_steps:
            break
    if args.rank == 0:
        print(f"Finished Training ({steps} steps, {time.time() - t0:.1f}s)")

    PATH = os.environ.get("MI355X_CKPT", "./cifar_net.pth")
    if args.rank == 0:  # rank-gated (fixes reference quirk 7); keys keep
        torch.save(net.state_dict(), PATH)  # the 'module.' prefix

    accuracy = DistAccuracy(dist_sync_on_step=True, device=device)
    net.eval()
    with torch.no_grad():
        for n, (images, labels) in enumerate(testloader):
            images, labels = images.to(device), labels.to(device)
            outputs = net(images)
            _, predicted = torch.max(outputs, 1)
            accuracy.update(predicted, labels)
            if max_steps and n >= max_steps:
                break
    acc = accuracy.compute()
    if args.rank == 0:
        # reference print format (/root/reference/cifar_example_ddp.py:135)
        print("Accuracy of the network on the 10000 test images: %d %%" %
              (100 * acc))
    comm.destroy()


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--world_size", default=None,
                        help="unused in distributed mode (env wins, as in "
                             "the reference)")
    main(parser.parse_args())
