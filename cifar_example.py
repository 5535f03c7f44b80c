"""Single-process CIFAR-10 training — the reference's serial script surface
(/root/reference/cifar_example.py) on the mi355x framework.

Same CLI (no args), hyperparameters (batch 4, SGD lr=0.001 momentum=0.9,
2 epochs), print cadence (running loss every 2000 minibatches), checkpoint
path/format ('./cifar_net.pth', plain state_dict keys), and final accuracy
print. Differences are conscious fixes of reference bugs:
no dead `dataiter.next()` (removed API), no dead imshow/matplotlib.

Env overrides (benchmark configs; CLI shape unchanged):
  MI355X_MODEL=net|resnet18|resnet50        (default net)
  MI355X_SYNTHETIC=1                        synthetic 32x32 data (no download)
  MI355X_BATCH / MI355X_EPOCHS / MI355X_STEPS / MI355X_LR
  MI355X_DEVICE=cpu|cuda                    (default cpu, like the reference)
  MI355X_CKPT=path                          checkpoint path (default the
                                            reference's ./cifar_net.pth)
"""

import os
import time

import torch

from mi355x import optim
from mi355x.data import CIFAR10, DataLoader, SyntheticImageDataset
from mi355x.models import build_model
from mi355x.ops import cross_entropy
from mi355x.parallel.flat import FlatState


def get_datasets():
    if os.environ.get("MI355X_SYNTHETIC", "0") == "1":
        return (SyntheticImageDataset(50000, seed=1),
                SyntheticImageDataset(10000, seed=2))
    # download=True bootstraps a clean box, like the reference's
    # torchvision.datasets.CIFAR10(root='./data', download=True)
    # (/root/reference/cifar_example.py:40-44)
    return (CIFAR10("./data", train=True, download=True),
            CIFAR10("./data", train=False, download=True))


def main():
    device = torch.device(os.environ.get("MI355X_DEVICE", "cpu"))
    batch = int(os.environ.get("MI355X_BATCH", "4"))
    epochs = int(os.environ.get("MI355X_EPOCHS", "2"))
    lr = float(os.environ.get("MI355X_LR", "0.001"))
    max_steps = int(os.environ.get("MI355X_STEPS", "0"))  # 0 = full epochs

    trainset, testset = get_datasets()
    # num_workers=2 mirrors the reference surface (cifar_example.py:47,52);
    # our loader needs no worker processes (see mi355x/data.py)
    trainloader = DataLoader(trainset, batch_size=batch, shuffle=True,
                             num_workers=2,
                             device=device if device.type == "cuda" else None)
    testloader = DataLoader(testset, batch_size=batch, shuffle=False,
                            num_workers=2,
                            device=device if device.type == "cuda" else None)

    net = build_model(os.environ.get("MI355X_MODEL", "net")).to(device)
    flat = FlatState(net)
    optimizer = optim.SGD(flat, lr=lr, momentum=0.9)

    meter = None
    if os.environ.get("MI355X_LOG_SPEED", "0") == "1":
        from mi355x.utils import SpeedMeter
        meter = SpeedMeter(print_every=100)

    t0 = time.time()
    steps = 0
    for epoch in range(epochs):
        running_loss = 0.0
        for i, (inputs, labels) in enumerate(trainloader):
            inputs, labels = inputs.to(device), labels.to(device)
            optimizer.zero_grad()
            outputs = net(inputs)
            loss = cross_entropy(outputs, labels)
            loss.backward()
            optimizer.step()

            if meter is not None:
                meter.step(inputs.shape[0])
            running_loss += loss.item()
            if i % 2000 == 1999:
                print("[%d, %5d] loss: %.3f" %
                      (epoch + 1, i + 1, running_loss / 2000))
                running_loss = 0.0
            steps += 1
            if max_steps and steps >= max_steps:
                break
        if max_steps and steps >= max_steps:
            break
    print(f"Finished Training ({steps} steps, {time.time() - t0:.1f}s)")

    PATH = os.environ.get("MI355X_CKPT", "./cifar_net.pth")
    torch.save(net.state_dict(), PATH)

    correct = 0
    total = 0
    net.eval()
    with torch.no_grad():
        for images, labels in testloader:
            images, labels = images.to(device), labels.to(device)
            outputs = net(images)
            _, predicted = torch.max(outputs, 1)
            total += labels.size(0)
            correct += (predicted == labels).sum().item()
            if max_steps and total >= max_steps * batch:
                break

    print("Accuracy of the network on the 10000 test images: %d %%" %
          (100 * correct / max(total, 1)))


if __name__ == "__main__":
    main()
