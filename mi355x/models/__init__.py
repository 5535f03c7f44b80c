from .net import Net  # noqa: F401
from .resnet import ResNet, resnet18, resnet50  # noqa: F401


def build_model(name: str, num_classes: int = 10):
    name = name.lower()
    if name == "net":
        return Net()
    if name == "resnet18":
        return resnet18(num_classes=num_classes, stem="cifar")
    if name == "resnet18_imagenet":
        return resnet18(num_classes=num_classes, stem="imagenet")
    if name == "resnet50":
        return resnet50(num_classes=num_classes, stem="imagenet")
    if name == "resnet50_cifar":
        return resnet50(num_classes=num_classes, stem="cifar")
    raise ValueError(f"unknown model {name!r}")
