"""The reference `Net` — LeNet-style CNN for 10-class 32x32 input.

Architecture and state_dict keys/shapes exactly match the reference
(/root/reference/cifar_example.py:17-34): conv1 Conv2d(3,6,5) -> ReLU ->
MaxPool(2,2) -> conv2 Conv2d(6,16,5) -> ReLU -> MaxPool -> flatten (NCHW
order, so fc weights interchange byte-for-byte) -> fc1(400,120) -> ReLU ->
fc2(120,84) -> ReLU -> fc3(84,10). 62,006 parameters.
"""

import torch
from torch import nn

from .layers import Conv2d, Linear, MaxPool2d, to_model_layout


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = Conv2d(3, 6, 5, act="relu")
        self.conv2 = Conv2d(6, 16, 5, act="relu")
        self.pool = MaxPool2d(2, 2)
        self.fc1 = Linear(16 * 5 * 5, 120, act="relu")
        self.fc2 = Linear(120, 84, act="relu")
        self.fc3 = Linear(84, 10)

    def forward(self, x):
        x = to_model_layout(x)           # NCHW -> NHWC (16-bit on GPU)
        x = self.pool(self.conv1(x))
        x = self.pool(self.conv2(x))
        # flatten in NCHW element order for fc1-weight parity with the
        # reference (x.view(-1, 16*5*5) on an NCHW tensor,
        # /root/reference/cifar_example.py:30)
        x = x.permute(0, 3, 1, 2).reshape(x.shape[0], -1)
        if not x.is_contiguous():
            x = x.contiguous()
        x = self.fc1(x)
        x = self.fc2(x)
        return self.fc3(x)
