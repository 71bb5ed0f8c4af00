"""LeNet for MNIST (parity with /root/reference/src/model_ops/lenet.py:20-41).

The reference's LeNetSplit (manual per-op backward to interleave MPI sends,
lenet.py:43-341) is intentionally NOT reproduced: the framework gets the same
comm/compute overlap from flat gradient views + stream-ordered collectives
(draco_amd/parallel/flat.py), so one plain autograd module serves both roles.
"""
import torch.nn as nn
import torch.nn.functional as F


class LeNet(nn.Module):
    def __init__(self, num_classes: int = 10, in_channels: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_channels, 20, 5, 1)
        self.conv2 = nn.Conv2d(20, 50, 5, 1)
        self.fc1 = nn.Linear(4 * 4 * 50, 500)
        self.fc2 = nn.Linear(500, num_classes)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.conv1(x)), 2, 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2, 2)
        x = x.flatten(1)
        x = F.relu(self.fc1(x))
        return self.fc2(x)
