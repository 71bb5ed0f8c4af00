"""784-800-500-10 MLP (parity with /root/reference/src/model_ops/fc_nn.py:21-40)."""
import torch.nn as nn


class FC_NN(nn.Module):
    def __init__(self, num_classes: int = 10, in_features: int = 784):
        super().__init__()
        self.net = nn.Sequential(
            nn.Flatten(),
            nn.Linear(in_features, 800),
            nn.ReLU(),
            nn.Linear(800, 500),
            nn.ReLU(),
            nn.Linear(500, num_classes),
        )

    def forward(self, x):
        return self.net(x)
