"""784-800-500-10 MLP (reference: model_ops/fc_nn.py:12-44)."""

import torch.nn as nn
import torch.nn.functional as F


class FC_NN(nn.Module):
    def __init__(self, num_classes: int = 10, in_features: int = 784):
        super().__init__()
        self.fc1 = nn.Linear(in_features, 800)
        self.fc2 = nn.Linear(800, 500)
        self.fc3 = nn.Linear(500, num_classes)

    def forward(self, x):
        x = x.flatten(1)
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        return self.fc3(x)
