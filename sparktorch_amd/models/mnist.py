"""MNIST models from the reference examples, used by the benchmark configs.

* ``MnistMLP``: 784-256-256-10 MLP (reference examples/simple_dnn.py:21-27) —
  the BASELINE.json flagship.
* ``MnistCNN``: Conv2d(1,16,5) -> Conv2d(16,32,3) -> maxpool -> dropout ->
  fc(3872,10) with the in-forward unflatten (reference examples/cnn_network.py:6-24).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


class MnistMLP(nn.Module):
    def __init__(self, in_dim: int = 784, hidden: int = 256, classes: int = 10):
        super().__init__()
        self.fc1 = nn.Linear(in_dim, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, classes)

    def forward(self, x):
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        return self.fc3(x)


class MnistCNN(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 16, kernel_size=5)
        self.conv2 = nn.Conv2d(16, 32, kernel_size=3)
        self.dropout = nn.Dropout2d(p=0.25)
        self.fc = nn.Linear(3872, 10)

    def forward(self, x):
        # rows arrive flattened (784,); unflatten like the reference does
        # (cnn_network.py:16).
        x = x.view(-1, 1, 28, 28)
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.max_pool2d(x, 2)
        x = self.dropout(x)
        x = torch.flatten(x, 1)
        return self.fc(x)
