"""Test/demo networks.  Parity with reference sparktorch/tests/simple_net.py:5-65
(same layer sizes so the reference's test matrix carries over verbatim)."""

import torch
import torch.nn as nn
import torch.nn.functional as F


class Net(nn.Module):
    """10 -> 20 -> 1 regression MLP (reference simple_net.py:5-16)."""

    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(10, 20)
        self.fc2 = nn.Linear(20, 1)

    def forward(self, x):
        x = F.relu(self.fc1(x))
        return self.fc2(x)


class AutoEncoder(nn.Module):
    """10 -> 5 -> 2 -> 5 -> 10 (reference simple_net.py:19-36)."""

    def __init__(self):
        super().__init__()
        self.enc1 = nn.Linear(10, 5)
        self.enc2 = nn.Linear(5, 2)
        self.dec1 = nn.Linear(2, 5)
        self.dec2 = nn.Linear(5, 10)

    def forward(self, x):
        x = F.relu(self.enc1(x))
        x = self.enc2(x)
        x = F.relu(self.dec1(x))
        return self.dec2(x)


class ClassificationNet(nn.Module):
    """10 -> 20 -> 2 with log_softmax (reference simple_net.py:39-51)."""

    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(10, 20)
        self.fc2 = nn.Linear(20, 2)

    def forward(self, x):
        x = F.relu(self.fc1(x))
        return F.log_softmax(self.fc2(x), dim=1)


class NetworkWithParameters(nn.Module):
    """Constructor-kwarg net for the lazy-serialization path
    (reference simple_net.py:54-65)."""

    def __init__(self, input_dim=10, hidden_dim=20, output_dim=1):
        super().__init__()
        self.fc1 = nn.Linear(input_dim, hidden_dim)
        self.fc2 = nn.Linear(hidden_dim, output_dim)

    def forward(self, x):
        x = F.relu(self.fc1(x))
        return self.fc2(x)
