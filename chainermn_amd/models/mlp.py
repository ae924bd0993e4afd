"""3-layer MLP — the MNIST/plumbing model (BASELINE.json config 1).

Mirrors the reference example's architecture
(examples/mnist/train_mnist.py: 784 → n_units → n_units → 10).
"""

import torch.nn as nn
import torch.nn.functional as F


class MLP(nn.Module):

    def __init__(self, n_in=784, n_units=1000, n_out=10):
        super().__init__()
        self.l1 = nn.Linear(n_in, n_units)
        self.l2 = nn.Linear(n_units, n_units)
        self.l3 = nn.Linear(n_units, n_out)

    def forward(self, x):
        x = x.reshape(x.shape[0], -1)
        h = F.relu(self.l1(x))
        h = F.relu(self.l2(h))
        return self.l3(h)
