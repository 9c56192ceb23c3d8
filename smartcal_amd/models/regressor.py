"""MLP regressor: metadata → direction-selection probabilities.

Parity with `demixing_rl/regressor_net.py:6-28`: n_input → 32 → 32 →
n_output with ELU hidden activations and tanh output.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["RegressorNet"]


class RegressorNet(nn.Module):
    def __init__(self, n_input: int, n_output: int, n_hidden: int = 32):
        super().__init__()
        self.fc1 = nn.Linear(n_input, n_hidden)
        self.fc2 = nn.Linear(n_hidden, n_hidden)
        self.fc3 = nn.Linear(n_hidden, n_output)

    def forward(self, x):
        x = F.elu(self.fc1(x))
        x = F.elu(self.fc2(x))
        return torch.tanh(self.fc3(x))
