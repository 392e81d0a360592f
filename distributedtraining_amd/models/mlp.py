"""Small fixture models for CPU-scale protocol tests.

Mirrors the reference's test-model strategy: every role has
an MNIST/MLP-scale twin so the full miner/validator/averager protocol runs
as a unit test. Reference: FeedforwardNN, training_manager.py:440-459
(5-layer 784→…→10 MLP) and SimpleCNN, new_training_manager.py:173-189.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .gpt2 import CausalLMOutput


class FeedforwardNN(nn.Module):
    """Classification MLP with the reference fixture's 5-layer shape."""

    def __init__(self, in_dim: int = 784, hidden: int = 64, classes: int = 10):
        super().__init__()
        self.fc1 = nn.Linear(in_dim, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, hidden)
        self.fc4 = nn.Linear(hidden, hidden)
        self.fc5 = nn.Linear(hidden, classes)

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None, **_) -> CausalLMOutput:
        x = input_ids
        h = F.relu(self.fc1(x.flatten(1)))
        h = F.relu(self.fc2(h))
        h = F.relu(self.fc3(h))
        h = F.relu(self.fc4(h))
        logits = self.fc5(h)
        loss = F.cross_entropy(logits, labels) if labels is not None else None
        return CausalLMOutput(loss=loss, logits=logits)
