"""Small conv-net fixture for the classifier protocol tests.

The reference uses a SimpleCNN on MNIST for its experimental loop
(/root/reference/hivetrain/new_training_manager.py:173-189) and a 5-layer
FeedforwardNN for the MNIST delta/gradient loops
(training_manager.py:440-459) — test-scale models exercising the full
miner/validator/averager protocol without an LLM. This is the CNN twin
(models/mlp.py is the feedforward twin); conv/pool run through PyTorch's
MIOpen bindings on ROCm — fixtures aren't a hot path.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .gpt2 import CausalLMOutput


class SimpleCNN(nn.Module):
    """Conv(1->8) -> pool -> Conv(8->16) -> pool -> FC, for [B,1,28,28]
    inputs (or flattened [B, 784]). Same call contract as the LM models
    (labels= -> .loss) so the validator/averager roles drive it unchanged
    — the reference's MNIST validator/averager twins
    (validation_logic.py:265-318, averaging_logic.py:586-760)."""

    def __init__(self, n_classes: int = 10):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 8, 3, padding=1)
        self.conv2 = nn.Conv2d(8, 16, 3, padding=1)
        self.fc1 = nn.Linear(16 * 7 * 7, 64)
        self.fc2 = nn.Linear(64, n_classes)

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None, **_) -> CausalLMOutput:
        x = input_ids
        if x.dim() == 2:                       # flattened batches
            x = x.view(-1, 1, 28, 28)
        x = F.max_pool2d(F.relu(self.conv1(x)), 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = x.flatten(1)
        logits = self.fc2(F.relu(self.fc1(x)))
        loss = F.cross_entropy(logits, labels) if labels is not None else None
        return CausalLMOutput(loss=loss, logits=logits)
