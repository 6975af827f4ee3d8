"""fit_a_line — the reference's smallest example job
(example/fit_a_line/train.py): linear regression on 13 features.
Used as BASELINE config 1 (CPU/gloo world_size=2 plumbing check)."""
import torch.nn as nn


class FitALine(nn.Module):
    def __init__(self, in_features=13):
        super().__init__()
        self.fc = nn.Linear(in_features, 1)

    def forward(self, x):
        return self.fc(x)
