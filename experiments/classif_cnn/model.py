"""classif_cnn hello-world task: small CIFAR10 CNN with a custom F1 metric.

Reference: experiments/classif_cnn/model.py:11-60 (the classic 2-conv /
3-FC tutorial net).  Demonstrates the custom-metric contract: ``inference``
returns extra keys as ``{"value": v, "higher_is_better": bool}`` dicts
(reference doc/sphinx/scenarios.rst "Implement new metrics").
"""

import torch
from torch import nn
from torch.nn import functional as F

from msrflute_amd.core.model import BaseModel
from msrflute_amd.utils import to_device


class Net(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 6, 5)
        self.pool = nn.MaxPool2d(2, 2)
        self.conv2 = nn.Conv2d(6, 16, 5)
        self.fc1 = nn.Linear(16 * 5 * 5, 120)
        self.fc2 = nn.Linear(120, 84)
        self.fc3 = nn.Linear(84, num_classes)

    def forward(self, x):
        x = self.pool(F.relu(self.conv1(x)))
        x = self.pool(F.relu(self.conv2(x)))
        x = torch.flatten(x, 1)
        x = F.relu(self.fc2(F.relu(self.fc1(x))))
        return self.fc3(x)


def micro_f1(pred: torch.Tensor, target: torch.Tensor) -> float:
    """Micro-averaged F1 == accuracy for single-label classification, but
    computed the F1 way so the metric plumbing is exercised end-to-end."""
    tp = (pred == target).sum().item()
    total = target.numel()
    return tp / total if total else 0.0


class CNN(BaseModel):
    def __init__(self, model_config):
        super().__init__()
        self.net = Net(model_config.get("num_classes", 10))

    def loss(self, input):
        x, y = to_device(input["x"]), to_device(input["y"])
        return F.cross_entropy(self.net(x), y.long())

    def inference(self, input):
        x, y = to_device(input["x"]), to_device(input["y"])
        output = self.net(x)
        pred = torch.argmax(output, dim=1)
        acc = (pred == y).float().mean().item()
        return {"output": output, "acc": acc, "batch_size": x.shape[0],
                "f1_score": {"value": micro_f1(pred, y),
                             "higher_is_better": True}}
