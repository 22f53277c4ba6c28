"""Shared classification-model wrapper.

Implements the BaseModel loss/inference contract once (CE loss; accuracy
metric; reference per-task copies: experiments/*/model.py) so task plugins
only provide the network.
"""

from __future__ import annotations

import torch
from torch import nn

from ..core.model import BaseModel
from ..utils import to_device


class ClassificationModel(BaseModel):
    """CE-loss classifier over ``{'x': features, 'y': labels}`` batches."""

    def __init__(self, net: nn.Module = None, **kwargs):
        super().__init__()
        if net is not None:
            self.net = net
        self.criterion = nn.CrossEntropyLoss()

    def loss(self, input) -> torch.Tensor:
        features = to_device(input["x"])
        labels = to_device(input["y"])
        output = self.net(features)
        return self.criterion(output, labels.long())

    def inference(self, input):
        features = to_device(input["x"])
        labels = to_device(input["y"])
        output = self.net(features)
        n_samples = features.shape[0]
        accuracy = torch.mean(
            (torch.argmax(output, dim=1) == labels).float()).item()
        return {"output": output, "acc": accuracy, "batch_size": n_samples}
