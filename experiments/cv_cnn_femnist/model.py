"""CNN-FEMNIST task: 2×conv + 2×FC with dropout, 62 classes.

Benchmark task 2 and the north-star bench model (reference:
experiments/cv_cnn_femnist/model.py:12-105; architecture recommended by
"Adaptive Federated Optimization", arXiv:2003.00295).
"""

import torch
from torch import nn

from msrflute_amd.models.classification import ClassificationModel


class FEMNISTNet(nn.Module):
    def __init__(self, num_classes=62):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, kernel_size=3)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=3)
        self.pool = nn.MaxPool2d(2, stride=2)
        self.drop1 = nn.Dropout(0.25)
        self.fc1 = nn.Linear(9216, 128)
        self.drop2 = nn.Dropout(0.5)
        self.fc2 = nn.Linear(128, num_classes)

    def forward(self, x):
        if x.dim() == 3:
            x = x.unsqueeze(1)  # [B, 28, 28] -> [B, 1, 28, 28]
        x = torch.relu(self.conv1(x))
        x = torch.relu(self.conv2(x))
        x = self.drop1(self.pool(x))
        x = torch.flatten(x, 1)
        x = self.drop2(torch.relu(self.fc1(x)))
        return self.fc2(x)


class CNN(ClassificationModel):
    def __init__(self, model_config):
        super().__init__(net=FEMNISTNet(model_config.get("num_classes", 62)))
