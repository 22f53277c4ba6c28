"""Fed-CIFAR100 task: ResNet-18 with optional GroupNorm.

Benchmark task 3 (reference: experiments/cv_resnet_fedcifar100/model.py:16-271).
The reference emulates GroupNorm by reshaping into ``F.batch_norm``
(group_normalization.py:10-90); here we use the native ``nn.GroupNorm``
(same math, one fused MIOpen/HIP op instead of reshape+batch_norm+reshape).
Config knobs: ``num_classes`` (reference default 1000), ``group_norm`` =
channels-per-group (0 = BatchNorm, the reference's effective default since
``resnet18()`` is called with no kwargs at model.py:253).
"""

import torch
from torch import nn
from torch.nn import functional as F

from msrflute_amd.models.classification import ClassificationModel


def _norm(planes: int, channels_per_group: int) -> nn.Module:
    if channels_per_group > 0:
        return nn.GroupNorm(max(1, planes // channels_per_group), planes)
    return nn.BatchNorm2d(planes)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1, cpg=0):
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride,
                               padding=1, bias=False)
        self.n1 = _norm(planes, cpg)
        self.conv2 = nn.Conv2d(planes, planes, 3, padding=1, bias=False)
        self.n2 = _norm(planes, cpg)
        self.down = None
        if stride != 1 or in_planes != planes:
            self.down = nn.Sequential(
                nn.Conv2d(in_planes, planes, 1, stride=stride, bias=False),
                _norm(planes, cpg))

    def forward(self, x):
        out = F.relu(self.n1(self.conv1(x)))
        out = self.n2(self.conv2(out))
        out = out + (self.down(x) if self.down is not None else x)
        return F.relu(out)


class ResNetNet(nn.Module):
    """ResNet-18/34-style network with an ImageNet stem (7x7/2 + maxpool),
    matching the reference topology for checkpoint-shape parity."""

    def __init__(self, layers=(2, 2, 2, 2), num_classes=1000, cpg=0):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.n1 = _norm(64, cpg)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.in_planes = 64
        self.layer1 = self._make(64, layers[0], 1, cpg)
        self.layer2 = self._make(128, layers[1], 2, cpg)
        self.layer3 = self._make(256, layers[2], 2, cpg)
        self.layer4 = self._make(512, layers[3], 2, cpg)
        self.fc = nn.Linear(512, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")

    def _make(self, planes, blocks, stride, cpg):
        strides = [stride] + [1] * (blocks - 1)
        seq = []
        for s in strides:
            seq.append(BasicBlock(self.in_planes, planes, s, cpg))
            self.in_planes = planes
        return nn.Sequential(*seq)

    def forward(self, x):
        x = self.maxpool(F.relu(self.n1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = torch.flatten(F.adaptive_avg_pool2d(x, 1), 1)
        return self.fc(x)


class ResNet(ClassificationModel):
    def __init__(self, model_config):
        super().__init__(net=ResNetNet(
            num_classes=model_config.get("num_classes", 1000),
            cpg=model_config.get("group_norm", 0)))
