"""semisupervision task: CIFAR-stem ResNet (FedLabels strategy).

Reference: experiments/semisupervision/model.py:16-183 (``Res`` BaseModel
over a CIFAR ResNet-18).  ``forward`` returns raw logits because the
FedLabels trainer calls the model directly for pseudo-label estimation
(msrflute_amd/extensions/fedlabels_train.py).
"""

import os
from importlib.machinery import SourceFileLoader

import torch
from torch.nn import functional as F

from msrflute_amd.core.model import BaseModel
from msrflute_amd.utils import to_device

_cv = SourceFileLoader(
    "cv_model_shared",
    os.path.join(os.path.dirname(__file__), "..", "cv", "model.py")
).load_module()


class Res(BaseModel):
    def __init__(self, model_config):
        super().__init__()
        self.net = _cv.CifarResNet(
            num_classes=model_config.get("num_classes", 100),
            cpg=model_config.get("group_norm", 0))

    def forward(self, x):
        return self.net(to_device(x))

    def loss(self, input):
        if isinstance(input, dict):
            x, y = input["x"], input["y"]
        else:
            x, y = input
        x, y = to_device(x), to_device(y)
        return F.cross_entropy(self.net(x), y.long())

    def inference(self, input):
        if isinstance(input, dict):
            x, y = input["x"], input["y"]
        else:
            x, y = input
        x, y = to_device(x), to_device(y)
        output = self.net(x)
        acc = (torch.argmax(output, dim=1) == y).float().mean().item()
        return {"output": output, "acc": acc, "batch_size": x.shape[0]}
