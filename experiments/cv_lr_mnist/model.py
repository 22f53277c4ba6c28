"""LR-MNIST task: 784→10 logistic regression.

Benchmark task 1 (reference: experiments/cv_lr_mnist/model.py:12-47;
architecture from the FLUTE/FedML MNIST benchmark — sigmoid over a single
linear layer fed into CE loss, kept for accuracy parity).
"""

import torch

from msrflute_amd.models.classification import ClassificationModel


class LogisticRegressionNet(torch.nn.Module):
    def __init__(self, input_dim, output_dim):
        super().__init__()
        self.linear = torch.nn.Linear(input_dim, output_dim)
        self.input_dim = input_dim

    def forward(self, x):
        return torch.sigmoid(self.linear(x.view(-1, self.input_dim)))


class LR(ClassificationModel):
    def __init__(self, model_config):
        super().__init__(net=LogisticRegressionNet(
            model_config.get("input_dim", 784),
            model_config.get("output_dim", 10)))
