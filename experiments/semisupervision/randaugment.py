"""Tensor-space RandAugment for the semisupervision task.

Equivalent role to the reference's PIL-based RandAugment
(experiments/semisupervision/randaugment.py, 289 LoC).  Offline/MI355X
divergence: operates directly on normalized float CHW tensors (no PIL in
the image), with geometric ops (flip / rot90 / integer translate / cutout)
and photometric ops (brightness, contrast, invert, solarize, noise).
"""

import random

import torch


def _translate(x, mag):
    sh = int(mag * x.shape[-1])
    if sh == 0:
        return x
    dims = (-1,) if random.random() < 0.5 else (-2,)
    return torch.roll(x, shifts=sh if random.random() < 0.5 else -sh,
                      dims=dims)


def _cutout(x, mag):
    h, w = x.shape[-2:]
    size = max(1, int(mag * h))
    cy, cx = random.randrange(h), random.randrange(w)
    y0, y1 = max(0, cy - size // 2), min(h, cy + size // 2 + 1)
    x0, x1 = max(0, cx - size // 2), min(w, cx + size // 2 + 1)
    out = x.clone()
    out[..., y0:y1, x0:x1] = 0.0
    return out


OPS = [
    lambda x, m: x,                                          # identity
    lambda x, m: torch.flip(x, dims=(-1,)),                  # hflip
    lambda x, m: torch.rot90(x, 1 if random.random() < 0.5 else 3,
                             dims=(-2, -1)),                 # rot90
    _translate,
    _cutout,
    lambda x, m: x + m,                                      # brightness
    lambda x, m: x * (1.0 + m),                              # contrast
    lambda x, m: -x,                                         # invert
    lambda x, m: torch.where(x > m, -x, x),                  # solarize
    lambda x, m: x + m * torch.randn_like(x),                # noise
]


class RandAugment:
    def __init__(self, n=2, magnitude=0.3):
        self.n = n
        self.magnitude = magnitude

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        for op in random.sample(OPS, self.n):
            m = random.uniform(0, self.magnitude)
            x = op(x, m)
        return x
