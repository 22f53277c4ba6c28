"""Privacy attack metrics (reference: extensions/privacy/metrics.py:10-76).

Two attacks evaluated client-side to decide whether a client's update leaks
too much (clients over threshold are dropped — reference client.py:474-506):

* embedding-gradient token extraction: tokens present in a batch leave
  higher-L2 rows in the embedding gradient;
* practical-ε leakage: perplexity ratio before/after an attacker Adamax
  step on the client's gradient.
"""

from __future__ import annotations

import logging
from copy import deepcopy

import numpy as np
import torch

from ...utils import make_optimizer, print_rank


def extract_indices_from_embeddings(gradients, batch, embed_size, vocab_size):
    """Reference: privacy/metrics.py:10-22.  ``gradients`` is the flat
    gradient vector whose first vocab×embed block is the embedding grad."""
    batch = torch.cat([b.reshape(-1) for b in batch]).cpu().detach().numpy()
    embed_grad = gradients[:embed_size * vocab_size].reshape(vocab_size, embed_size)
    valid_batch = batch[batch > 0]
    tot_tokens = len(batch)
    extracted_indices = torch.argsort(
        embed_grad.norm(dim=-1), descending=True)[:tot_tokens].cpu().numpy()
    extracted_ratio = float(np.isin(valid_batch, extracted_indices).mean()) \
        if len(valid_batch) else 0.0
    return extracted_ratio, np.intersect1d(extracted_indices, valid_batch)


def compute_perplexity(encoded_batch, model):
    """Per-token log-softmax at the target indices
    (reference: privacy/metrics.py:25-30)."""
    outputs = model.inference(encoded_batch)
    out = outputs["output"]
    batch_size, seq_len, vocab_size = out.shape
    logp = torch.nn.functional.log_softmax(out, dim=-1)
    flat_idx = encoded_batch.reshape(-1).cpu().clamp(min=0)  # pads < 0 -> 0
    return logp.reshape(-1, vocab_size)[
        np.arange(batch_size * seq_len), flat_idx].reshape(batch_size, seq_len)


def practical_epsilon_leakage(original_params, model, encoded_batches,
                              is_weighted_leakage=True, max_ratio=1e9,
                              optimizer_config=None):
    """Reference: privacy/metrics.py:33-76."""
    current_params = deepcopy(model.state_dict())
    current_gradients = {n: p.grad.clone().detach()
                         for n, p in model.named_parameters()}
    model.load_state_dict(original_params)
    pre_perplex, post_perplex = [], []
    tolerance = 1 / max_ratio
    max_leakage = 0.0

    model.loss(encoded_batches[0][:1]).backward()
    for p in model.parameters():
        if p.grad is not None:
            p.grad.zero_()

    with torch.no_grad():
        for encoded_batch in encoded_batches:
            pre_perplex.append(compute_perplexity(encoded_batch, model))
        for n, p in model.named_parameters():
            p.grad = current_gradients[n]
        if optimizer_config is None:
            optimizer_config = {"lr": 0.03, "amsgrad": False, "type": "adamax"}
        make_optimizer(dict(optimizer_config), model).step()
        for encoded_batch in encoded_batches:
            post_perplex.append(compute_perplexity(encoded_batch, model))
        for pre, post in zip(pre_perplex, post_perplex):
            leakage = ((pre + tolerance) / (post + tolerance)).clamp_(0, max_ratio)
            if is_weighted_leakage:
                weight_leakage = torch.max(pre.exp(), post.exp()) * leakage
            else:
                weight_leakage = leakage
            max_leakage = max(max_leakage, weight_leakage.max().item())
    print_rank(f"raw max leakage: {max_leakage}", loglevel=logging.DEBUG)

    model.load_state_dict(current_params)
    for n, p in model.named_parameters():
        p.grad = current_gradients[n]
    return max(float(np.log(max_leakage)) if max_leakage > 0 else 0.0, 0.0)
