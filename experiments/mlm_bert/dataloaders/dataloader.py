"""mlm_bert dataloader with built-in MLM masking collation.

Reference: experiments/mlm_bert/dataloaders/dataloader.py:57-61 uses HF
``DataCollatorForLanguageModeling``; here the 80/10/10 masking is applied
directly (works with the offline pretokenized tokenizer as well as a local
HF tokenizer), batches are ``{input_ids, attention_mask, labels}``.
"""

import os
from importlib.machinery import SourceFileLoader

import numpy as np
import torch

from msrflute_amd.core.dataloader import BaseDataLoader

_ds_mod = SourceFileLoader(
    "mlm_bert_dataset",
    os.path.join(os.path.dirname(__file__), "dataset.py")).load_module()
_Dataset = _ds_mod.Dataset


class DataLoader(BaseDataLoader):
    def __init__(self, data, user_idx=0, mode="train", args=None, **kwargs):
        args = args or {}
        self.mode = mode
        self.args = args
        self.batch_size = int(args.get("batch_size", 8))
        self.mlm_prob = float(args.get("mlm_probability", 0.15))
        self.dataset = _Dataset(
            data, args=args, test_only=(mode != "train"),
            user_idx=user_idx if mode == "train" else -1)
        tok = self.dataset.tokenizer
        if tok is not None:
            self.mask_id = tok.mask_token_id
            self.vocab_size = len(tok)
            self.special_ids = set(tok.all_special_ids)
        else:
            self.mask_id = _ds_mod.MASK
            self.vocab_size = int(args.get("vocab_size", 1000))
            self.special_ids = {_ds_mod.PAD, _ds_mod.MASK, _ds_mod.CLS,
                                _ds_mod.SEP}
        self._rng = np.random.default_rng(
            torch.randint(0, 2 ** 31 - 1, (1,)).item())

    def create_loader(self):
        return self

    def __len__(self):
        n = len(self.dataset)
        return max(1, (n + self.batch_size - 1) // self.batch_size) if n else 0

    def _collate(self, idxs):
        ids = np.stack([self.dataset[i][0] for i in idxs])
        attn = np.stack([self.dataset[i][1] for i in idxs])
        labels = np.full_like(ids, -100)
        # eligible = real tokens that are not special
        eligible = (attn == 1)
        for s in self.special_ids:
            eligible &= ids != s
        pick = (self._rng.random(ids.shape) < self.mlm_prob) & eligible
        labels[pick] = ids[pick]
        r = self._rng.random(ids.shape)
        ids = ids.copy()
        ids[pick & (r < 0.8)] = self.mask_id                      # 80% [MASK]
        rand = (r >= 0.8) & (r < 0.9) & pick                      # 10% random
        ids[rand] = self._rng.integers(0, self.vocab_size,
                                       size=int(rand.sum()))
        return {"input_ids": torch.from_numpy(ids),
                "attention_mask": torch.from_numpy(attn),
                "labels": torch.from_numpy(labels)}

    def __iter__(self):
        n = len(self.dataset)
        if n == 0:
            return
        order = (torch.randperm(n) if self.mode == "train"
                 else torch.arange(n)).tolist()
        for s in range(0, n, self.batch_size):
            yield self._collate(order[s:s + self.batch_size])
