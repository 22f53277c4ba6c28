"""nlg_gru dataset: per-user utterance lists (variable-length token-id
sequences).  Reference: experiments/nlg_gru/dataloaders/dataset.py.

Blob convention: ``user_data[user]['x']`` is a list of utterances; each
utterance is a list of token ids (``preencoded: true``) or words.
"""

import json
import os
from importlib.machinery import SourceFileLoader

import numpy as np

from msrflute_amd.core.dataset import BaseDataset

_util = SourceFileLoader(
    "nlg_gru_utility",
    os.path.join(os.path.dirname(__file__), "..", "utils", "utility.py")
).load_module()


class Dataset(BaseDataset):
    def __init__(self, data, min_num_words=2, max_num_words=25,
                 test_only=False, user_idx=0, vocab_dict=None,
                 preencoded=False, args=None, **kwargs):
        args = args or {}
        self.test_only = test_only
        self.min_num_words = int(args.get("min_num_words", min_num_words))
        self.max_num_words = int(args.get("max_num_words", max_num_words))
        self.preencoded = bool(args.get("preencoded", preencoded))
        self.vocab = _util.load_vocab(args.get("vocab_dict", vocab_dict))
        self.vocab_size = len(self.vocab) if self.vocab else None
        self.utt_list = []
        self.load_data(data, user_idx)

    def load_data(self, data=None, user_idx=-1):
        if isinstance(data, str):
            from msrflute_amd.models.generic_data import load_blob
            data = load_blob(data)
        self.user_list = list(data["users"])
        self.num_samples = list(data["num_samples"])
        self.user_data = data["user_data"]
        self.user_data_label = data.get("user_data_label")
        self.user = ("test_only" if self.test_only or user_idx == -1
                     else self.user_list[user_idx])
        users = (self.user_list if self.test_only or user_idx == -1
                 else [self.user_list[user_idx]])
        for u in users:
            ud = self.user_data[u]
            for utt in (ud["x"] if isinstance(ud, dict) else ud):
                toks = utt if isinstance(utt, list) else utt.split()
                if len(toks) <= self.min_num_words:
                    continue
                toks = toks[: self.max_num_words]
                self.utt_list.append({"src_text": toks,
                                      "duration": len(toks),
                                      "loss_weight": 1.0})

    def __len__(self):
        return len(self.utt_list)

    def __getitem__(self, idx):
        toks = self.utt_list[idx]["src_text"]
        if self.preencoded:
            ids = np.asarray(toks, dtype=np.int64)
        else:
            toks = [_util.case_backoff(w, self.vocab.term_to_idx) for w in toks]
            ids = np.asarray(_util.to_indices(self.vocab, toks), dtype=np.int64)
        return ids, self.user
