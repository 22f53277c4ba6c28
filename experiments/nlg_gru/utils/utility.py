"""nlg_gru vocab helpers (reference: experiments/nlg_gru/utils/utility.py).

Vocab file format: JSON — either a list of tokens or a ``{token: count}``
dict (ordered by insertion).  Index 0 is reserved for OOV/``<UNK>``.
"""

import json


class Vocab:
    def __init__(self, tokens):
        self.idx_to_term = ["<UNK>"] + [t for t in tokens if t != "<UNK>"]
        self.term_to_idx = {t: i for i, t in enumerate(self.idx_to_term)}

    def __len__(self):
        return len(self.idx_to_term)


def load_vocab(path_or_tokens):
    if path_or_tokens is None:
        return None
    if isinstance(path_or_tokens, (list, tuple)):
        return Vocab(path_or_tokens)
    with open(path_or_tokens, "r") as f:
        obj = json.load(f)
    return Vocab(list(obj.keys()) if isinstance(obj, dict) else list(obj))


def case_backoff(word, term_to_idx):
    """Pick the casing variant of ``word`` present in the vocab
    (reference utility: case_backoff_batch)."""
    for cand in (word, word.lower(), word.capitalize(), word.upper()):
        if cand in term_to_idx:
            return cand
    return word


def to_indices(vocab, words):
    return [vocab.term_to_idx.get(w, 0) for w in words]
