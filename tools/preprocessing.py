#!/usr/bin/env python3
"""Dataset preprocessing utilities.

Equivalent role to the reference's ``utils/preprocessing/`` tools
(create-json.py: reddit TSV -> per-user JSON; from_json_to_hdf5.py /
create-hdf5.py: JSON -> HDF5).  h5py is not in this image, so the packed
binary format here is torch ``.pt`` / numpy ``.npz`` — both readable by
``msrflute_amd.models.generic_data.load_blob``.
"""

from __future__ import annotations

import argparse
import csv
import json
import os

import numpy as np


def tsv_to_json(tsv_path, out_path, user_col=0, text_col=1, delimiter="\t",
                min_utts=1):
    """Group a TSV of (user, text) rows into the universal FLUTE blob
    (reference: utils/preprocessing/create-json.py)."""
    per_user = {}
    with open(tsv_path, newline="") as f:
        for row in csv.reader(f, delimiter=delimiter):
            if len(row) <= max(user_col, text_col):
                continue
            per_user.setdefault(row[user_col], []).append(row[text_col])
    users, num_samples, user_data = [], [], {}
    for u, texts in per_user.items():
        if len(texts) < min_utts:
            continue
        users.append(u)
        num_samples.append(len(texts))
        user_data[u] = {"x": texts}
    blob = {"users": users, "num_samples": num_samples,
            "user_data": user_data}
    with open(out_path, "w") as f:
        json.dump(blob, f)
    return blob


def json_to_packed(json_path, out_path):
    """Convert a JSON blob to the packed binary format (.pt or .npz)
    (reference: utils/preprocessing/from_json_to_hdf5.py, hdf5 replaced)."""
    with open(json_path) as f:
        blob = json.load(f)
    if out_path.endswith(".npz"):
        np.savez_compressed(out_path,
                            **{k: np.asarray(v, dtype=object)
                               if isinstance(v, (dict,)) else v
                               for k, v in blob.items()})
    else:
        import torch
        torch.save(blob, out_path)
    return out_path


def build_vocab(json_path, out_path, vocab_size=10000):
    """Build a frequency-ordered vocab JSON from a text blob
    (reference: testing/build_vocab.py role)."""
    with open(json_path) as f:
        blob = json.load(f)
    counts = {}
    for u in blob["users"]:
        ud = blob["user_data"][u]
        for utt in (ud["x"] if isinstance(ud, dict) else ud):
            words = utt.split() if isinstance(utt, str) else utt
            for w in words:
                counts[str(w)] = counts.get(str(w), 0) + 1
    top = dict(sorted(counts.items(), key=lambda kv: -kv[1])[:vocab_size])
    with open(out_path, "w") as f:
        json.dump(top, f)
    return top


def main():
    ap = argparse.ArgumentParser()
    sub = ap.add_subparsers(dest="cmd", required=True)
    p1 = sub.add_parser("tsv-to-json")
    p1.add_argument("tsv"); p1.add_argument("out")
    p1.add_argument("--user-col", type=int, default=0)
    p1.add_argument("--text-col", type=int, default=1)
    p2 = sub.add_parser("json-to-packed")
    p2.add_argument("json"); p2.add_argument("out")
    p3 = sub.add_parser("build-vocab")
    p3.add_argument("json"); p3.add_argument("out")
    p3.add_argument("--vocab-size", type=int, default=10000)
    args = ap.parse_args()
    if args.cmd == "tsv-to-json":
        tsv_to_json(args.tsv, args.out, args.user_col, args.text_col)
    elif args.cmd == "json-to-packed":
        json_to_packed(args.json, args.out)
    else:
        build_vocab(args.json, args.out, args.vocab_size)


if __name__ == "__main__":
    main()
