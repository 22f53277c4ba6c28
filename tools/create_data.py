#!/usr/bin/env python3
"""Synthetic federated-data generator.

Equivalent role to the reference's dummy-data generator
(testing/create_data.py:81-173): emits data blobs in the universal FLUTE
layout ``{users, num_samples, user_data, user_data_label}``.  All data is
synthetic (random): there is no network access for real datasets; bench
and tests consume these shapes only.
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import torch


def make_classification_blob(n_users, samples_per_user, feature_shape,
                             n_classes, seed=0, dtype=np.float32,
                             flat=False):
    rng = np.random.default_rng(seed)
    users, num_samples, user_data, user_data_label = [], [], {}, {}
    for u in range(n_users):
        name = f"user{u:05d}"
        n = samples_per_user if np.isscalar(samples_per_user) else int(
            rng.integers(samples_per_user[0], samples_per_user[1] + 1))
        shape = (n, int(np.prod(feature_shape))) if flat else (n, *feature_shape)
        x = rng.standard_normal(shape).astype(dtype)
        y = rng.integers(0, n_classes, size=n)
        users.append(name)
        num_samples.append(n)
        user_data[name] = {"x": x}
        user_data_label[name] = y
    return {"users": users, "num_samples": num_samples,
            "user_data": user_data, "user_data_label": user_data_label}


def make_mnist_blob(n_users=1000, samples_per_user=60, seed=0):
    return make_classification_blob(n_users, samples_per_user, (784,), 10,
                                    seed=seed, flat=True)


def make_femnist_blob(n_users=3400, samples_per_user=100, seed=0):
    return make_classification_blob(n_users, samples_per_user, (28, 28), 62,
                                    seed=seed)


def make_char_lm_blob(n_users, samples_per_user, seq_len=80, vocab=90, seed=0):
    """Shakespeare-style char-LM shards: x int sequences, y next-char
    sequences."""
    rng = np.random.default_rng(seed)
    users, num_samples, user_data, user_data_label = [], [], {}, {}
    for u in range(n_users):
        name = f"user{u:05d}"
        n = samples_per_user
        x = rng.integers(1, vocab, size=(n, seq_len))
        y = np.roll(x, -1, axis=1)
        users.append(name)
        num_samples.append(n)
        user_data[name] = {"x": x}
        user_data_label[name] = y
    return {"users": users, "num_samples": num_samples,
            "user_data": user_data, "user_data_label": user_data_label}


def make_fedcifar100_blob(n_users=500, samples_per_user=100, seed=0):
    return make_classification_blob(n_users, samples_per_user, (3, 24, 24),
                                    100, seed=seed)


def make_cifar10_blob(n_users=50, samples_per_user=100, seed=0):
    return make_classification_blob(n_users, samples_per_user, (3, 32, 32),
                                    10, seed=seed)


def make_ecg_blob(n_users=50, samples_per_user=100, seed=0):
    return make_classification_blob(n_users, samples_per_user, (187,), 5,
                                    seed=seed, flat=True)


def make_shakespeare_blob(n_users=715, samples_per_user=50, seed=0):
    return make_char_lm_blob(n_users, samples_per_user, seq_len=80, vocab=90,
                             seed=seed)


def make_nlg_blob(n_users=25, utts_per_user=20, vocab=1000, max_words=25,
                  seed=0):
    """Reddit-style variable-length preencoded utterances (nlg_gru /
    mlm_bert shape)."""
    rng = np.random.default_rng(seed)
    users, num_samples, user_data = [], [], {}
    for u in range(n_users):
        name = f"user{u:05d}"
        utts = [rng.integers(1, vocab, size=int(
            rng.integers(3, max_words + 1))).tolist()
            for _ in range(utts_per_user)]
        users.append(name)
        num_samples.append(len(utts))
        user_data[name] = {"x": utts}
    return {"users": users, "num_samples": num_samples,
            "user_data": user_data}


def make_newsrec_blob(n_users=20, samples_per_user=8, vocab=5000, hist=50,
                      title=30, k=1 + 4, seed=0):
    """MIND-style pretokenized click/candidate slates (fednewsrec shape).
    Train labels: clicked index; eval consumes the same as binary-ish."""
    rng = np.random.default_rng(seed)
    users, num_samples, user_data, user_labels = [], [], {}, {}
    for u in range(n_users):
        name = f"user{u:05d}"
        n = samples_per_user
        user_data[name] = {
            "history": rng.integers(1, vocab, size=(n, hist, title)),
            "candidates": rng.integers(1, vocab, size=(n, k, title)),
            "labels": rng.integers(0, k, size=n),
        }
        users.append(name)
        num_samples.append(n)
        user_labels[name] = user_data[name]["labels"]
    return {"users": users, "num_samples": num_samples,
            "user_data": user_data, "user_data_label": user_labels}


def make_newsrec_eval_blob(n_users=4, samples_per_user=8, vocab=5000,
                           hist=50, title=30, k=1 + 4, seed=0):
    """Eval variant: labels are [n, k] binary relevance vectors."""
    blob = make_newsrec_blob(n_users, samples_per_user, vocab, hist, title,
                             k, seed)
    rng = np.random.default_rng(seed + 99)
    for u in blob["users"]:
        n = len(blob["user_data"][u]["labels"])
        lab = np.zeros((n, k), dtype=np.int64)
        lab[np.arange(n), rng.integers(0, k, size=n)] = 1
        blob["user_data"][u]["labels"] = lab
        blob["user_data_label"][u] = lab
    return blob


def make_cv_flat_blob(n_samples=2000, seed=0, img=32, n_classes=10, **kw):
    """Flat (non-federated) CIFAR-shaped arrays; the cv task partitions
    them Dirichlet-style on load (experiments/cv/data.py)."""
    rng = np.random.default_rng(seed)
    return {"x": rng.standard_normal((n_samples, 3, img, img)).astype(np.float32),
            "y": rng.integers(0, n_classes, size=n_samples)}


def save_blob(blob, path):
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    if path.endswith(".pt"):
        torch.save(blob, path)
    else:
        import json

        def clean(o):
            if isinstance(o, np.ndarray):
                return o.tolist()
            if isinstance(o, dict):
                return {k: clean(v) for k, v in o.items()}
            if isinstance(o, list):
                return [clean(v) for v in o]
            return o
        with open(path, "w") as f:
            json.dump(clean(blob), f)


TASKS = {
    "cv_lr_mnist": (make_mnist_blob, dict(n_users=1000, samples_per_user=60)),
    "cv_cnn_femnist": (make_femnist_blob, dict(n_users=3400, samples_per_user=100)),
    "cv_resnet_fedcifar100": (make_fedcifar100_blob,
                              dict(n_users=500, samples_per_user=100)),
    "classif_cnn": (make_cifar10_blob, dict(n_users=50, samples_per_user=100)),
    "ecg_cnn": (make_ecg_blob, dict(n_users=50, samples_per_user=100)),
    "nlp_rnn_fedshakespeare": (make_shakespeare_blob,
                               dict(n_users=715, samples_per_user=50)),
    "nlg_gru": (make_nlg_blob, dict(n_users=25, utts_per_user=20)),
    "mlm_bert": (make_nlg_blob, dict(n_users=25, utts_per_user=20)),
    "cv": (make_cv_flat_blob, dict(n_samples=2000)),
    "semisupervision": (make_cifar10_blob,
                        dict(n_users=20, samples_per_user=40)),
    "fednewsrec": (make_newsrec_blob, dict(n_users=20, samples_per_user=8)),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--task", required=True, choices=sorted(TASKS))
    ap.add_argument("--out", required=True, help="output directory")
    ap.add_argument("--users", type=int, default=None)
    ap.add_argument("--samples", type=int, default=None)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    fn, kw = TASKS[args.task]
    kw = dict(kw)
    if args.users:
        kw["n_users"] = args.users
    if args.samples:
        kw["samples_per_user"] = args.samples
    train = fn(seed=args.seed, **kw)
    kw_eval = dict(kw, n_users=max(kw["n_users"] // 20, 2))
    val = fn(seed=args.seed + 1, **kw_eval)
    test = fn(seed=args.seed + 2, **kw_eval)
    save_blob(train, os.path.join(args.out, args.task, "train_data.pt"))
    save_blob(val, os.path.join(args.out, args.task, "val_data.pt"))
    save_blob(test, os.path.join(args.out, args.task, "test_data.pt"))
    print(f"wrote {args.task} blobs to {os.path.join(args.out, args.task)}")


if __name__ == "__main__":
    main()
