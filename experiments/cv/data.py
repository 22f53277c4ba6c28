"""Non-IID partitioners for the cv personalization task.

Reference: experiments/cv/data.py (Dirichlet partitioner from FedML at
data.py:119-172, fixed-label-distribution mode at 67-115, per-client
rotation transforms at 44-60).  Operates on in-memory arrays — there is no
dataset download path; blobs come from tools/create_data.py or user files.
"""

import numpy as np


def dirichlet_partition(labels, n_clients, alpha, num_classes, seed):
    """Per-class Dirichlet split with the FedML balancing rule: resample
    until every client has at least ``num_classes`` samples."""
    labels = np.asarray(labels)
    N = len(labels)
    rng = np.random.RandomState(seed)
    min_size = 0
    while min_size < num_classes:
        idx_batch = [[] for _ in range(n_clients)]
        for k in range(num_classes):
            idx_k = np.where(labels == k)[0]
            rng.shuffle(idx_k)
            prop = rng.dirichlet(np.repeat(alpha, n_clients))
            # zero out clients already at their fair share
            prop = np.array([p * (len(b) < N / n_clients)
                             for p, b in zip(prop, idx_batch)])
            prop = prop / prop.sum()
            cuts = (np.cumsum(prop) * len(idx_k)).astype(int)[:-1]
            for b, chunk in zip(idx_batch, np.split(idx_k, cuts)):
                b.extend(chunk.tolist())
        min_size = min(len(b) for b in idx_batch)
    for b in idx_batch:
        rng.shuffle(b)
    return idx_batch


def fixed_label_partition(labels, n_clients, labels_per_client, seed):
    """Each client sees a fixed subset of classes (reference fixed-label
    mode, data.py:67-115): classes assigned round-robin, samples of each
    class split evenly among its clients."""
    labels = np.asarray(labels)
    num_classes = int(labels.max()) + 1
    rng = np.random.RandomState(seed)
    client_classes = [[(c * labels_per_client + j) % num_classes
                       for j in range(labels_per_client)]
                      for c in range(n_clients)]
    holders = {k: [i for i, cc in enumerate(client_classes) if k in cc]
               for k in range(num_classes)}
    idx_batch = [[] for _ in range(n_clients)]
    for k in range(num_classes):
        idx_k = np.where(labels == k)[0]
        rng.shuffle(idx_k)
        who = holders[k]
        if not who:
            continue
        for j, chunk in enumerate(np.array_split(idx_k, len(who))):
            idx_batch[who[j]].extend(chunk.tolist())
    for b in idx_batch:
        rng.shuffle(b)
    return idx_batch


def rotation_for_client(client_idx, n_rotations=4):
    """Per-client rotation angle in multiples of 90° (reference per-client
    transform, data.py:44-60) — applied as a tensor rot90, no PIL."""
    return (client_idx % n_rotations)


def partition_blob(x, y, n_clients, alpha=1.0, seed=2020, mode="dirichlet",
                   labels_per_client=2):
    """Turn a flat (x, y) array pair into the universal federated blob."""
    num_classes = int(np.asarray(y).max()) + 1
    if mode == "dirichlet":
        idx_batch = dirichlet_partition(y, n_clients, alpha, num_classes, seed)
    else:
        idx_batch = fixed_label_partition(y, n_clients, labels_per_client,
                                          seed)
    users, num_samples, user_data, user_labels = [], [], {}, {}
    for c, idxs in enumerate(idx_batch):
        name = f"client{c:05d}"
        users.append(name)
        num_samples.append(len(idxs))
        user_data[name] = {"x": np.asarray(x)[idxs]}
        user_labels[name] = np.asarray(y)[idxs]
    return {"users": users, "num_samples": num_samples,
            "user_data": user_data, "user_data_label": user_labels}
