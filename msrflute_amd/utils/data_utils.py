"""Batch samplers (reference: utils/data_utils.py:9-119)."""

from __future__ import annotations

import random

from torch.utils.data import sampler


class BatchSampler(sampler.Sampler):
    """Contiguous batches in dataset order, shuffled at the batch level, so
    nearby (similar-length) samples share a batch.
    Reference: utils/data_utils.py:9-39.
    """

    def __init__(self, dataset, batch_size, randomize=True, drop_last=False):
        self.dataset = dataset
        self.batch_size = batch_size
        self.randomize = randomize
        n = len(dataset)
        batches = [list(range(b, min(b + batch_size, n))) for b in range(0, n, batch_size)]
        if drop_last and batches and len(batches[-1]) < batch_size:
            del batches[-1]
        self.batches = batches

    def __iter__(self):
        if self.randomize:
            random.shuffle(self.batches)
        return iter(self.batches)

    def __len__(self):
        return len(self.batches) * self.batch_size


class DynamicBatchSampler(sampler.Sampler):
    """Duration-aware packing: sort utterances by duration and pack batches
    under a total-frame budget (``frames_threshold``) and optional
    ``max_batch_size``; in ``unsorted_batch`` mode pack purely by count.
    Reference: utils/data_utils.py:42-119.
    """

    def __init__(self, base_sampler, frames_threshold, max_batch_size=0,
                 unsorted_batch=False, fps=1000 / 30):
        self.sampler = base_sampler
        self.frames_threshold = frames_threshold
        self.max_batch_size = max_batch_size
        self.unsorted_batch = unsorted_batch

        dataset = self.sampler.dataset
        indices = [(idx, dataset.utt_list[idx]["duration"]) for idx in self.sampler]
        if not unsorted_batch:
            indices.sort(key=lambda elem: elem[1])

        batches = []
        batch, batch_frames, max_frames_in_batch = [], 0.0, 0.0
        for idx, duration in indices:
            if duration <= 0:
                continue
            frames = duration * fps
            max_frames_in_batch = max(max_frames_in_batch, frames)
            fits = (
                (unsorted_batch and len(batch) < max_batch_size)
                or (not unsorted_batch
                    and batch_frames + frames <= self.frames_threshold
                    and (max_batch_size == 0 or len(batch) < max_batch_size))
            )
            if fits:
                batch.append(idx)
                batch_frames += frames
            else:
                if batch:
                    batches.append(batch)
                batch, batch_frames = [idx], frames
                max_frames_in_batch = frames
        if batch:
            batches.append(batch)
        self.batches = batches

    def __iter__(self):
        random.shuffle(self.batches)
        return iter(self.batches)

    def __len__(self):
        return len(self.batches)
