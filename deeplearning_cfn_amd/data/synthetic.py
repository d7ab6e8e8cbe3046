"""Synthetic data generators for the benchmark configs.

There is no network access for datasets (synthetic data /
random-init weights); these generate ImageNet-shape, CIFAR-shape and
BERT-token batches deterministically per (seed, rank, step) so loss
curves are exactly reproducible across world sizes — the correctness
check of SURVEY.md §5 item 4.
"""

from __future__ import annotations

from typing import Iterator, Tuple

import torch


class SyntheticImages:
    """Yields (images, labels); shapes per config.

    ImageNet-shape: 3x224x224, 1000 classes (config #2/#3/#5);
    CIFAR-shape:    3x32x32, 10 classes (config #1).
    """

    def __init__(self, batch_size: int, image_size: int = 224,
                 num_classes: int = 1000, seed: int = 0, rank: int = 0,
                 dtype: torch.dtype = torch.float32,
                 device: str = "cpu", channels_last: bool = False,
                 pinned: bool = False, pool: int = 0):
        self.batch_size = batch_size
        self.image_size = image_size
        self.num_classes = num_classes
        self.dtype = dtype
        self.device = device
        self.channels_last = channels_last
        self.pinned = pinned
        # pool > 0: cycle a FIXED set of batches.  Endless fresh random
        # labels have ln(num_classes) as their optimum, so a decreasing
        # loss (memorization) needs a finite pool — the synthetic-data
        # analogue of the reference's "CIFAR example converges" check.
        self.pool = pool
        self._pool_batches = []
        self.gen = torch.Generator().manual_seed(seed * 100003 + rank)

    def batch(self) -> Tuple[torch.Tensor, torch.Tensor]:
        x = torch.randn(self.batch_size, 3, self.image_size, self.image_size,
                        generator=self.gen, dtype=torch.float32)
        y = torch.randint(0, self.num_classes, (self.batch_size,),
                          generator=self.gen)
        x = x.to(self.dtype)
        if self.channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        if self.pinned:
            x, y = x.pin_memory(), y.pin_memory()
        if self.device != "cpu":
            x = x.to(self.device, non_blocking=True)
            y = y.to(self.device, non_blocking=True)
        return x, y

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        if self.pool > 0:
            i = 0
            while True:
                if len(self._pool_batches) < self.pool:
                    self._pool_batches.append(self.batch())
                yield self._pool_batches[i % self.pool]
                i += 1
        else:
            while True:
                yield self.batch()


class SyntheticBertBatches:
    """Yields (input_ids, mlm_labels, nsp_labels) for BERT pretraining."""

    def __init__(self, batch_size: int, seq_len: int = 128,
                 vocab_size: int = 30522, mask_frac: float = 0.15,
                 seed: int = 0, rank: int = 0, device: str = "cpu",
                 pinned: bool = False, pool: int = 0):
        self.batch_size, self.seq_len = batch_size, seq_len
        self.vocab_size, self.mask_frac = vocab_size, mask_frac
        self.device, self.pinned = device, pinned
        self.pool = pool            # >0: cycle a fixed batch set (see
        self._pool_batches = []     # SyntheticImages — memorization)
        self.gen = torch.Generator().manual_seed(seed * 100003 + rank)

    def batch(self):
        ids = torch.randint(0, self.vocab_size,
                            (self.batch_size, self.seq_len), generator=self.gen)
        labels = ids.clone()
        mask = torch.rand(ids.shape, generator=self.gen) < self.mask_frac
        labels[~mask] = -100
        nsp = torch.randint(0, 2, (self.batch_size,), generator=self.gen)
        if self.pinned:
            ids, labels, nsp = ids.pin_memory(), labels.pin_memory(), nsp.pin_memory()
        if self.device != "cpu":
            ids = ids.to(self.device, non_blocking=True)
            labels = labels.to(self.device, non_blocking=True)
            nsp = nsp.to(self.device, non_blocking=True)
        return ids, labels, nsp

    def __iter__(self):
        if self.pool > 0:
            i = 0
            while True:
                if len(self._pool_batches) < self.pool:
                    self._pool_batches.append(self.batch())
                yield self._pool_batches[i % self.pool]
                i += 1
        else:
            while True:
                yield self.batch()
