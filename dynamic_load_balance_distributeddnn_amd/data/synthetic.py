"""Synthetic datasets shaped like the reference's real ones.

The reference trains on FashionMNIST / CIFAR-10 / CIFAR-100 / WikiText-2
(dataloader.py:59-103).  This environment has no network (and the
reference's own wikitext-2 train split is a stripped blob), so we generate
deterministic random data of the same shapes — BASELINE.json mandates
synthetic data / random-init weights for every measured config.

Memory design: images are served from a small pool of pre-generated
tensors (index i -> pool[i % POOL]); the dataset advertises the full
nominal length so epoch/step accounting matches the real dataset, while
holding ~100 MB instead of ~600 MB per process.  Throughput and loss-shape
are unaffected; content never repeats within a batch for pools >= batch.
"""

from __future__ import annotations

import os

import torch
from torch.utils.data import Dataset

# name -> (C, H, W, num_classes, train_len, test_len)
DATASET_SPECS = {
    "mnist": (1, 28, 28, 10, 60_000, 10_000),
    "cifar10": (3, 32, 32, 10, 50_000, 10_000),
    "cifar100": (3, 32, 32, 100, 50_000, 10_000),
}

# WikiText-2 word-level sizes (ntokens hardcoded at reference dbs.py:337).
WIKITEXT2_VOCAB = 33_278
WIKITEXT2_TRAIN_TOKENS = 2_088_628
WIKITEXT2_TEST_TOKENS = 245_569

_POOL = 4096


def _nominal_len(default: int) -> int:
    """Optional shrink for fast CI runs via DLB_SYNTH_SCALE (0 < s <= 1)."""
    scale = float(os.environ.get("DLB_SYNTH_SCALE", "1"))
    return max(1, int(default * scale))


class SyntheticImages(Dataset):
    """Normalized-image-shaped random data with balanced labels."""

    def __init__(self, name: str, train: bool, seed: int = 1234):
        c, h, w, ncls, ntrain, ntest = DATASET_SPECS[name]
        self.num_classes = ncls
        self._len = _nominal_len(ntrain if train else ntest)
        g = torch.Generator().manual_seed(seed + (0 if train else 1))
        pool = min(_POOL, self._len)
        self.images = torch.randn(pool, c, h, w, generator=g)
        self.labels = torch.randint(0, ncls, (pool,), generator=g)

    def __len__(self) -> int:
        return self._len

    def __getitem__(self, i):
        j = i % self.images.shape[0]
        return self.images[j], self.labels[j]


def make_cv_dataset(name: str, train: bool, seed: int = 1234) -> SyntheticImages:
    return SyntheticImages(name, train, seed)


def make_lm_tokens(train: bool, seed: int = 1234) -> torch.Tensor:
    """WikiText-2-shaped int64 token stream (Zipf-ish over the real vocab)."""
    n = _nominal_len(WIKITEXT2_TRAIN_TOKENS if train else WIKITEXT2_TEST_TOKENS)
    g = torch.Generator().manual_seed(seed + (10 if train else 11))
    # Zipf-like skew: word frequency in real text is heavy-headed; an
    # exponential transform of uniforms gives a cheap approximation.
    u = torch.rand(n, generator=g)
    ids = (WIKITEXT2_VOCAB * u.pow(3.0)).long().clamp_(0, WIKITEXT2_VOCAB - 1)
    return ids
