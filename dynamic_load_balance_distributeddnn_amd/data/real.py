"""Real-dataset loading with synthetic fallback.

The reference trains on FashionMNIST / CIFAR / WikiText-2 from disk
(dataloader.py:59-103).  When those files exist locally (prepare_data.py
or rnn_data/) we use them; otherwise the synthetic generators of the same
shapes (the benchmarked configuration — BASELINE.json) are used.
"""

from __future__ import annotations

import logging

import torch

from .corpus import Corpus, corpus_available
from .synthetic import make_cv_dataset, make_lm_tokens

_CV_NORM = {
    "mnist": ((0.1307,), (0.3081,)),
    "cifar10": ((0.4914, 0.4822, 0.4465), (0.2023, 0.1994, 0.2010)),
    "cifar100": ((0.5071, 0.4865, 0.4409), (0.2673, 0.2564, 0.2762)),
}


def load_cv_dataset(name: str, train: bool, seed: int = 1234):
    """torchvision dataset from ./data when present, else synthetic.
    'mnist' maps to FashionMNIST like the reference (dataloader.py:60)."""
    try:
        from torchvision import datasets, transforms

        mean, std = _CV_NORM[name]
        if name == "mnist":
            tfm = transforms.Compose([transforms.ToTensor(),
                                      transforms.Normalize(mean, std)])
            return datasets.FashionMNIST("./data", train=train,
                                         download=False, transform=tfm)
        aug = [transforms.RandomCrop(32, padding=4),
               transforms.RandomHorizontalFlip(),
               transforms.ToTensor(), transforms.Normalize(mean, std)]
        # (the reference applies the train augmentation to the test set
        # too — dataloader.py:78-84; we keep deterministic eval instead)
        tfm = transforms.Compose(aug if train else aug[2:])
        ctor = datasets.CIFAR10 if name == "cifar10" else datasets.CIFAR100
        return ctor("./data", train=train, download=False, transform=tfm)
    except (ImportError, FileNotFoundError, RuntimeError) as e:
        # Only the "dataset not present / torchvision absent" class of
        # failures falls back to synthetic — and loudly, so a corrupt
        # ./data tree can't silently train on random tensors.
        logging.getLogger(__name__).warning(
            "real %s unavailable (%s: %s); using synthetic data",
            name, type(e).__name__, e)
        return make_cv_dataset(name, train, seed)


def load_lm_tokens(train: bool, seed: int = 1234) -> torch.Tensor:
    if corpus_available():
        corpus = Corpus()
        return corpus.train if train else corpus.test
    return make_lm_tokens(train, seed)
