"""Word-level corpus tokenizer (reference Corpus/Dictionary parity,
dataloader.py:120-163).

Used when real WikiText-2 text files exist on disk
(``rnn_data/wikitext-2/{train,valid,test}.txt``); the benchmark configs
use the synthetic stream in synthetic.py (this environment is offline and
the reference's own train split is a stripped blob).
"""

from __future__ import annotations

import os

import torch

__all__ = ["Dictionary", "Corpus", "corpus_available"]

CORPUS_DIR = "rnn_data/wikitext-2"


def corpus_available(path: str = CORPUS_DIR) -> bool:
    return all(os.path.exists(os.path.join(path, f + ".txt"))
               for f in ("train", "valid", "test"))


class Dictionary:
    def __init__(self):
        self.word2idx: dict[str, int] = {}
        self.idx2word: list[str] = []

    def add_word(self, word: str) -> int:
        if word not in self.word2idx:
            self.word2idx[word] = len(self.idx2word)
            self.idx2word.append(word)
        return self.word2idx[word]

    def __len__(self) -> int:
        return len(self.idx2word)


class Corpus:
    """Tokenizes train/valid/test splits; every line is terminated with
    an ``<eos>`` token (the reference's convention)."""

    def __init__(self, path: str = CORPUS_DIR):
        self.dictionary = Dictionary()
        self.train = self.tokenize(os.path.join(path, "train.txt"))
        self.valid = self.tokenize(os.path.join(path, "valid.txt"))
        self.test = self.tokenize(os.path.join(path, "test.txt"))

    def tokenize(self, path: str) -> torch.Tensor:
        assert os.path.exists(path), path
        ids: list[int] = []
        with open(path, encoding="utf8") as f:
            for line in f:
                for word in line.split() + ["<eos>"]:
                    ids.append(self.dictionary.add_word(word))
        return torch.tensor(ids, dtype=torch.int64)
