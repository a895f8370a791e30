"""Data layer: exact-sum shards, common step counts, LM batchify."""

import numpy as np
import torch

from dynamic_load_balance_distributeddnn_amd.data import (
    batchify, bptt_batch, make_cv_dataset, make_lm_tokens, partition_cv,
    partition_lm)


def test_cv_partition_common_steps_and_sizes():
    ds = make_cv_dataset("cifar10", train=True)
    batches = np.array([10, 20, 14, 20])  # B = 64, deliberately uneven
    loaders = []
    steps_seen = set()
    for rank in range(4):
        loader, steps = partition_cv(ds, batches, rank, seed=1, epoch=0)
        loaders.append(loader)
        steps_seen.add(steps)
        assert len(loader) == steps  # drop_last exactness
    assert len(steps_seen) == 1
    # each loader yields its own batch size every iteration
    for rank, loader in enumerate(loaders):
        x, y = next(iter(loader))
        assert x.shape[0] == batches[rank]


def test_cv_partition_disjoint_cover():
    ds = make_cv_dataset("cifar10", train=True)
    batches = np.array([3, 5])
    idx_all = []
    for rank in range(2):
        loader, steps = partition_cv(ds, batches, rank, seed=7, epoch=3)
        idx_all.extend(loader.dataset.indices)
    assert len(idx_all) == len(set(idx_all))  # disjoint
    assert len(idx_all) == (len(ds) // 8) * 8  # full cover up to remainder


def test_cv_partition_epoch_reshuffles():
    ds = make_cv_dataset("cifar10", train=True)
    batches = np.array([4, 4])
    l0, _ = partition_cv(ds, batches, 0, seed=7, epoch=0)
    l1, _ = partition_cv(ds, batches, 0, seed=7, epoch=1)
    assert l0.dataset.indices != l1.dataset.indices


def test_batchify_layout():
    t = torch.arange(12)
    sheet = batchify(t, 3)  # rows = 4
    assert sheet.shape == (4, 3)
    # column-major semantics: column j is the j-th contiguous chunk
    assert sheet[:, 0].tolist() == [0, 1, 2, 3]
    assert sheet[:, 1].tolist() == [4, 5, 6, 7]


def test_bptt_batch_shapes():
    sheet = torch.arange(200).view(40, 5)
    data, target = bptt_batch(sheet, 0, 35)
    assert data.shape == (35, 5)
    assert target.shape == (35 * 5,)
    # target is data shifted by one row
    assert target[:5].tolist() == sheet[1].tolist()
    # tail window clips
    data, target = bptt_batch(sheet, 35, 35)
    assert data.shape[0] == 4


def test_lm_partition_common_steps():
    tokens = make_lm_tokens(train=True)
    batches = np.array([7, 13, 12])  # B = 32
    step_counts = set()
    for rank in range(3):
        sheet, steps = partition_lm(tokens, batches, rank)
        assert sheet.shape[1] == batches[rank]
        step_counts.add(steps)
        # every rank has the same number of rows
        assert sheet.shape[0] == tokens.numel() // 32
    assert len(step_counts) == 1


def test_lm_tokens_in_vocab():
    tokens = make_lm_tokens(train=True)
    assert tokens.min() >= 0
    assert tokens.max() < 33278


def test_corpus_tokenize_real_files(tmp_path):
    """End-to-end Corpus over real text files (the path used when
    rnn_data/wikitext-2 is present — reference dataloader.py:120-163)."""
    from dynamic_load_balance_distributeddnn_amd.data.corpus import (
        Corpus, corpus_available)

    d = tmp_path / "wikitext-2"
    d.mkdir()
    (d / "train.txt").write_text("the cat sat\nthe dog ran\n")
    (d / "valid.txt").write_text("the cat ran\n")
    (d / "test.txt").write_text("a dog\n")
    assert corpus_available(str(d))
    c = Corpus(str(d))
    # every line ends with <eos>; words are interned in first-seen order
    assert c.train.tolist()[:4] == [0, 1, 2, 3]          # the cat sat <eos>
    assert c.train[-1].item() == c.dictionary.word2idx["<eos>"]
    assert len(c.dictionary) == 7  # the cat sat <eos> dog ran a
    # valid/test reuse the shared dictionary (reference behavior)
    assert c.valid[0].item() == c.dictionary.word2idx["the"]
    assert c.test[0].item() == c.dictionary.word2idx["a"]


def test_corpus_available_false_when_missing(tmp_path):
    from dynamic_load_balance_distributeddnn_amd.data.corpus import \
        corpus_available

    assert not corpus_available(str(tmp_path / "nope"))
