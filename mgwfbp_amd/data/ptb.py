"""Penn Treebank reader (reference ptb_reader.py:14-102)."""
import collections
import os

import numpy as np
import torch
from torch.utils.data import Dataset


def _read_words(filename):
    with open(filename, 'r') as f:
        return f.read().replace('\n', ' <eos> ').split()


def build_vocab(filename):
    data = _read_words(filename)
    counter = collections.Counter(data)
    pairs = sorted(counter.items(), key=lambda x: (-x[1], x[0]))
    words = [w for w, _ in pairs]
    return dict(zip(words, range(len(words))))


def _file_to_ids(filename, word_to_id):
    data = _read_words(filename)
    return [word_to_id[w] for w in data if w in word_to_id]


def ptb_raw_data(data_path):
    """Load PTB train/valid/test token-id streams + vocab size
    (reference ptb_reader.py:32-54)."""
    train_path = os.path.join(data_path, 'ptb.train.txt')
    valid_path = os.path.join(data_path, 'ptb.valid.txt')
    test_path = os.path.join(data_path, 'ptb.test.txt')
    word_to_id = build_vocab(train_path)
    train = _file_to_ids(train_path, word_to_id)
    valid = _file_to_ids(valid_path, word_to_id)
    test = _file_to_ids(test_path, word_to_id)
    return train, valid, test, len(word_to_id)


class PTBDataset(Dataset):
    """Sliding windows of num_steps tokens (reference
    ptb_reader.py:56-102)."""

    def __init__(self, raw_data, batch_size=20, num_steps=35):
        self.data = np.asarray(raw_data, dtype=np.int64)
        self.num_steps = num_steps
        self.batch_size = batch_size

    def __len__(self):
        return (len(self.data) - 1) // self.num_steps

    def __getitem__(self, idx):
        s = idx * self.num_steps
        x = torch.from_numpy(self.data[s:s + self.num_steps].copy())
        y = torch.from_numpy(self.data[s + 1:s + self.num_steps + 1].copy())
        return x, y
