"""PTB word-level LSTM language model (reference models/lstm.py:5-47).

2x1500 LSTM + tied-size embedding, truncated BPTT via
``repackage_hidden``.
"""
import torch
import torch.nn as nn


class lstm(nn.Module):
    def __init__(self, vocab_size=10000, embedding_dim=1500, hidden_dim=1500,
                 num_layers=2, num_steps=35, batch_size=20, dropout=0.65):
        super().__init__()
        self.ntokens = vocab_size
        self.num_layers = num_layers
        self.hidden_dim = hidden_dim
        self.num_steps = num_steps
        self.batch_size = batch_size
        self.drop = nn.Dropout(dropout)
        self.embedding = nn.Embedding(vocab_size, embedding_dim)
        self.rnn = nn.LSTM(embedding_dim, hidden_dim, num_layers,
                           dropout=dropout, batch_first=True)
        self.fc = nn.Linear(hidden_dim, vocab_size)
        self.init_weights()

    def init_weights(self):
        rng = 0.1
        nn.init.uniform_(self.embedding.weight, -rng, rng)
        nn.init.uniform_(self.fc.weight, -rng, rng)
        nn.init.zeros_(self.fc.bias)

    def init_hidden(self):
        w = next(self.parameters())
        return (w.new_zeros(self.num_layers, self.batch_size,
                            self.hidden_dim),
                w.new_zeros(self.num_layers, self.batch_size,
                            self.hidden_dim))

    def forward(self, x, hidden):
        emb = self.drop(self.embedding(x))
        out, hidden = self.rnn(emb, hidden)
        out = self.drop(out)
        return self.fc(out), hidden


def repackage_hidden(h):
    """Detach hidden state from the graph (truncated BPTT; reference
    models/lstm.py:42-47)."""
    if isinstance(h, torch.Tensor):
        return h.detach()
    return tuple(repackage_hidden(v) for v in h)
