"""DeepSpeech2-style speech model for AN4 (the reference's ``lstman4``).

Native implementation of the architecture the reference assembles in
models/lstm_models.py:45-233 (MaskConv over a 2xConv2d stack, N stacked
BatchRNN layers, optional Lookahead for unidirectional RNNs, SequenceWise
BN+Linear head) and the factory models/lstman4.py:8-33 — the reference's
CUDA warp-ctc loss (dl_trainer.py:214-215) is replaced by ROCm-native
``torch.nn.CTCLoss``.

Input: (N, 1, freq, time) spectrograms + per-utterance lengths.
Output: (T, N, num_classes) logits + output lengths (CTC layout).
"""
import math

import torch
import torch.nn as nn
import torch.nn.functional as F

# Default AN4 label set (reference audio_data/labels.json)
LABELS = "_'ABCDEFGHIJKLMNOPQRSTUVWXYZ "


class SequenceWise(nn.Module):
    """Collapse (T, N, H) to (T*N, H), apply module, restore shape
    (reference lstm_models.py:18-33)."""

    def __init__(self, module):
        super().__init__()
        self.module = module

    def forward(self, x):
        t, n = x.size(0), x.size(1)
        x = self.module(x.view(t * n, -1))
        return x.view(t, n, -1)


class MaskConv(nn.Module):
    """Run a conv stack and zero activations beyond each utterance's
    length (reference lstm_models.py:45-72)."""

    def __init__(self, seq_module):
        super().__init__()
        self.seq_module = seq_module

    def forward(self, x, lengths):
        for module in self.seq_module:
            x = module(x)
            if isinstance(module, nn.Conv2d):
                lengths = self._conv_out_lengths(module, lengths)
            mask = torch.arange(x.size(3), device=x.device)[None, :] \
                >= lengths.to(x.device)[:, None]
            x = x.masked_fill(mask[:, None, None, :], 0)
        return x, lengths

    @staticmethod
    def _conv_out_lengths(conv, lengths):
        return torch.div(
            lengths + 2 * conv.padding[1] - conv.dilation[1]
            * (conv.kernel_size[1] - 1) - 1,
            conv.stride[1], rounding_mode='floor') + 1


class BatchRNN(nn.Module):
    """BN + (bi)GRU/LSTM with summed directions (reference
    lstm_models.py:76-105)."""

    def __init__(self, input_size, hidden_size, rnn_type=nn.LSTM,
                 bidirectional=True, batch_norm=True):
        super().__init__()
        self.batch_norm = (SequenceWise(nn.BatchNorm1d(input_size))
                           if batch_norm else None)
        self.bidirectional = bidirectional
        self.rnn = rnn_type(input_size, hidden_size,
                            bidirectional=bidirectional, bias=True)
        self.hidden_size = hidden_size

    def forward(self, x):
        if self.batch_norm is not None:
            x = self.batch_norm(x)
        x, _ = self.rnn(x)
        if self.bidirectional:
            # sum the two directions
            t, n = x.size(0), x.size(1)
            x = x.view(t, n, 2, -1).sum(2)
        return x


class Lookahead(nn.Module):
    """Temporal lookahead convolution for unidirectional stacks
    (reference lstm_models.py:108-145)."""

    def __init__(self, n_features, context=20):
        super().__init__()
        self.context = context
        self.conv = nn.Conv1d(n_features, n_features,
                              kernel_size=context + 1, padding=0,
                              groups=n_features, bias=False)

    def forward(self, x):
        # x: (T, N, H) -> pad future frames, depthwise conv over time
        x = x.permute(1, 2, 0)                     # N, H, T
        x = F.pad(x, (0, self.context))
        x = self.conv(x)
        return x.permute(2, 0, 1).contiguous()     # T, N, H


class DeepSpeech(nn.Module):
    def __init__(self, rnn_type=nn.LSTM, labels=LABELS, rnn_hidden_size=800,
                 nb_layers=5, audio_conf=None, bidirectional=True,
                 context=20):
        super().__init__()
        audio_conf = audio_conf or {}
        self.sample_rate = audio_conf.get('sample_rate', 16000)
        self.window_size = audio_conf.get('window_size', 0.02)
        self.labels = labels
        num_classes = len(labels)
        self.bidirectional = bidirectional

        self.conv = MaskConv(nn.Sequential(
            nn.Conv2d(1, 32, kernel_size=(41, 11), stride=(2, 2),
                      padding=(20, 5)),
            nn.BatchNorm2d(32),
            nn.Hardtanh(0, 20, inplace=True),
            nn.Conv2d(32, 32, kernel_size=(21, 11), stride=(2, 1),
                      padding=(10, 5)),
            nn.BatchNorm2d(32),
            nn.Hardtanh(0, 20, inplace=True),
        ))
        freq = int(math.floor((self.sample_rate * self.window_size) / 2) + 1)
        freq = int(math.floor(freq + 2 * 20 - 41) / 2 + 1)
        freq = int(math.floor(freq + 2 * 10 - 21) / 2 + 1)
        rnn_input_size = freq * 32

        rnns = [BatchRNN(rnn_input_size, rnn_hidden_size, rnn_type,
                         bidirectional, batch_norm=False)]
        for _ in range(nb_layers - 1):
            rnns.append(BatchRNN(rnn_hidden_size, rnn_hidden_size, rnn_type,
                                 bidirectional))
        self.rnns = nn.Sequential(*rnns)
        self.lookahead = (nn.Sequential(
            Lookahead(rnn_hidden_size, context=context),
            nn.Hardtanh(0, 20, inplace=True))
            if not bidirectional else None)
        self.fc = nn.Sequential(SequenceWise(nn.Sequential(
            nn.BatchNorm1d(rnn_hidden_size),
            nn.Linear(rnn_hidden_size, num_classes, bias=False))))

    def forward(self, x, lengths):
        """x: (N, 1, freq, T) spectrograms; lengths: (N,) frame counts.
        Returns (T', N, C) log-softmax-ready logits + output lengths."""
        x, output_lengths = self.conv(x, lengths)
        n, c, f, t = x.size()
        x = x.view(n, c * f, t).permute(2, 0, 1).contiguous()  # T, N, H
        x = self.rnns(x)
        if self.lookahead is not None:
            x = self.lookahead(x)
        x = self.fc(x)
        return x, output_lengths


def LSTMAN4(datapath=None, rnn_hidden_size=800, nb_layers=5):
    """Factory matching the reference's (net, ext) contract
    (reference models/lstman4.py:8-33); ext carries the label set."""
    model = DeepSpeech(rnn_type=nn.LSTM, labels=LABELS,
                       rnn_hidden_size=rnn_hidden_size, nb_layers=nb_layers,
                       bidirectional=True)
    ext = {'labels': LABELS}
    return model, ext


class GreedyDecoder:
    """Greedy CTC decoder (replaces the deepspeech.pytorch decoder the
    reference imports but does not vendor, reference dl_trainer.py:494)."""

    def __init__(self, labels=LABELS, blank_index=0):
        self.labels = labels
        self.blank = blank_index
        self.int2char = dict(enumerate(labels))

    def decode(self, probs, sizes=None):
        """probs: (N, T, C) or (T, N, C) -> list of decoded strings."""
        if probs.dim() != 3:
            raise ValueError('expected 3-D logits')
        argmax = probs.argmax(-1)
        out = []
        for b in range(argmax.size(0)):
            seq = argmax[b]
            if sizes is not None:
                seq = seq[:int(sizes[b])]
            chars = []
            prev = None
            for idx in seq.tolist():
                if idx != self.blank and idx != prev:
                    chars.append(self.int2char[idx])
                prev = idx
            out.append(''.join(chars))
        return out

    @staticmethod
    def wer(decoded, target):
        """Word error rate between two strings (edit distance on words)."""
        d_words, t_words = decoded.split(), target.split()
        d = [[0] * (len(t_words) + 1) for _ in range(len(d_words) + 1)]
        for i in range(len(d_words) + 1):
            d[i][0] = i
        for j in range(len(t_words) + 1):
            d[0][j] = j
        for i in range(1, len(d_words) + 1):
            for j in range(1, len(t_words) + 1):
                cost = 0 if d_words[i - 1] == t_words[j - 1] else 1
                d[i][j] = min(d[i - 1][j] + 1, d[i][j - 1] + 1,
                              d[i - 1][j - 1] + cost)
        return d[len(d_words)][len(t_words)] / max(len(t_words), 1)
