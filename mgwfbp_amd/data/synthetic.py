"""Synthetic datasets — first-class data path (BASELINE runs are
synthetic / random-init).

Shapes match the reference's real pipelines (reference
dl_trainer.py:317-539) so the benchmark measures the same tensor traffic:
imagenet 3x224x224/1000, cifar10 3x32x32/10, mnist 1x28x28/10, PTB token
windows, AN4-style spectrograms with variable lengths.

GPU-resident variants keep a pool of pre-generated batches on-device so
the input pipeline costs nothing in the timed region (the reference's
HDF5 io_time pollutes its comm comparison — SURVEY.md §7.4.6).
"""
import torch
from torch.utils.data import Dataset


class SyntheticImageDataset(Dataset):
    def __init__(self, shape=(3, 224, 224), num_classes=1000, length=51200,
                 seed=0):
        self.shape = tuple(shape)
        self.num_classes = num_classes
        self.length = length
        g = torch.Generator().manual_seed(seed)
        # small pool of distinct images, indexed modulo (keeps memory flat)
        self._pool = torch.randn((64,) + self.shape, generator=g)
        self._labels = torch.randint(0, num_classes, (length,), generator=g)

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        return self._pool[idx % self._pool.size(0)], int(self._labels[idx])


def synthetic_imagenet(length=51200, image_size=224):
    return SyntheticImageDataset((3, image_size, image_size), 1000, length)


def synthetic_cifar10(length=50000):
    return SyntheticImageDataset((3, 32, 32), 10, length)


def synthetic_mnist(length=60000):
    return SyntheticImageDataset((1, 28, 28), 10, length)


class SyntheticPTBDataset(Dataset):
    """Token windows of (num_steps,) input + target (reference
    ptb_reader.py:56-102 shape contract)."""

    def __init__(self, vocab_size=10000, num_steps=35, length=20000,
                 seed=0):
        g = torch.Generator().manual_seed(seed)
        self.tokens = torch.randint(0, vocab_size, (length + num_steps + 1,),
                                    generator=g)
        self.num_steps = num_steps
        self.length = length
        self.vocab_size = vocab_size

    def __len__(self):
        return self.length // self.num_steps

    def __getitem__(self, idx):
        s = idx * self.num_steps
        x = self.tokens[s:s + self.num_steps]
        y = self.tokens[s + 1:s + self.num_steps + 1]
        return x, y


class SyntheticAudioDataset(Dataset):
    """AN4-style (1, freq, T) spectrograms with variable lengths +
    integer transcript targets for CTC."""

    def __init__(self, freq=161, min_t=100, max_t=400, num_classes=29,
                 max_transcript=30, length=948, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.freq = freq
        self.lengths = torch.randint(min_t, max_t + 1, (length,),
                                     generator=g)
        self.tlens = torch.randint(5, max_transcript + 1, (length,),
                                   generator=g)
        self.num_classes = num_classes
        self.length = length
        self.max_t = max_t
        self.seed = seed

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        t = int(self.lengths[idx])
        g = torch.Generator().manual_seed(self.seed + idx)
        spect = torch.randn(1, self.freq, t, generator=g)
        # labels 1..C-1 (0 is CTC blank)
        target = torch.randint(1, self.num_classes, (int(self.tlens[idx]),),
                               generator=g)
        return spect, target


def an4_collate(batch, pad_to=128):
    """Pad a batch of (spect, target) to the max T (reference
    audio_data loader contract: padded batch + per-utterance lengths).

    max_t is rounded up to a multiple of ``pad_to``: MIOpen's fused RNN
    tunes per (T, N) shape, and un-quantized padding gives every batch a
    fresh T — measured ~8 s of RNN re-tuning PER STEP on MI355X
    (benchmarks/an4_probe.py). With a coarse grid only ~4 shapes exist
    and each tunes once. CTC masks padded frames via output_sizes, so
    numerics are unchanged.
    """
    batch = sorted(batch, key=lambda b: b[0].size(2), reverse=True)
    freq = batch[0][0].size(1)
    max_t = batch[0][0].size(2)
    if pad_to > 1:
        max_t = (max_t + pad_to - 1) // pad_to * pad_to
    n = len(batch)
    inputs = torch.zeros(n, 1, freq, max_t)
    input_lengths = torch.zeros(n, dtype=torch.long)
    targets = []
    target_lengths = torch.zeros(n, dtype=torch.long)
    for i, (spect, tgt) in enumerate(batch):
        t = spect.size(2)
        inputs[i, :, :, :t] = spect
        input_lengths[i] = t
        targets.append(tgt)
        target_lengths[i] = tgt.numel()
    return inputs, torch.cat(targets), input_lengths, target_lengths


class GPUBatchPool:
    """Pre-generated on-device batches, cycled round-robin.

    Removes H2D copies and host dataloader jitter from the timed region —
    the right default for weak-scaling throughput benchmarking on a
    288 GB-HBM part where a few dozen resident batches are free.
    """

    def __init__(self, batches):
        self._batches = batches
        self._i = 0

    def next(self):
        b = self._batches[self._i]
        self._i = (self._i + 1) % len(self._batches)
        return b

    @classmethod
    def images(cls, batch_size, shape, num_classes, device, n_batches=8,
               dtype=torch.float32, seed=0, channels_last=False):
        g = torch.Generator(device='cpu').manual_seed(seed)
        batches = []
        for _ in range(n_batches):
            x = torch.randn((batch_size,) + tuple(shape), generator=g)
            y = torch.randint(0, num_classes, (batch_size,), generator=g)
            x = x.to(device=device, dtype=dtype)
            if channels_last and x.dim() == 4:
                x = x.to(memory_format=torch.channels_last)
            batches.append((x, y.to(device)))
        return cls(batches)
