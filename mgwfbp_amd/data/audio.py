"""Audio (AN4/DeepSpeech) data pipeline.

Native equivalents of the deepspeech.pytorch pieces the reference imports
but does not vendor (reference dl_trainer.py:493-494: SpectrogramDataset,
AudioDataLoader, BucketingSampler, DistributedBucketingSampler) plus the
manifest format of reference audio_data/an4.py:68-84. Spectrogram
extraction uses torch.stft (no librosa/scipy-signal dependency).
"""
import math
import os

import torch
from torch.utils.data import DataLoader, Dataset, Sampler

from .synthetic import an4_collate  # shared padded-batch collate

WINDOWS = {'hamming': torch.hamming_window, 'hann': torch.hann_window,
           'blackman': torch.blackman_window,
           'bartlett': torch.bartlett_window}


class SpectrogramParser:
    def __init__(self, audio_conf=None, normalize=True):
        audio_conf = audio_conf or {}
        self.sample_rate = audio_conf.get('sample_rate', 16000)
        self.window_size = audio_conf.get('window_size', 0.02)
        self.window_stride = audio_conf.get('window_stride', 0.01)
        self.window = audio_conf.get('window', 'hamming')
        self.normalize = normalize

    def parse_audio_tensor(self, samples):
        """samples: 1-D float tensor -> (freq, T) log-magnitude
        spectrogram."""
        n_fft = int(self.sample_rate * self.window_size)
        hop = int(self.sample_rate * self.window_stride)
        win = WINDOWS[self.window](n_fft)
        spect = torch.stft(samples, n_fft=n_fft, hop_length=hop,
                           win_length=n_fft, window=win,
                           return_complex=True, center=True)
        spect = spect.abs().clamp_min(1e-10).log1p()
        if self.normalize:
            spect = (spect - spect.mean()) / (spect.std() + 1e-6)
        return spect

    def parse_audio(self, path):
        samples = load_wav(path)
        return self.parse_audio_tensor(samples)


def load_wav(path):
    """Minimal PCM16 WAV reader (no torchaudio in the image)."""
    import struct
    import wave
    with wave.open(path, 'rb') as w:
        nframes = w.getnframes()
        raw = w.readframes(nframes)
        data = struct.unpack('<%dh' % (len(raw) // 2), raw)
    return torch.tensor(data, dtype=torch.float32) / 32768.0


class SpectrogramDataset(Dataset):
    """Manifest CSV of "wav_path,txt_path" rows -> (spect, target)."""

    def __init__(self, audio_conf, manifest_filepath, labels,
                 normalize=True):
        with open(manifest_filepath) as f:
            ids = [line.strip().split(',') for line in f if line.strip()]
        self.ids = ids
        self.parser = SpectrogramParser(audio_conf, normalize)
        self.labels_map = {c: i for i, c in enumerate(labels)}

    def __len__(self):
        return len(self.ids)

    def __getitem__(self, index):
        wav_path, txt_path = self.ids[index][0], self.ids[index][1]
        spect = self.parser.parse_audio(wav_path).unsqueeze(0)
        with open(txt_path) as f:
            transcript = f.read().strip()
        target = torch.tensor(
            [self.labels_map[c] for c in transcript if c in self.labels_map],
            dtype=torch.long)
        return spect, target


class BucketingSampler(Sampler):
    """Batch utterances of adjacent (duration-sorted) indices so padding
    is minimal."""

    def __init__(self, data_source, batch_size=1):
        super().__init__()
        ids = list(range(len(data_source)))
        self.bins = [ids[i:i + batch_size]
                     for i in range(0, len(ids), batch_size)]

    def __iter__(self):
        return iter(self.bins)

    def __len__(self):
        return len(self.bins)

    def shuffle(self, epoch):
        g = torch.Generator().manual_seed(epoch)
        perm = torch.randperm(len(self.bins), generator=g).tolist()
        self.bins = [self.bins[i] for i in perm]


class DistributedBucketingSampler(Sampler):
    """Rank-sharded bucketing sampler (world-size aware)."""

    def __init__(self, data_source, batch_size=1, num_replicas=1, rank=0):
        super().__init__()
        self.num_replicas = num_replicas
        self.rank = rank
        ids = list(range(len(data_source)))
        bins = [ids[i:i + batch_size]
                for i in range(0, len(ids), batch_size)]
        self.num_samples = int(math.ceil(len(bins) / num_replicas))
        self.total_size = self.num_samples * num_replicas
        bins = bins + bins[:self.total_size - len(bins)]
        self.all_bins = bins
        self.epoch = 0

    def __iter__(self):
        g = torch.Generator().manual_seed(self.epoch)
        perm = torch.randperm(len(self.all_bins), generator=g).tolist()
        shard = perm[self.rank:self.total_size:self.num_replicas]
        return iter([self.all_bins[i] for i in shard])

    def __len__(self):
        return self.num_samples

    def set_epoch(self, epoch):
        self.epoch = epoch

    def shuffle(self, epoch):
        self.set_epoch(epoch)


class AudioDataLoader(DataLoader):
    def __init__(self, dataset, **kwargs):
        super().__init__(dataset, collate_fn=an4_collate, **kwargs)


def wav_duration(path):
    """Duration in seconds from the RIFF header (stdlib wave — the
    reference shells out to sox for this, audio_data/utils.py:24-27)."""
    import wave
    with wave.open(path, 'rb') as w:
        rate = w.getframerate()
        return w.getnframes() / float(rate) if rate else 0.0


def create_manifest(data_path, manifest_path, min_duration=None,
                    max_duration=None):
    """Write "wav,txt" manifest rows sorted by duration, optionally
    dropping clips outside [min_duration, max_duration] seconds
    (reference audio_data/utils.py:11-37, sox-free)."""
    pairs = []
    for root, _, files in os.walk(data_path):
        for fn in files:
            if fn.endswith('.wav'):
                wav = os.path.join(root, fn)
                txt = wav.replace('/wav/', '/txt/').replace('.wav', '.txt')
                if not os.path.exists(txt):
                    continue
                try:
                    dur = wav_duration(wav)
                except Exception:
                    dur = os.path.getsize(wav) / 32000.0
                if min_duration is not None and dur < min_duration:
                    continue
                if max_duration is not None and dur > max_duration:
                    continue
                pairs.append((dur, wav, txt))
    pairs.sort()
    with open(manifest_path, 'w') as f:
        for _, wav, txt in pairs:
            f.write('%s,%s\n' % (wav, txt))
