"""Data-pipeline tests: audio spectrogram/manifest path, PTB reader,
HDF5 gating, evaluate end-to-end."""
import math
import os
import struct
import wave

import pytest
import torch

from mgwfbp_amd import data as D
from mgwfbp_amd import evaluate as ev
from mgwfbp_amd.dl_trainer import DLTrainer


def _write_wav(path, n=16000, freq=440):
    with wave.open(str(path), 'wb') as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(16000)
        samples = [int(10000 * math.sin(2 * math.pi * freq * i / 16000))
                   for i in range(n)]
        w.writeframes(struct.pack('<%dh' % n, *samples))


class TestAudio:
    def test_wav_spectrogram_manifest(self, tmp_path):
        wavdir = tmp_path / 'wav'
        txtdir = tmp_path / 'txt'
        wavdir.mkdir()
        txtdir.mkdir()
        for i in range(3):
            _write_wav(wavdir / ('u%d.wav' % i), n=8000 + 4000 * i)
            (txtdir / ('u%d.txt' % i)).write_text('HELLO WORLD')
        manifest = tmp_path / 'manifest.csv'
        D.create_manifest(str(tmp_path), str(manifest))
        lines = manifest.read_text().strip().splitlines()
        assert len(lines) == 3
        from mgwfbp_amd.models.deepspeech import LABELS
        ds = D.SpectrogramDataset({}, str(manifest), LABELS)
        spect, target = ds[0]
        assert spect.dim() == 3 and spect.size(1) == 161
        assert target.numel() == len('HELLO WORLD')
        # bucketing sampler over it
        sampler = D.BucketingSampler(ds, batch_size=2)
        bins = list(iter(sampler))
        assert sum(len(b) for b in bins) == 3
        loader = D.AudioDataLoader(ds, batch_sampler=sampler)
        x, tgt, xlen, tlen = next(iter(loader))
        assert x.size(1) == 1 and x.size(2) == 161

    def test_distributed_bucketing_shards(self):
        ds = D.SyntheticAudioDataset(length=20)
        s0 = D.DistributedBucketingSampler(ds, batch_size=2,
                                           num_replicas=2, rank=0)
        s1 = D.DistributedBucketingSampler(ds, batch_size=2,
                                           num_replicas=2, rank=1)
        b0 = list(iter(s0))
        b1 = list(iter(s1))
        assert len(b0) == len(b1) == 5


class TestPTB:
    def test_reader_roundtrip(self, tmp_path):
        text = 'the cat sat on the mat\nthe dog sat too\n'
        for split in ('train', 'valid', 'test'):
            (tmp_path / ('ptb.%s.txt' % split)).write_text(text)
        train, valid, test, vocab = D.ptb_raw_data(str(tmp_path))
        assert vocab == len(set(text.replace('\n', ' <eos> ').split()))
        ds = D.PTBDataset(train, batch_size=2, num_steps=3)
        x, y = ds[0]
        assert x.shape == (3,) and y.shape == (3,)
        # y is x shifted by one
        assert train[1] == int(y[0])


class TestHDF5Gate:
    def test_raises_without_h5py(self):
        from mgwfbp_amd.data.hdf5 import HAS_H5PY, DatasetHDF5
        if HAS_H5PY:
            pytest.skip('h5py installed here')
        with pytest.raises(RuntimeError):
            DatasetHDF5('/nonexistent.h5')


class TestEvaluateEndToEnd:
    def test_epoch_walk(self, tmp_path):
        t = DLTrainer(0, 1, dist=False, batch_size=4, ngpus=0,
                      data_dir='', dataset='cifar10', dnn='lenet', lr=0.1,
                      nworkers=1, prefix='e2e', synthetic=True)
        rundir = tmp_path / 'lenet-n1-bs4-lr0.1000'
        rundir.mkdir()
        for epoch in (0, 1):
            t.train_epoch = epoch
            t.save_checkpoint(filename=str(
                rundir / ('lenet-rank0-epoch%d.pth' % epoch)))
        best, best_epoch = ev.evaluate(str(rundir), dataset='cifar10',
                                       start_epoch=0, nepochs=1)
        assert 0.0 <= best <= 1.0
        assert best_epoch in (0, 1)
