"""Dataset preparation tools: AN4 raw->wav formatter + HDF5 converter
(VERDICT r01 missing items 3-4; reference scripts/create_hdf5.py and
audio_data/an4.py)."""
import os
import sys
import wave

import numpy as np
import pytest

SCRIPTS = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), 'scripts')
sys.path.insert(0, SCRIPTS)


def _make_an4_tree(root, n_train=3, n_val=2, rate=16000):
    """Synthesize a mini an4 raw layout (big-endian PCM + etc files)."""
    os.makedirs(os.path.join(root, 'etc'), exist_ok=True)
    sents = ['HELLO WORLD', 'YES', 'ENTER FIFTY', 'NO', 'RUBOUT G M E']
    for split, n, folder in (('train', n_train, 'an4_clstk'),
                             ('test', n_val, 'an4test_clstk')):
        ids, trans = [], []
        wav_dir = os.path.join(root, 'wav', folder, 'spk')
        os.makedirs(wav_dir, exist_ok=True)
        for i in range(n):
            utt = '%s/spk/%s-%d' % (folder, split, i)
            samples = (np.sin(np.arange(rate) * 0.01 * (i + 1))
                       * 3000).astype('>i2')
            samples.tofile(os.path.join(root, 'wav', utt + '.raw'))
            ids.append(utt)
            trans.append('<s> %s </s> (%s-%d)' % (sents[i % len(sents)],
                                                  split, i))
        with open(os.path.join(root, 'etc', 'an4_%s.fileids' % split),
                  'w') as f:
            f.write('\n'.join(ids) + '\n')
        with open(os.path.join(root, 'etc',
                               'an4_%s.transcription' % split),
                  'w') as f:
            f.write('\n'.join(trans) + '\n')


def test_prepare_an4_roundtrip(tmp_path, monkeypatch):
    import prepare_an4
    root = str(tmp_path / 'an4')
    target = str(tmp_path / 'out')
    _make_an4_tree(root)
    monkeypatch.chdir(tmp_path)
    nt = prepare_an4.format_split(root, target, 'train', 'an4_clstk')
    nv = prepare_an4.format_split(root, target, 'val', 'an4test_clstk')
    assert nt == 3 and nv == 2
    # wavs are little-endian RIFF mono 16k and byte-identical samples
    wavs = []
    for r, _, fs in os.walk(os.path.join(target, 'train')):
        wavs += [os.path.join(r, f) for f in fs if f.endswith('.wav')]
    assert len(wavs) == 3
    with wave.open(wavs[0], 'rb') as w:
        assert w.getframerate() == 16000
        assert w.getnchannels() == 1
        assert w.getnframes() == 16000
    # transcript extraction strips tags and utterance ids
    txts = [p.replace('/wav/', '/txt/').replace('.wav', '.txt')
            for p in wavs]
    body = open(txts[0]).read()
    assert '<s>' not in body and '(' not in body and body == body.upper()
    # manifest generation with duration bounds
    from mgwfbp_amd.data.audio import create_manifest
    man = str(tmp_path / 'm.csv')
    create_manifest(os.path.join(target, 'train'), man,
                    min_duration=0.5, max_duration=15.0)
    rows = open(man).read().strip().split('\n')
    assert len(rows) == 3
    for row in rows:
        wav, txt = row.split(',')
        assert os.path.exists(wav) and os.path.exists(txt)


def test_raw_to_wav_byteswap(tmp_path):
    import prepare_an4
    raw = str(tmp_path / 'x.raw')
    wav = str(tmp_path / 'x.wav')
    samples = np.array([1, -2, 30000, -30000, 255], dtype='>i2')
    samples.tofile(raw)
    prepare_an4.raw_to_wav(raw, wav)
    with wave.open(wav, 'rb') as w:
        got = np.frombuffer(w.readframes(5), dtype='<i2')
    assert (got == samples.astype(np.int16)).all()


def test_create_hdf5_roundtrip(tmp_path):
    h5py = pytest.importorskip('h5py')
    from PIL import Image
    import create_hdf5
    datadir = tmp_path / 'imagenet'
    for split, classes in (('train', ('cat', 'dog')), ('val', ('cat',))):
        for cls in classes:
            d = datadir / split / cls
            d.mkdir(parents=True)
            for i in range(2):
                Image.new('RGB', (40 + i, 30), (i * 40, 100, 200)).save(
                    d / ('%s%d.JPEG' % (cls, i)))
    out = create_hdf5.convert(str(datadir), str(tmp_path), 'x.hdf5', 32)
    with h5py.File(out, 'r') as hf:
        assert hf['train_img'].shape == (4, 32, 32, 3)
        assert hf['val_img'].shape == (2, 32, 32, 3)
        assert list(hf['train_labels'][...]) == [0, 0, 1, 1]
    assert (tmp_path / 'imagenet_label_mapping.csv').exists()
