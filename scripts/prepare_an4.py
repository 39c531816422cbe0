"""Prepare the AN4 speech dataset from a locally provided archive.

Reference parity: /root/reference/audio_data/an4.py:30-84 — but
MI355X-native in two ways: no network fetch (this pool has no egress;
point --archive at an an4_raw.bigendian.tar.gz you already have, or
--rawdir at an extracted tree) and no sox dependency: the raw files
are 16-bit BIG-ENDIAN mono PCM at 16 kHz, converted with numpy
byteswap + the stdlib wave module.

Output layout (what mgwfbp_amd/data/audio.py loads):
    <target>/{train,val}/an4/wav/<id>.wav
    <target>/{train,val}/an4/txt/<id>.txt
    an4_train_manifest.csv, an4_val_manifest.csv   (wav_path,txt_path)

Usage:
    python scripts/prepare_an4.py --archive an4_raw.bigendian.tar.gz
    python scripts/prepare_an4.py --rawdir an4/ --target-dir an4_dataset
"""
import argparse
import os
import sys
import tarfile
import wave

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from mgwfbp_amd.data.audio import create_manifest  # noqa: E402


def raw_to_wav(raw_path, wav_path, rate=16000):
    """16-bit big-endian mono PCM -> little-endian RIFF wav."""
    data = np.fromfile(raw_path, dtype='>i2').astype('<i2')
    with wave.open(wav_path, 'wb') as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(rate)
        w.writeframes(data.tobytes())


def extract_transcript(line):
    # reference an4.py:63-65: strip the utterance-id suffix and <s> tags
    return (line.split('(')[0].strip('<s>').split('<')[0]
            .strip().upper())


def format_split(rawdir, target, split, wav_folder):
    """Walk the AN4 fileids/transcripts pair for one split, writing
    wav + txt pairs under <target>/<split>/an4/."""
    etc = os.path.join(rawdir, 'etc')
    tag = 'train' if split == 'train' else 'test'
    fileids = os.path.join(etc, 'an4_%s.fileids' % tag)
    transcripts = os.path.join(etc, 'an4_%s.transcription' % tag)
    wav_root = os.path.join(rawdir, 'wav', wav_folder)
    out_wav = os.path.join(target, split, 'an4', 'wav')
    out_txt = os.path.join(target, split, 'an4', 'txt')
    os.makedirs(out_wav, exist_ok=True)
    os.makedirs(out_txt, exist_ok=True)
    with open(fileids) as f:
        ids = [ln.strip() for ln in f if ln.strip()]
    with open(transcripts) as t:
        lines = [ln for ln in t]
    n = 0
    for i, utt in enumerate(ids):
        raw = os.path.join(rawdir, 'wav', utt + '.raw')
        if not os.path.exists(raw):
            raw = os.path.join(wav_root, os.path.basename(utt) + '.raw')
        if not os.path.exists(raw):
            continue
        base = os.path.basename(utt)
        raw_to_wav(raw, os.path.join(out_wav, base + '.wav'))
        with open(os.path.join(out_txt, base + '.txt'), 'w') as fo:
            fo.write(extract_transcript(lines[i]))
        n += 1
    return n


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--archive', default=None,
                    help='an4_raw.bigendian.tar.gz (local file)')
    ap.add_argument('--rawdir', default=None,
                    help='already-extracted an4/ tree')
    ap.add_argument('--target-dir', default='an4_dataset')
    ap.add_argument('--min-duration', type=float, default=1.0)
    ap.add_argument('--max-duration', type=float, default=15.0)
    args = ap.parse_args()
    rawdir = args.rawdir
    if args.archive:
        with tarfile.open(args.archive) as tar:
            tar.extractall()
        rawdir = 'an4'
    if not rawdir or not os.path.isdir(rawdir):
        sys.exit('need --archive or --rawdir (no network on this pool: '
                 'fetch an4_raw.bigendian.tar.gz elsewhere)')
    nt = format_split(rawdir, args.target_dir, 'train', 'an4_clstk')
    nv = format_split(rawdir, args.target_dir, 'val', 'an4test_clstk')
    print('converted %d train / %d val utterances' % (nt, nv))
    create_manifest(os.path.join(args.target_dir, 'train'),
                    'an4_train_manifest.csv', args.min_duration,
                    args.max_duration)
    create_manifest(os.path.join(args.target_dir, 'val'),
                    'an4_val_manifest.csv')


if __name__ == '__main__':
    main()
