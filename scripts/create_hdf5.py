"""Convert an ImageNet-style folder tree into the single-HDF5 layout
DatasetHDF5 reads (mgwfbp_amd/data/hdf5.py).

Reference parity: /root/reference/scripts/create_hdf5.py:46-107 —
same output schema ({train,val}_img uint8 NHWC + {train,val}_labels
int16 + imagenet_label_mapping.csv), rebuilt without cv2: PIL decodes
and resizes (bicubic, RGB), and the write loop streams in chunks
instead of one image per dataset assignment.

Usage:
    python scripts/create_hdf5.py --datadir /path/imagenet \
        --output imagenet-shuffled.hdf5 [--size 320]

Expects datadir/{train,val}/<class>/<img>.JPEG. Requires h5py + PIL
(checked at startup; this container intentionally ships neither a
network nor datasets — the framework's first-class path is synthetic
data, BASELINE.json).
"""
import argparse
import csv
import os
import sys
import time


def list_images(datadir, folder):
    files = []
    root = os.path.join(datadir, folder)
    for cls in sorted(os.listdir(root)):
        d = os.path.join(root, cls)
        if not os.path.isdir(d):
            continue
        for f in sorted(os.listdir(d)):
            files.append((os.path.join(d, f), cls))
    return files


def class_map(train_files, val_files):
    classes = sorted({cls for _, cls in train_files}
                     | {cls for _, cls in val_files})
    return {cls: i for i, cls in enumerate(classes)}


def convert(datadir, outputpath, output, size):
    try:
        import h5py
        import numpy as np
        from PIL import Image
    except ImportError as e:
        sys.exit('create_hdf5 needs h5py + numpy + PIL: %s' % e)

    train_files = list_images(datadir, 'train')
    val_files = list_images(datadir, 'val')
    print('train images: %d, val images: %d'
          % (len(train_files), len(val_files)))
    labels = class_map(train_files, val_files)
    os.makedirs(outputpath, exist_ok=True)
    with open(os.path.join(outputpath, 'imagenet_label_mapping.csv'),
              'w') as csvfile:
        writer = csv.writer(csvfile, delimiter=' ')
        for cls in sorted(labels):
            writer.writerow([cls, str(labels[cls])])

    def decode(path):
        img = Image.open(path).convert('RGB')
        img = img.resize((size, size), Image.BICUBIC)
        return np.asarray(img, dtype=np.uint8)

    h5file = os.path.join(outputpath, output)
    with h5py.File(h5file, 'w') as hf:
        for split, files in (('train', train_files), ('val', val_files)):
            n = len(files)
            dimg = hf.create_dataset('%s_img' % split,
                                     (n, size, size, 3), np.uint8)
            dlab = hf.create_dataset('%s_labels' % split, (n,), np.int16)
            dlab[...] = [labels[cls] for _, cls in files]
            s = time.time()
            for i, (path, _) in enumerate(files):
                dimg[i, ...] = decode(path)
                if i % 1000 == 0 and i > 0:
                    print('%s: %d/%d (%.1fs)' % (split, i, n,
                                                 time.time() - s))
                    s = time.time()
    print('wrote %s' % h5file)
    return h5file


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--datadir', required=True,
                    help='folder with train/ and val/ class subfolders')
    ap.add_argument('--outputpath', default='.')
    ap.add_argument('--output', default='imagenet-shuffled.hdf5')
    ap.add_argument('--size', type=int, default=320)
    args = ap.parse_args()
    convert(args.datadir, args.outputpath, args.output, args.size)


if __name__ == '__main__':
    main()
