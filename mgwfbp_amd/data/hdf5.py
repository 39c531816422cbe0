"""HDF5 ImageNet dataset (reference datasets.py:8-36).

Gated on h5py availability (not installed in the MI355X image; synthetic
data is the first-class path — BASELINE runs are synthetic/random-init).
"""
import numpy as np
import torch
from torch.utils.data import Dataset

try:
    import h5py
    HAS_H5PY = True
except ImportError:
    HAS_H5PY = False


class DatasetHDF5(Dataset):
    """SWMR HDF5 file with {train,val}_img uint8 NHWC + labels."""

    def __init__(self, hdf5fn, t='train', transform=None,
                 target_transform=None):
        if not HAS_H5PY:
            raise RuntimeError('h5py is not available; use the synthetic '
                               'dataset path (data.synthetic)')
        self.hf = h5py.File(hdf5fn, 'r', libver='latest', swmr=True)
        self.t = t
        self.n_images = self.hf['%s_img' % t].shape[0]
        self.d_labels = self.hf['%s_labels' % t]
        self.d_imgs = self.hf['%s_img' % t]
        self.transform = transform
        self.target_transform = target_transform

    def __len__(self):
        return self.n_images

    def __getitem__(self, index):
        img = self.d_imgs[index]          # HWC uint8
        target = int(self.d_labels[index]) - 1
        if self.transform is not None:
            img = self.transform(img)
        else:
            img = torch.from_numpy(
                np.ascontiguousarray(img.transpose(2, 0, 1))).float() / 255.
        if self.target_transform is not None:
            target = self.target_transform(target)
        return img, target
