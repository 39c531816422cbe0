"""Data pipelines: synthetic-first (BASELINE contract), plus native HDF5 /
PTB / audio equivalents of the reference's loaders."""
from .synthetic import (SyntheticImageDataset, SyntheticPTBDataset,
                        SyntheticAudioDataset, GPUBatchPool, an4_collate,
                        synthetic_imagenet, synthetic_cifar10,
                        synthetic_mnist)
from .ptb import ptb_raw_data, PTBDataset, build_vocab
from .audio import (SpectrogramDataset, BucketingSampler,
                    DistributedBucketingSampler, AudioDataLoader,
                    SpectrogramParser, create_manifest)
from .hdf5 import DatasetHDF5, HAS_H5PY
