"""Raw MNIST IDX-gz loader (the reference's mnist_dataset.py:4-26).

The reference reads the original LeCun gz files with FixedLengthRecordDataset
(header_bytes 16 for images / 8 for labels, record_bytes 784 / 1), converts
uint8 -> float32/255 and reshapes 28x28x1. This is the torch-native
equivalent; ``load`` returns {"train": ArrayDataset, "test": ArrayDataset}
with image tensors [N, 28, 28, 1] in [0, 1] and int64 labels, ready for
data/input_fn.py's shard/shuffle/batch/repeat composition.

No network access is assumed: point ``load`` at a directory that already
contains train-images-idx3-ubyte.gz etc. (plain uncompressed files work too).
"""

from __future__ import annotations

import gzip
import os
import struct
from typing import Dict

import numpy as np
import torch

from .input_fn import ArrayDataset

FILES = {
    "train": ("train-images-idx3-ubyte.gz", "train-labels-idx1-ubyte.gz"),
    "test": ("t10k-images-idx3-ubyte.gz", "t10k-labels-idx1-ubyte.gz"),
}

_IMAGE_HEADER = 16  # magic, count, rows, cols (4 x int32 BE)
_LABEL_HEADER = 8   # magic, count
_RECORD = 28 * 28


def _read(path: str) -> bytes:
    opener = gzip.open if path.endswith(".gz") else open
    with opener(path, "rb") as f:
        return f.read()


def read_images(path: str) -> torch.Tensor:
    """[N, 28, 28, 1] float32 in [0,1] (reference read_image: uint8/255)."""
    buf = _read(path)
    magic, n, rows, cols = struct.unpack(">IIII", buf[:_IMAGE_HEADER])
    if magic != 2051 or rows != 28 or cols != 28:
        raise ValueError(f"not an MNIST image IDX file: {path} (magic={magic})")
    a = np.frombuffer(buf, dtype=np.uint8, offset=_IMAGE_HEADER, count=n * _RECORD)
    return torch.from_numpy(a.astype(np.float32) / 255.0).reshape(n, 28, 28, 1)


def read_labels(path: str) -> torch.Tensor:
    """[N] int64 (reference read_label: uint8 -> int32)."""
    buf = _read(path)
    magic, n = struct.unpack(">II", buf[:_LABEL_HEADER])
    if magic != 2049:
        raise ValueError(f"not an MNIST label IDX file: {path} (magic={magic})")
    a = np.frombuffer(buf, dtype=np.uint8, offset=_LABEL_HEADER, count=n)
    return torch.from_numpy(a.astype(np.int64))


def load(data_dir: str) -> Dict[str, ArrayDataset]:
    """{"train","test"} datasets from a directory of IDX(.gz) files
    (mnist_dataset.py:24-26's zipped dict). Missing .gz falls back to the
    uncompressed name."""
    out: Dict[str, ArrayDataset] = {}
    for split, (img_name, lab_name) in FILES.items():
        paths = []
        for name in (img_name, lab_name):
            p = os.path.join(data_dir, name)
            if not os.path.exists(p) and p.endswith(".gz"):
                p = p[:-3]
            paths.append(p)
        images = read_images(paths[0])
        labels = read_labels(paths[1])
        if images.shape[0] != labels.shape[0]:
            raise ValueError(f"{split}: {images.shape[0]} images vs "
                             f"{labels.shape[0]} labels")
        out[split] = ArrayDataset(images, labels)
    return out
