"""Data-layer parity tests: CSV feature columns (another-example.py:19-95),
raw MNIST IDX loading (mnist_dataset.py:4-26), TF_CONFIG bootstrap
(03:68-74)."""

import gzip
import json
import os
import struct

import numpy as np
import torch


def test_csv_feature_columns(tmp_path):
    from gradient_accumulation_tf_estimator_amd.data.csv import (
        CategoricalColumn, NumericColumn, build_features, csv_input_fn, parse_csv)

    p = tmp_path / "housing.csv"
    p.write_text(
        "crim,chas,rm,medv\n"
        "0.1,0,6.5,24.0\n"
        "0.2,1,7.1,30.1\n"
        "0.3,0,5.9,18.2\n"
        ",1,,10.0\n"  # empty cells -> defaults
    )
    cols = [NumericColumn("crim", default=0.05),
            CategoricalColumn("chas", vocabulary=["0", "1"]),
            NumericColumn("rm", default=6.0)]
    raw, labels = parse_csv(str(p), cols, "medv")
    assert raw["crim"] == [0.1, 0.2, 0.3, 0.05]
    assert raw["rm"][3] == 6.0
    assert labels == [24.0, 30.1, 18.2, 10.0]

    x, stats = build_features(raw, cols)
    assert x.shape == (4, 4)  # crim + 2 one-hot + rm
    # z-scored numeric columns have ~0 mean
    np.testing.assert_allclose(x[:, 0].mean().item(), 0.0, atol=1e-5)
    np.testing.assert_allclose(x[:, 1:3].sum(1).numpy(), np.ones(4))  # one-hot
    # eval split reuses train stats rather than recomputing
    x2, stats2 = build_features(raw, cols, stats=stats)
    assert stats2 == stats
    np.testing.assert_allclose(x2.numpy(), x.numpy())

    fn, dim, _ = csv_input_fn(str(p), cols, "medv", batch_size=2, num_epochs=1)
    batches = list(fn())
    assert dim == 4
    assert sum(b[1].numel() for b in batches) == 4


def _write_idx(tmp_path, n=7):
    img = np.arange(n * 784, dtype=np.uint8).reshape(n, 784) % 251
    lab = (np.arange(n) % 10).astype(np.uint8)
    with gzip.open(tmp_path / "train-images-idx3-ubyte.gz", "wb") as f:
        f.write(struct.pack(">IIII", 2051, n, 28, 28) + img.tobytes())
    with gzip.open(tmp_path / "train-labels-idx1-ubyte.gz", "wb") as f:
        f.write(struct.pack(">II", 2049, n) + lab.tobytes())
    # test split: uncompressed fallback
    with open(tmp_path / "t10k-images-idx3-ubyte", "wb") as f:
        f.write(struct.pack(">IIII", 2051, 2, 28, 28) + img[:2].tobytes())
    with open(tmp_path / "t10k-labels-idx1-ubyte", "wb") as f:
        f.write(struct.pack(">II", 2049, 2) + lab[:2].tobytes())
    return img, lab


def test_mnist_idx_loader(tmp_path):
    from gradient_accumulation_tf_estimator_amd.data import mnist_idx

    img, lab = _write_idx(tmp_path)
    ds = mnist_idx.load(str(tmp_path))
    x, y = ds["train"].features, ds["train"].labels
    assert x.shape == (7, 28, 28, 1) and x.dtype == torch.float32
    assert float(x.max()) <= 1.0
    np.testing.assert_allclose(x[0].flatten().numpy() * 255.0,
                               img[0].astype(np.float32), atol=1e-4)
    assert y.tolist() == lab.tolist()
    assert len(ds["test"]) == 2  # uncompressed fallback path


def test_tf_config_bootstrap(monkeypatch):
    from gradient_accumulation_tf_estimator_amd.parallel.launch import (
        init_from_tf_config)

    for k in ("WORLD_SIZE", "RANK", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("TF_CONFIG", json.dumps({
        "cluster": {"worker": ["127.0.0.1:23456"]},
        "task": {"type": "worker", "index": 0},
    }))
    ctx = init_from_tf_config()
    # single worker -> no process group, but the env mapping happened
    assert os.environ["MASTER_ADDR"] == "127.0.0.1"
    assert os.environ["MASTER_PORT"] == "23456"
    assert ctx.world_size == 1 and ctx.rank == 0
