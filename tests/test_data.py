import numpy as np
import torch

from gan_deeplearning4j_amd.data import (
    CSVRecordReader,
    RecordReaderDataSetIterator,
    pixel_lattice_images,
    transactions_tabular,
    write_synthetic_csv,
)


def test_pixel_lattice_shapes():
    x, y = pixel_lattice_images(32, 28, 28, 1, 10, seed=1)
    assert x.shape == (32, 1, 28, 28)
    assert x.min() >= 0 and x.max() <= 1
    assert y.shape == (32,) and y.max() < 10


def test_transactions_shapes():
    x, y = transactions_tabular(64, 32, 2, seed=1)
    assert x.shape == (64, 32)
    assert x.min() >= 0 and x.max() <= 1.0001


def test_csv_roundtrip(tmp_path):
    # CSV format matches the reference notebook writer (cell 2):
    # 784 floats + integer label per row
    p = write_synthetic_csv(tmp_path / "train.csv", "pixel_lattice", n=50,
                            height=28, width=28, channels=1)
    reader = CSVRecordReader().initialize(p)
    assert reader.records.shape == (50, 785)
    it = RecordReaderDataSetIterator(reader, batch_size=20, label_index=784,
                                     num_classes=10)
    batches = list(it)
    assert len(batches) == 3
    assert batches[0].features.shape == (20, 784)
    assert batches[0].labels.shape == (20, 10)
    # one-hot labels
    assert torch.all(batches[0].labels.sum(dim=1) == 1)
    assert batches[2].features.shape == (10, 784)


def test_iterator_shuffle_determinism(tmp_path):
    p = write_synthetic_csv(tmp_path / "t.csv", "transactions", n=40,
                            num_features=8)
    r = CSVRecordReader().initialize(p)
    a = list(RecordReaderDataSetIterator(r, 16, 8, 2, shuffle=True, seed=3))
    b = list(RecordReaderDataSetIterator(r, 16, 8, 2, shuffle=True, seed=3))
    assert torch.equal(a[0].features, b[0].features)


def test_native_csv_loader_matches_numpy(tmp_path):
    """The _C extension's multithreaded CSV parser must agree with numpy."""
    from gan_deeplearning4j_amd.ops.backend import has_hip_ext, hip_ext

    if not has_hip_ext():
        import pytest

        pytest.skip("extension not built")
    p = write_synthetic_csv(tmp_path / "n.csv", "transactions", n=123,
                            num_features=17)
    a = hip_ext().csv_load(str(p), 0).numpy()
    b = np.loadtxt(p, delimiter=",", dtype=np.float32)
    assert a.shape == b.shape
    assert np.allclose(a, b, atol=1e-5)


def test_tensor_dataset_iterator():
    import pytest
    import torch
    from gan_deeplearning4j_amd.data import TensorDataSetIterator

    x = torch.arange(20, dtype=torch.float32).reshape(10, 2)
    y = torch.tensor([0, 1, 2, 0, 1, 2, 0, 1, 2, 0])
    it = TensorDataSetIterator(x, y, batch_size=4, num_classes=3)
    batches = list(it)
    assert [b.num_examples() for b in batches] == [4, 4, 2]
    assert batches[0].labels.shape == (4, 3)
    assert torch.equal(batches[0].features, x[:4])
    assert batches[0].labels[0].argmax() == 0
    # drop_last + shuffle determinism per epoch
    it2 = TensorDataSetIterator(x, y, 4, 3, shuffle=True, seed=7,
                                drop_last=True)
    e1 = [b.features.clone() for b in it2]
    assert len(e1) == 2 and it2.num_batches() == 2
    e2 = [b.features for b in it2]
    assert not all(torch.equal(a, b) for a, b in zip(e1, e2))  # reshuffled
    # one-hot passthrough + validation
    oh = torch.eye(3)[y]
    it3 = TensorDataSetIterator(x, oh, 5)
    assert next(iter(it3)).labels.shape == (5, 3)
    with pytest.raises(ValueError):
        TensorDataSetIterator(x, y, 4)      # index labels, no num_classes
    with pytest.raises(ValueError):
        TensorDataSetIterator(x, y[:5], 4, 3)
