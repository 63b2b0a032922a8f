"""CSV data pipeline.

Recreates the reference's DataVec ingestion layer (reference Java:372-377,
395-400): `CSVRecordReader` + `RecordReaderDataSetIterator(reader, batchSize,
labelIndex, numClasses)` producing `DataSet{features, labels}` minibatches.

File format matches the reference notebook's writer (Python/gan.ipynb cell 2):
comma-delimited rows of `num_features` floats followed by one integer label.
"""

from __future__ import annotations

from dataclasses import dataclass
from pathlib import Path
from typing import Iterator, Optional

import numpy as np
import torch


@dataclass
class DataSet:
    """A minibatch: features [N, F] and one-hot labels [N, C].

    Mirrors ND4J's DataSet (features/labels pair) used throughout the
    reference training loop (Java:410-423, 462-466).
    """

    features: torch.Tensor
    labels: torch.Tensor

    def num_examples(self) -> int:
        return int(self.features.shape[0])

    def to(self, device, dtype=None) -> "DataSet":
        f = self.features.to(device=device, dtype=dtype or self.features.dtype)
        l = self.labels.to(device=device, dtype=dtype or self.labels.dtype)
        return DataSet(f, l)


class CSVRecordReader:
    """Reads comma-delimited numeric records (DataVec CSVRecordReader analog).

    Loads the whole file into one float32 ndarray (MNIST-scale data easily
    fits host RAM; large files are read in streaming chunks).
    """

    def __init__(self, skip_lines: int = 0, delimiter: str = ","):
        self.skip_lines = skip_lines
        self.delimiter = delimiter
        self._data: Optional[np.ndarray] = None

    def initialize(self, path: str | Path) -> "CSVRecordReader":
        path = Path(path)
        # native multithreaded parser from the _C extension when available
        # (~50x numpy.loadtxt); numpy fallback keeps the reader usable
        # without the built extension (and for non-comma delimiters)
        if self.delimiter == ",":
            try:
                from ..ops.backend import hip_ext

                t = hip_ext().csv_load(str(path), self.skip_lines)
                if t.numel():
                    self._data = t.numpy()
                    return self
            except (RuntimeError, ImportError):
                pass
        self._data = np.loadtxt(
            path, delimiter=self.delimiter, skiprows=self.skip_lines,
            dtype=np.float32, ndmin=2,
        )
        return self

    @property
    def records(self) -> np.ndarray:
        if self._data is None:
            raise RuntimeError("CSVRecordReader.initialize() not called")
        return self._data


class RecordReaderDataSetIterator:
    """Iterator producing DataSet minibatches from a record reader.

    Args mirror the reference construction (Java:377):
        RecordReaderDataSetIterator(reader, batch_size, label_index, num_classes)

    label_index: column index where the label sits (num_features).
    num_classes: one-hot width of the labels.
    """

    def __init__(
        self,
        reader: CSVRecordReader,
        batch_size: int,
        label_index: int,
        num_classes: int,
        shuffle: bool = False,
        seed: int = 0,
        drop_last: bool = False,
    ):
        self.reader = reader
        self.batch_size = batch_size
        self.label_index = label_index
        self.num_classes = num_classes
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self._epoch = 0

    def __iter__(self) -> Iterator[DataSet]:
        data = self.reader.records
        n = data.shape[0]
        order = np.arange(n)
        if self.shuffle:
            rng = np.random.default_rng(self.seed + self._epoch)
            rng.shuffle(order)
        self._epoch += 1
        for start in range(0, n, self.batch_size):
            idx = order[start : start + self.batch_size]
            if self.drop_last and len(idx) < self.batch_size:
                break
            rows = data[idx]
            feats = torch.from_numpy(rows[:, : self.label_index].copy())
            raw_labels = rows[:, self.label_index].astype(np.int64)
            labels = torch.zeros(len(idx), self.num_classes)
            labels[torch.arange(len(idx)), torch.from_numpy(raw_labels)] = 1.0
            yield DataSet(feats, labels)

    def num_batches(self) -> int:
        n = self.reader.records.shape[0]
        if self.drop_last:
            return n // self.batch_size
        return (n + self.batch_size - 1) // self.batch_size


class TensorDataSetIterator:
    """DataSet minibatches straight from in-memory tensors — skips the
    CSV round trip for programmatic pipelines (ND4J's
    ListDataSetIterator analog). Labels may be class indices [N] (one-
    hot encoded with num_classes) or already-one-hot [N, C].
    """

    def __init__(self, features: torch.Tensor, labels: torch.Tensor,
                 batch_size: int, num_classes: int = 0,
                 shuffle: bool = False, seed: int = 0,
                 drop_last: bool = False):
        if features.shape[0] != labels.shape[0]:
            raise ValueError("features/labels row mismatch")
        self.features = features
        if labels.dim() == 1:
            if num_classes <= 0:
                raise ValueError("index labels need num_classes")
            oh = torch.zeros(labels.shape[0], num_classes)
            oh[torch.arange(labels.shape[0]), labels.long()] = 1.0
            labels = oh
        self.labels = labels
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self._epoch = 0

    def __iter__(self) -> Iterator[DataSet]:
        n = self.features.shape[0]
        order = torch.arange(n)
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self._epoch)
            order = torch.randperm(n, generator=g)
        self._epoch += 1
        for start in range(0, n, self.batch_size):
            idx = order[start:start + self.batch_size]
            if self.drop_last and len(idx) < self.batch_size:
                break
            yield DataSet(self.features[idx], self.labels[idx])

    def num_batches(self) -> int:
        n = self.features.shape[0]
        if self.drop_last:
            return n // self.batch_size
        return (n + self.batch_size - 1) // self.batch_size
