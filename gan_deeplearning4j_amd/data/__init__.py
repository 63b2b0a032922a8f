from .csv_reader import CSVRecordReader, RecordReaderDataSetIterator, DataSet
from .synthetic import (
    pixel_lattice_images,
    transactions_tabular,
    write_synthetic_csv,
)

__all__ = [
    "CSVRecordReader",
    "RecordReaderDataSetIterator",
    "DataSet",
    "pixel_lattice_images",
    "transactions_tabular",
    "write_synthetic_csv",
]
