from .csv_reader import (CSVRecordReader, DataSet,
                         RecordReaderDataSetIterator, TensorDataSetIterator)
from .synthetic import (
    pixel_lattice_images,
    transactions_tabular,
    write_synthetic_csv,
)

__all__ = [
    "CSVRecordReader",
    "RecordReaderDataSetIterator",
    "TensorDataSetIterator",
    "DataSet",
    "pixel_lattice_images",
    "transactions_tabular",
    "write_synthetic_csv",
]
