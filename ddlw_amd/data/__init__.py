from .preprocess import preprocess_bytes, preprocess_batch, preprocess_pil
from .pipeline import build_tables, read_table, table_path
from .synthetic import make_synthetic_jpeg_tree, make_synthetic_dataset
from .loader import ShardedParquetLoader, make_converter
from .decode import ParallelDecoder, decode_resize_u8

__all__ = [
    "ParallelDecoder",
    "decode_resize_u8",
    "preprocess_bytes",
    "preprocess_batch",
    "preprocess_pil",
    "build_tables",
    "read_table",
    "table_path",
    "make_synthetic_jpeg_tree",
    "make_synthetic_dataset",
    "ShardedParquetLoader",
    "make_converter",
]
