from .array_table import ArrayTable
from .kv_table import KVTable
from .matrix_table import MatrixTable

__all__ = ["ArrayTable", "MatrixTable", "KVTable"]
