from .array_table import ArrayTable
from .kv_table import KVTable
from .matrix_table import MatrixTable
from .sparse_matrix import SparseMatrixTable

__all__ = ["ArrayTable", "MatrixTable", "SparseMatrixTable", "KVTable"]
