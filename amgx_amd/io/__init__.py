from .matrix_market import read_matrix_market, write_matrix_market, read_system

__all__ = ["read_matrix_market", "write_matrix_market", "read_system"]
