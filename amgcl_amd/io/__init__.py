"""Matrix I/O: MatrixMarket and raw binary (parity: amgcl/io/mm.hpp:52,
amgcl/io/binary.hpp:70)."""
from .mm import mm_read, mm_write
from .binary import read_crs, write_crs, read_dense, write_dense
