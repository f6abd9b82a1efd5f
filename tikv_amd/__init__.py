"""tikv_amd — MI355X-native TiKV coprocessor batch-executor engine.

Product layout:
  include/copr_gpu.h   — the C-ABI drop-in boundary
  tikv_amd/csrc/       — C++ runner + HIP (gfx950) kernels -> libcopr.so
  tikv_amd/_ffi.py     — ctypes mirror of the descriptor structs
  tikv_amd/runner.py   — request builder + engine wrappers (plumbing)
"""
from . import _ffi  # noqa: F401
from .runner import (  # noqa: F401
    Col, Expr, DagSelect, Engine, Region, GenRegion,
    count_star, count_col, sum_col, avg_col, sum_real, avg_real, max_col, min_col, first_col, bit_op,
    cmp_col_const, field_type, gen_blocks,
)
