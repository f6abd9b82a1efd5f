"""Product library on CPU: it loads, exports the full C-ABI, and fails
LOUDLY (COPR_ERR_NO_GPU) without a GPU — no silent CPU fallback."""
import ctypes as C

import pytest
import torch

import tikv_amd
from tikv_amd import _ffi as F


def test_abi_symbols_present():
    lib = F.load_lib()
    # every entry point include/copr_gpu.h declares must resolve
    import os
    import re
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    hdr = open(os.path.join(root, "include", "copr_gpu.h")).read()
    declared = sorted(set(re.findall(r"\b(copr_[a-z0-9_]+)\s*\(", hdr)))
    assert len(declared) >= 15
    for sym in declared:
        assert getattr(lib, sym) is not None


@pytest.mark.skipif(torch.cuda.is_available(), reason="GPU present")
def test_no_gpu_fails_loudly():
    lib = F.load_lib()
    h = C.c_void_p()
    st = lib.copr_engine_create(0, C.byref(h))
    assert st == F.COPR_ERR_NO_GPU


def test_generator_runs_on_cpu():
    g = tikv_amd.GenRegion(config_index=0, n_rows=100)
    try:
        assert g.n_kv == 100
        assert g.key_bytes() == 100 * 19
        assert g.val_bytes() > 0
    finally:
        g.close()
