"""ctypes binding for the ORACLE library (test infrastructure ONLY).

Importable only from tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg. Reuses the descriptor struct definitions from
tikv_amd._ffi (pure ctypes declarations; no product library is loaded by
importing them).
"""
import ctypes as C
import os
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))
_ROOT = os.path.dirname(_HERE)
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)

from tikv_amd._ffi import CoprDagRequest  # noqa: E402


class OrcResult(C.Structure):
    _fields_ = [("data", C.POINTER(C.c_uint8)), ("data_len", C.c_uint64),
                ("n_rows", C.c_uint64), ("resume_row", C.c_uint64)]


class OrcRegion(C.Structure):
    _fields_ = [("keys", C.POINTER(C.c_uint8)), ("key_offs", C.POINTER(C.c_uint64)),
                ("vals", C.POINTER(C.c_uint8)), ("val_offs", C.POINTER(C.c_uint64)),
                ("n_kv", C.c_uint64)]


_lib = None


def load_lib():
    global _lib
    if _lib is not None:
        return _lib
    lib = C.CDLL(os.path.join(_HERE, "liboracle.so"))
    lib.orc_last_error.restype = C.c_char_p
    lib.orc_dag_run.restype = C.c_int
    lib.orc_dag_run.argtypes = [C.POINTER(CoprDagRequest),
                                C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                C.c_uint64, C.POINTER(OrcResult)]
    lib.orc_result_free.argtypes = [C.POINTER(OrcResult)]
    lib.orc_checksum.restype = C.c_int
    lib.orc_checksum.argtypes = [C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                 C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                 C.c_uint64, C.POINTER(C.c_uint64),
                                 C.POINTER(C.c_uint64), C.POINTER(C.c_uint64)]
    lib.orc_mvcc_filter.restype = C.c_int
    lib.orc_mvcc_filter.argtypes = [C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                    C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
                                    C.c_uint64, C.c_uint64, C.POINTER(OrcRegion)]
    lib.orc_mvcc_filter2.restype = C.c_int
    lib.orc_mvcc_filter2.argtypes = [
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
        C.c_uint64, C.POINTER(OrcRegion)]
    lib.orc_region_free.argtypes = [C.POINTER(OrcRegion)]
    lib.orc_block_parse.restype = C.c_int
    lib.orc_block_parse.argtypes = [C.POINTER(C.c_uint8),
                                    C.POINTER(C.c_uint64), C.c_uint32,
                                    C.POINTER(OrcRegion)]
    lib.orc_crc64_xz.restype = C.c_uint64
    lib.orc_crc64_xz.argtypes = [C.c_char_p, C.c_uint64]
    lib.orc_test_memcmp_encode.restype = C.c_uint64
    lib.orc_test_memcmp_encode.argtypes = [C.c_char_p, C.c_uint64, C.c_int, C.c_char_p]
    lib.orc_test_memcmp_decode.restype = C.c_uint64
    lib.orc_test_memcmp_decode.argtypes = [C.c_char_p, C.c_uint64, C.c_char_p,
                                           C.POINTER(C.c_uint64)]
    lib.orc_test_var_i64_encode.restype = C.c_uint64
    lib.orc_test_var_i64_encode.argtypes = [C.c_int64, C.c_char_p]
    lib.orc_test_var_i64_decode.restype = C.c_int
    lib.orc_test_var_i64_decode.argtypes = [C.c_char_p, C.c_uint64,
                                            C.POINTER(C.c_int64), C.POINTER(C.c_uint64)]
    lib.orc_test_row_key.argtypes = [C.c_int64, C.c_int64, C.c_char_p]
    lib.orc_test_int_handle.restype = C.c_int
    lib.orc_test_int_handle.argtypes = [C.c_char_p, C.c_uint64, C.POINTER(C.c_int64)]
    lib.orc_test_row_v2_col.restype = C.c_int
    lib.orc_test_row_v2_col.argtypes = [C.c_char_p, C.c_uint64, C.c_int64,
                                        C.c_int32, C.c_uint32, C.c_char_p,
                                        C.POINTER(C.c_int)]
    lib.orc_test_dec_add_encode.restype = C.c_int
    lib.orc_test_dec_add_encode.argtypes = [C.c_char_p, C.c_uint64, C.c_char_p,
                                            C.c_uint64, C.c_char_p]
    lib.orc_test_dec_from_i64_encode.restype = C.c_int
    lib.orc_test_dec_from_i64_encode.argtypes = [C.c_int64, C.c_char_p]
    _lib = lib
    return lib


def dag_run(req, keys, key_offs, vals, val_offs, n_kv, with_resume=False):
    """Run the oracle pipeline. Buffer args are ctypes pointers (e.g. from
    GenRegion) or bytes (auto-wrapped)."""
    lib = load_lib()
    res = OrcResult()
    st = lib.orc_dag_run(C.byref(req), keys, key_offs, vals, val_offs, n_kv,
                         C.byref(res))
    if st != 0:
        raise RuntimeError("oracle: %s" % lib.orc_last_error().decode())
    data = C.string_at(res.data, res.data_len) if res.data_len else b""
    n = res.n_rows
    resume = res.resume_row
    lib.orc_result_free(C.byref(res))
    if with_resume:
        return data, n, resume
    return data, n


def checksum(keys, key_offs, vals, val_offs, n_kv):
    lib = load_lib()
    cs = C.c_uint64()
    kvs = C.c_uint64()
    byts = C.c_uint64()
    st = lib.orc_checksum(keys, key_offs, vals, val_offs, n_kv,
                          C.byref(cs), C.byref(kvs), C.byref(byts))
    assert st == 0
    return cs.value, kvs.value, byts.value


def block_parse(blocks, block_offs, n_blocks):
    """Oracle decode of RocksDB data blocks -> (keys, ko, vals, vo, n)."""
    lib = load_lib()
    out = OrcRegion()
    st = lib.orc_block_parse(blocks, block_offs, n_blocks, C.byref(out))
    if st != 0:
        raise RuntimeError("orc_block_parse: %d" % st)
    n = out.n_kv
    koffs = [out.key_offs[i] for i in range(n + 1)]
    voffs = [out.val_offs[i] for i in range(n + 1)]
    keysb = C.string_at(out.keys, koffs[-1]) if koffs[-1] else b""
    valsb = C.string_at(out.vals, voffs[-1]) if voffs[-1] else b""
    lib.orc_region_free(C.byref(out))
    return keysb, koffs, valsb, voffs, n


def sst_parse(file_bytes):
    """Oracle walk of a whole BlockBasedTable SST (footer + index +
    crc32c + decompress + block decode) -> (keys, ko, vals, vo, n)."""
    lib = load_lib()
    lib.orc_sst_parse.restype = C.c_int
    lib.orc_sst_parse.argtypes = [C.POINTER(C.c_uint8), C.c_uint64,
                                  C.POINTER(OrcRegion)]
    buf = (C.c_uint8 * max(len(file_bytes), 1)).from_buffer_copy(
        file_bytes or b"\0")
    out = OrcRegion()
    st = lib.orc_sst_parse(C.cast(buf, C.POINTER(C.c_uint8)),
                           len(file_bytes), C.byref(out))
    if st != 0:
        raise RuntimeError("orc_sst_parse: %d" % st)
    n = out.n_kv
    koffs = [out.key_offs[i] for i in range(n + 1)]
    voffs = [out.val_offs[i] for i in range(n + 1)]
    keysb = C.string_at(out.keys, koffs[-1]) if koffs[-1] else b""
    valsb = C.string_at(out.vals, voffs[-1]) if voffs[-1] else b""
    lib.orc_region_free(C.byref(out))
    return keysb, koffs, valsb, voffs, n


def mvcc_filter(keys, key_offs, vals, val_offs, n_kv, read_ts,
                default_cf=None):
    """Run the oracle MVCC filter; returns (keys, key_offs, vals, val_offs,
    n). default_cf = (dkeys, dkey_offs, dvals, dval_offs, n) resolves Puts
    without short values (forward.rs:433-515)."""
    lib = load_lib()
    out = OrcRegion()
    if default_cf is not None:
        dk, dko, dv, dvo, dn = default_cf
        st = lib.orc_mvcc_filter2(keys, key_offs, vals, val_offs, n_kv,
                                  dk, dko, dv, dvo, dn, read_ts, C.byref(out))
    else:
        st = lib.orc_mvcc_filter(keys, key_offs, vals, val_offs, n_kv,
                                 read_ts, C.byref(out))
    if st != 0:
        raise RuntimeError("orc_mvcc_filter: %d" % st)
    n = out.n_kv
    koffs = [out.key_offs[i] for i in range(n + 1)]
    voffs = [out.val_offs[i] for i in range(n + 1)]
    keysb = C.string_at(out.keys, koffs[-1]) if koffs[-1] else b""
    valsb = C.string_at(out.vals, voffs[-1]) if voffs[-1] else b""
    lib.orc_region_free(C.byref(out))
    return keysb, koffs, valsb, voffs, n
