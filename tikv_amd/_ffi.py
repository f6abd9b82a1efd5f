"""ctypes mirrors of include/copr_types.h + the libcopr.so C-ABI.

Python here is plumbing only (test/bench orchestration); the product compute
path is C-ABI -> C++ runner -> HIP kernels.
"""
import ctypes as C
import os

_HERE = os.path.dirname(os.path.abspath(__file__))


class CoprFieldType(C.Structure):
    _fields_ = [("tp", C.c_int32), ("flag", C.c_uint32), ("flen", C.c_int32),
                ("decimal", C.c_int32), ("collate", C.c_int32)]


class CoprColumnInfo(C.Structure):
    _fields_ = [("column_id", C.c_int64), ("ft", CoprFieldType),
                ("pk_handle", C.c_int32),
                ("default_val", C.POINTER(C.c_uint8)),
                ("default_val_len", C.c_uint32)]


class CoprExprNode(C.Structure):
    _fields_ = [("kind", C.c_int32), ("sig", C.c_int32), ("n_args", C.c_int32),
                ("ft", CoprFieldType), ("i64_val", C.c_int64),
                ("f64_val", C.c_double),
                ("bytes_val", C.POINTER(C.c_uint8)), ("bytes_len", C.c_uint32)]


class CoprExpr(C.Structure):
    _fields_ = [("nodes", C.POINTER(CoprExprNode)), ("n_nodes", C.c_uint32)]


class CoprAggDef(C.Structure):
    _fields_ = [("func", C.c_int32), ("arg", CoprExpr), ("out_ft", CoprFieldType)]


class CoprExecutor(C.Structure):
    _fields_ = [("kind", C.c_int32),
                ("columns", C.POINTER(CoprColumnInfo)), ("n_columns", C.c_uint32),
                ("desc", C.c_int32),
                ("conditions", C.POINTER(CoprExpr)), ("n_conditions", C.c_uint32),
                ("group_by", C.POINTER(CoprExpr)), ("n_group_by", C.c_uint32),
                ("aggs", C.POINTER(CoprAggDef)), ("n_aggs", C.c_uint32),
                ("limit", C.c_uint64),
                ("order_by", C.POINTER(CoprExpr)),
                ("order_desc", C.POINTER(C.c_int32)),
                ("n_order_by", C.c_uint32)]


class CoprDagRequest(C.Structure):
    _fields_ = [("executors", C.POINTER(CoprExecutor)), ("n_executors", C.c_uint32),
                ("output_offsets", C.POINTER(C.c_uint32)),
                ("n_output_offsets", C.c_uint32),
                ("flags", C.c_uint64), ("div_precision_increment", C.c_int32),
                ("paging_size", C.c_uint64), ("encode_type", C.c_int32)]


class CoprExecSummary(C.Structure):
    _fields_ = [("num_produced_rows", C.c_uint64), ("num_iterations", C.c_uint64),
                ("time_processed_ns", C.c_uint64)]


class CoprSelectResult(C.Structure):
    _fields_ = [("data", C.POINTER(C.c_uint8)), ("data_len", C.c_uint64),
                ("n_rows", C.c_uint64),
                ("summaries", C.POINTER(CoprExecSummary)),
                ("n_summaries", C.c_uint32), ("resume_row", C.c_uint64)]


class CoprGenSpec(C.Structure):
    _fields_ = [("config_index", C.c_int32), ("table_id", C.c_int64),
                ("n_rows", C.c_uint64), ("first_handle", C.c_uint64),
                ("n_cols", C.c_uint32), ("row_format", C.c_int32)]


class CoprGenOut(C.Structure):
    _fields_ = [("keys", C.POINTER(C.c_uint8)), ("key_offs", C.POINTER(C.c_uint64)),
                ("vals", C.POINTER(C.c_uint8)), ("val_offs", C.POINTER(C.c_uint64)),
                ("n_kv", C.c_uint64)]


# enums (copr_types.h)
TP_LONGLONG = 8
TP_DOUBLE = 5
TP_VARCHAR = 15
TP_NEWDECIMAL = 0xF6
FLAG_NOT_NULL = 1
FLAG_UNSIGNED = 1 << 5

EXPR_COLUMN_REF, EXPR_CONST_NULL, EXPR_CONST_INT, EXPR_CONST_UINT, \
    EXPR_CONST_REAL, EXPR_CONST_BYTES, EXPR_CONST_DECIMAL, EXPR_SCALAR_FUNC = range(8)

(SIG_LT_INT, SIG_LE_INT, SIG_GT_INT, SIG_GE_INT, SIG_EQ_INT, SIG_NE_INT,
 SIG_LT_REAL, SIG_LE_REAL, SIG_GT_REAL, SIG_GE_REAL, SIG_EQ_REAL, SIG_NE_REAL,
 SIG_LOGICAL_AND, SIG_LOGICAL_OR, SIG_UNARY_NOT,
 SIG_PLUS_INT, SIG_MINUS_INT, SIG_MULTIPLY_INT,
 SIG_INT_IS_NULL, SIG_INT_IS_TRUE, SIG_INT_IS_FALSE) = range(1, 22)

(AGG_COUNT, AGG_SUM, AGG_AVG, AGG_MAX, AGG_MIN, AGG_FIRST,
 AGG_BIT_AND, AGG_BIT_OR, AGG_BIT_XOR) = range(9)

(EXEC_TABLE_SCAN, EXEC_INDEX_SCAN, EXEC_SELECTION, EXEC_SIMPLE_AGG,
 EXEC_FAST_HASH_AGG, EXEC_SLOW_HASH_AGG, EXEC_STREAM_AGG, EXEC_LIMIT,
 EXEC_TOPN, EXEC_PROJECTION) = range(10)

COPR_OK = 0
COPR_ERR_NO_GPU = 1
COPR_ERR_UNSUPPORTED = 3

_lib = None


def load_lib():
    """Load the product engine library. Raises OSError if it is not built —
    the product path fails loudly rather than falling back."""
    global _lib
    if _lib is not None:
        return _lib
    path = os.path.join(_HERE, "libcopr.so")
    lib = C.CDLL(path)
    lib.copr_last_error.restype = C.c_char_p
    lib.copr_engine_create.restype = C.c_int
    lib.copr_engine_create.argtypes = [C.c_int, C.POINTER(C.c_void_p)]
    lib.copr_engine_destroy.argtypes = [C.c_void_p]
    lib.copr_region_create.restype = C.c_int
    lib.copr_region_create.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
        C.POINTER(C.c_void_p)]
    lib.copr_region_destroy.argtypes = [C.c_void_p]
    lib.copr_region_create_mvcc.restype = C.c_int
    lib.copr_region_create_mvcc.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64, C.c_uint64,
        C.POINTER(C.c_void_p)]
    lib.copr_region_create_mvcc_with_default.restype = C.c_int
    lib.copr_region_create_mvcc_with_default.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
        C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
        C.c_uint64, C.POINTER(C.c_void_p)]
    lib.copr_region_create_sst.restype = C.c_int
    lib.copr_region_create_sst.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_void_p)]
    lib.copr_region_create_sst_mvcc.restype = C.c_int
    lib.copr_region_create_sst_mvcc.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.c_uint64, C.c_uint64,
        C.POINTER(C.c_void_p)]
    lib.copr_region_dump.restype = C.c_int
    lib.copr_region_dump.argtypes = [C.c_void_p, C.c_void_p, C.POINTER(CoprGenOut)]
    lib.copr_region_num_kv.restype = C.c_uint64
    lib.copr_region_num_kv.argtypes = [C.c_void_p]
    lib.copr_dag_run.restype = C.c_int
    lib.copr_dag_run.argtypes = [C.c_void_p, C.POINTER(CoprDagRequest),
                                 C.POINTER(C.c_void_p), C.c_uint32,
                                 C.POINTER(CoprSelectResult)]
    lib.copr_result_free.argtypes = [C.POINTER(CoprSelectResult)]
    lib.copr_checksum.restype = C.c_int
    lib.copr_checksum.argtypes = [C.c_void_p, C.POINTER(C.c_void_p), C.c_uint32,
                                  C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
                                  C.POINTER(C.c_uint64)]
    lib.copr_gen_region.restype = C.c_int
    lib.copr_gen_region.argtypes = [C.POINTER(CoprGenSpec), C.POINTER(CoprGenOut)]
    lib.copr_gen_free.argtypes = [C.POINTER(CoprGenOut)]
    # RCCL merge surface (copr_comm.cpp)
    lib.copr_comm_id.restype = C.c_int
    lib.copr_comm_id.argtypes = [C.POINTER(C.c_uint8)]
    lib.copr_comm_create.restype = C.c_int
    lib.copr_comm_create.argtypes = [C.c_void_p, C.POINTER(C.c_uint8),
                                     C.c_int, C.c_int]
    lib.copr_comm_destroy.argtypes = [C.c_void_p]
    lib.copr_merge_count.restype = C.c_int
    lib.copr_merge_count.argtypes = [C.c_void_p, C.POINTER(C.c_uint64)]
    lib.copr_merge_checksum.restype = C.c_int
    lib.copr_merge_checksum.argtypes = [C.c_void_p, C.POINTER(C.c_uint64)]
    lib.copr_merge_sum_i128.restype = C.c_int
    lib.copr_merge_sum_i128.argtypes = [C.c_void_p, C.POINTER(C.c_uint64),
                                        C.POINTER(C.c_uint64)]
    lib.copr_merge_sum_f64.restype = C.c_int
    lib.copr_merge_sum_f64.argtypes = [C.c_void_p, C.POINTER(C.c_double)]
    _lib = lib
    return lib
