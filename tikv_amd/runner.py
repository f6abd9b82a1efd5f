"""High-level request builder + engine wrappers (test/bench orchestration).

`DagSelect` mirrors the reference's request-builder fixture
(test_coprocessor/src/dag.rs:21 DagSelect; ProductTable fixture.rs:25):
it assembles the decoded descriptor structs of include/copr_types.h that the
Rust shim would produce from a tipb::DagRequest.
"""
import ctypes as C

from . import _ffi as F


def field_type(tp=F.TP_LONGLONG, flag=0, flen=-1, decimal=-1, collate=63):
    return F.CoprFieldType(tp, flag, flen, decimal, collate)


class Col:
    """Scan column descriptor (tipb::ColumnInfo subset)."""

    def __init__(self, column_id, tp=F.TP_LONGLONG, flag=0, decimal=-1,
                 pk_handle=False, default_val=None):
        self.column_id = column_id
        self.ft = field_type(tp, flag, decimal=decimal)
        self.pk_handle = pk_handle
        self.default_val = default_val


class Expr:
    """RPN expression builder."""

    def __init__(self):
        self.nodes = []
        self._keep = []

    def col(self, offset, ft=None):
        n = F.CoprExprNode()
        n.kind = F.EXPR_COLUMN_REF
        n.i64_val = offset
        if ft is not None:
            n.ft = ft
        self.nodes.append(n)
        return self

    def const_int(self, v, unsigned=False):
        n = F.CoprExprNode()
        n.kind = F.EXPR_CONST_UINT if unsigned else F.EXPR_CONST_INT
        n.i64_val = C.c_int64(v).value if not unsigned else C.c_int64(v & (2**64 - 1) if v >= 0 else v).value
        n.ft = field_type(F.TP_LONGLONG, F.FLAG_UNSIGNED if unsigned else 0)
        self.nodes.append(n)
        return self

    def const_real(self, v):
        n = F.CoprExprNode()
        n.kind = F.EXPR_CONST_REAL
        n.f64_val = float(v)
        n.ft = field_type(F.TP_DOUBLE)
        self.nodes.append(n)
        return self

    def const_null(self):
        n = F.CoprExprNode()
        n.kind = F.EXPR_CONST_NULL
        n.ft = field_type(F.TP_LONGLONG)
        self.nodes.append(n)
        return self

    def func(self, sig, n_args=2, ft=None):
        n = F.CoprExprNode()
        n.kind = F.EXPR_SCALAR_FUNC
        n.sig = sig
        n.n_args = n_args
        n.ft = ft if ft is not None else field_type(F.TP_LONGLONG)
        self.nodes.append(n)
        return self

    def build(self):
        arr = (F.CoprExprNode * len(self.nodes))(*self.nodes)
        self._keep.append(arr)
        return F.CoprExpr(arr, len(self.nodes)), self._keep


class DagSelect:
    """Builds a CoprDagRequest over a table- or index-scan pipeline."""

    def __init__(self, columns, index=False):
        self.columns = columns
        self._keep = []
        self.executors = []
        self.output_offsets = None
        self._n_out_schema = len(columns)
        self._agg_out_cols = 0
        self._has_group = False

        ex = F.CoprExecutor()
        ex.kind = F.EXEC_INDEX_SCAN if index else F.EXEC_TABLE_SCAN
        cols = (F.CoprColumnInfo * len(columns))()
        for i, c in enumerate(columns):
            cols[i].column_id = c.column_id
            cols[i].ft = c.ft
            cols[i].pk_handle = 1 if c.pk_handle else 0
            if c.default_val:
                buf = (C.c_uint8 * len(c.default_val)).from_buffer_copy(c.default_val)
                self._keep.append(buf)
                cols[i].default_val = C.cast(buf, C.POINTER(C.c_uint8))
                cols[i].default_val_len = len(c.default_val)
        self._keep.append(cols)
        ex.columns = cols
        ex.n_columns = len(columns)
        self.executors.append(ex)

    def where(self, *exprs):
        """Selection; multiple conditions are ANDed
        (selection_executor.rs:86)."""
        built = []
        for expr in exprs:
            e, keep = expr.build()
            self._keep += keep
            built.append(e)
        conds = (F.CoprExpr * len(built))(*built)
        self._keep.append(conds)
        ex = F.CoprExecutor()
        ex.kind = F.EXEC_SELECTION
        ex.conditions = conds
        ex.n_conditions = len(built)
        self.executors.append(ex)
        return self

    def _agg(self, kind, aggs, group_by=None):
        ex = F.CoprExecutor()
        ex.kind = kind
        arr = (F.CoprAggDef * len(aggs))()
        out_cols = 0
        for i, (func, arg_expr, out_ft) in enumerate(aggs):
            e, keep = arg_expr.build()
            self._keep += keep
            arr[i].func = func
            arr[i].arg = e
            arr[i].out_ft = out_ft
            out_cols += 2 if func == F.AGG_AVG else 1
        self._keep.append(arr)
        ex.aggs = arr
        ex.n_aggs = len(aggs)
        if group_by is not None:
            ge, keep = group_by.build()
            self._keep += keep
            garr = (F.CoprExpr * 1)(ge)
            self._keep.append(garr)
            ex.group_by = garr
            ex.n_group_by = 1
            out_cols += 1
            self._has_group = True
        self.executors.append(ex)
        self._agg_out_cols = out_cols
        self._n_out_schema = out_cols
        return self

    def simple_agg(self, aggs):
        return self._agg(F.EXEC_SIMPLE_AGG, aggs)

    def hash_agg(self, aggs, group_by):
        return self._agg(F.EXEC_FAST_HASH_AGG, aggs, group_by)

    def stream_agg(self, aggs, group_by):
        """BatchStreamAggregationExecutor: groups are contiguous runs of
        equal keys in input order (stream_aggr_executor.rs:108-117)."""
        return self._agg(F.EXEC_STREAM_AGG, aggs, group_by)

    def topn(self, order_by, n, desc=False):
        """BatchTopNExecutor (top_n_executor.rs): n smallest rows under the
        order-by comparator (NULL first; desc reverses), emitted sorted.
        order_by: Expr or [(Expr, desc_bool), ...]."""
        ex = F.CoprExecutor()
        ex.kind = F.EXEC_TOPN
        ex.limit = n
        if not isinstance(order_by, list):
            order_by = [(order_by, desc)]
        earr = (F.CoprExpr * len(order_by))()
        darr = (C.c_int32 * len(order_by))()
        for i, (expr, d) in enumerate(order_by):
            e, keep = expr.build()
            self._keep += keep
            earr[i] = e
            darr[i] = 1 if d else 0
        self._keep += [earr, darr]
        ex.order_by = earr
        ex.order_desc = darr
        ex.n_order_by = len(order_by)
        self.executors.append(ex)
        return self

    def limit(self, n):
        ex = F.CoprExecutor()
        ex.kind = F.EXEC_LIMIT
        ex.limit = n
        self.executors.append(ex)
        return self

    def paging(self, size):
        self._paging = size
        return self

    def chunked(self):
        """TypeChunk response encoding (tipb EncodeType::TypeChunk)."""
        self._encode_type = 1
        return self

    def output(self, offsets):
        self.output_offsets = list(offsets)
        return self

    def build(self):
        if self.output_offsets is None:
            self.output_offsets = list(range(self._n_out_schema))
        exarr = (F.CoprExecutor * len(self.executors))(*self.executors)
        offs = (C.c_uint32 * len(self.output_offsets))(*self.output_offsets)
        self._keep += [exarr, offs]
        req = F.CoprDagRequest()
        req.executors = exarr
        req.n_executors = len(self.executors)
        req.output_offsets = offs
        req.n_output_offsets = len(self.output_offsets)
        req.paging_size = getattr(self, "_paging", 0)
        req.encode_type = getattr(self, "_encode_type", 0)
        self._req = req
        return req


# convenience agg-def helpers (out field types as TiDB would set them:
# count -> LongLong; sum/avg over int/decimal -> NewDecimal)
def count_star():
    return (F.AGG_COUNT, Expr().const_int(1), field_type(F.TP_LONGLONG))


def count_col(offset):
    return (F.AGG_COUNT, Expr().col(offset), field_type(F.TP_LONGLONG))


def sum_col(offset, decimal=-1):
    return (F.AGG_SUM, Expr().col(offset), field_type(F.TP_NEWDECIMAL, decimal=decimal))


def avg_col(offset, decimal=-1):
    return (F.AGG_AVG, Expr().col(offset), field_type(F.TP_NEWDECIMAL, decimal=decimal))


def sum_real(offset):
    return (F.AGG_SUM, Expr().col(offset), field_type(F.TP_DOUBLE))


def avg_real(offset):
    return (F.AGG_AVG, Expr().col(offset), field_type(F.TP_DOUBLE))


def max_col(offset, tp=F.TP_LONGLONG, flag=0):
    return (F.AGG_MAX, Expr().col(offset), field_type(tp, flag))


def min_col(offset, tp=F.TP_LONGLONG, flag=0):
    return (F.AGG_MIN, Expr().col(offset), field_type(tp, flag))


def first_col(offset, tp=F.TP_LONGLONG, flag=0):
    return (F.AGG_FIRST, Expr().col(offset), field_type(tp, flag))


def bit_op(func, offset):
    return (func, Expr().col(offset), field_type(F.TP_LONGLONG, F.FLAG_UNSIGNED))


def gen_blocks(gen, restart_interval=16, target_block_bytes=4096):
    """Pack a GenRegion's KV stream into RocksDB-format data blocks.
    Returns (blocks_ptr, block_offs_array, n_blocks, keepalive)."""
    lib = gen._lib
    blocks = C.POINTER(C.c_uint8)()
    offs = C.POINTER(C.c_uint64)()
    n = C.c_uint32()
    st = lib.copr_gen_blocks(gen.keys, gen.key_offs, gen.vals, gen.val_offs,
                             gen.n_kv, restart_interval, target_block_bytes,
                             C.byref(blocks), C.byref(offs), C.byref(n))
    if st != 0:
        raise RuntimeError("copr_gen_blocks: %d" % st)
    return blocks, offs, n.value, (blocks, offs)


def cmp_col_const(offset, sig, const, unsigned_const=False):
    return Expr().col(offset).const_int(const, unsigned_const).func(sig, 2)


class GenRegion:
    """Synthetic region (host buffers from the product generator)."""

    def __init__(self, config_index, n_rows, table_id=1, first_handle=0,
                 n_cols=0, row_format=1):
        lib = F.load_lib()
        spec = F.CoprGenSpec(config_index, table_id, n_rows, first_handle,
                             n_cols, row_format)
        self._out = F.CoprGenOut()
        st = lib.copr_gen_region(C.byref(spec), C.byref(self._out))
        if st != 0:
            raise RuntimeError("generator failed: %d" % st)
        self._lib = lib
        self.n_kv = self._out.n_kv

    @property
    def keys(self):
        return self._out.keys

    @property
    def key_offs(self):
        return self._out.key_offs

    @property
    def vals(self):
        return self._out.vals

    @property
    def val_offs(self):
        return self._out.val_offs

    def key_bytes(self):
        return self._out.key_offs[self.n_kv]

    def val_bytes(self):
        return self._out.val_offs[self.n_kv]

    def close(self):
        if self._out.keys:
            self._lib.copr_gen_free(C.byref(self._out))

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class Engine:
    def __init__(self, device=0):
        lib = F.load_lib()
        h = C.c_void_p()
        st = lib.copr_engine_create(device, C.byref(h))
        if st != 0:
            raise RuntimeError("copr_engine_create: %d (%s)" %
                               (st, lib.copr_last_error().decode()))
        self._lib = lib
        self._h = h

    def region(self, gen: GenRegion):
        r = C.c_void_p()
        st = self._lib.copr_region_create(self._h, gen.keys, gen.key_offs,
                                          gen.vals, gen.val_offs, gen.n_kv,
                                          C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_raw(self, keys, key_offs, vals, val_offs, n_kv):
        """Region from explicit host buffers (ctypes pointers/arrays)."""
        r = C.c_void_p()
        st = self._lib.copr_region_create(self._h, keys, key_offs, vals,
                                          val_offs, n_kv, C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_blocks(self, blocks, block_offs, n_blocks):
        """Region from uncompressed RocksDB data blocks (device parse)."""
        r = C.c_void_p()
        st = self._lib.copr_region_create_blocks(self._h, blocks, block_offs,
                                                 n_blocks, C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create_blocks: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_sst(self, file_bytes):
        """Region from one whole BlockBasedTable SST file: footer + index
        walk + crc32c verify + per-block decompression on host, block
        parse on device."""
        buf = (C.c_uint8 * max(len(file_bytes), 1)).from_buffer_copy(
            file_bytes or b"\0")
        r = C.c_void_p()
        st = self._lib.copr_region_create_sst(
            self._h, C.cast(buf, C.POINTER(C.c_uint8)), len(file_bytes),
            C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create_sst: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_sst_mvcc(self, file_bytes, read_ts):
        """SST of write-CF records: file walk + device MVCC filter."""
        buf = (C.c_uint8 * max(len(file_bytes), 1)).from_buffer_copy(
            file_bytes or b"\0")
        r = C.c_void_p()
        st = self._lib.copr_region_create_sst_mvcc(
            self._h, C.cast(buf, C.POINTER(C.c_uint8)), len(file_bytes),
            read_ts, C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create_sst_mvcc: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_mvcc_with_default(self, keys, key_offs, vals, val_offs, n_kv,
                                 dkeys, dkey_offs, dvals, dval_offs,
                                 n_default, read_ts):
        """MVCC region with the default CF beside the write CF: Puts without
        short values resolve from the default stream (forward.rs:433-515)."""
        r = C.c_void_p()
        st = self._lib.copr_region_create_mvcc_with_default(
            self._h, keys, key_offs, vals, val_offs, n_kv,
            dkeys, dkey_offs, dvals, dval_offs, n_default,
            C.c_uint64(read_ts), C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create_mvcc_with_default: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_blocks_mvcc(self, blocks, block_offs, n_blocks, read_ts):
        r = C.c_void_p()
        st = self._lib.copr_region_create_blocks_mvcc(
            self._h, blocks, block_offs, n_blocks,
            C.c_uint64(read_ts), C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create_blocks_mvcc: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def region_mvcc(self, gen: GenRegion, read_ts):
        """Build a visible-row region from a raw write-CF stream (the
        device MVCC version filter)."""
        r = C.c_void_p()
        st = self._lib.copr_region_create_mvcc(
            self._h, gen.keys, gen.key_offs, gen.vals, gen.val_offs,
            gen.n_kv, read_ts, C.byref(r))
        if st != 0:
            raise RuntimeError("copr_region_create_mvcc: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return Region(self, r)

    def dump_region(self, region):
        """test/debug: host copies of a resident region."""
        out = F.CoprGenOut()
        st = self._lib.copr_region_dump(self._h, region._h, C.byref(out))
        if st != 0:
            raise RuntimeError("copr_region_dump: %d" % st)
        n = out.n_kv
        koffs = [out.key_offs[i] for i in range(n + 1)]
        voffs = [out.val_offs[i] for i in range(n + 1)]
        keys = C.string_at(out.keys, koffs[-1]) if koffs[-1] else b""
        vals = C.string_at(out.vals, voffs[-1]) if voffs[-1] else b""
        self._lib.copr_gen_free(C.byref(out))
        return keys, koffs, vals, voffs, n

    def dag_run(self, req, regions, with_resume=False):
        arr = (C.c_void_p * len(regions))(*[r._h for r in regions])
        res = F.CoprSelectResult()
        st = self._lib.copr_dag_run(self._h, C.byref(req), arr, len(regions),
                                    C.byref(res))
        if st != 0:
            raise RuntimeError("copr_dag_run: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        data = C.string_at(res.data, res.data_len) if res.data_len else b""
        n_rows = res.n_rows
        kernel_ns = res.summaries[0].time_processed_ns if res.n_summaries else 0
        resume = res.resume_row
        self._lib.copr_result_free(C.byref(res))
        if with_resume:
            return data, n_rows, kernel_ns, resume
        return data, n_rows, kernel_ns

    # ---- RCCL merge surface (the engine's only collective).
    # The 128-byte id comes from rank 0's comm_id() and is distributed
    # out-of-band (the bench uses a gloo broadcast as the bootstrap channel,
    # mirroring NCCL's own bootstrap-over-sockets).
    @staticmethod
    def comm_id():
        lib = F.load_lib()
        buf = (C.c_uint8 * 128)()
        st = lib.copr_comm_id(buf)
        if st != 0:
            raise RuntimeError("copr_comm_id: %d (%s)" %
                               (st, lib.copr_last_error().decode()))
        return bytes(buf)

    def comm_create(self, comm_id: bytes, n_ranks: int, rank: int):
        buf = (C.c_uint8 * 128).from_buffer_copy(comm_id)
        st = self._lib.copr_comm_create(self._h, buf, n_ranks, rank)
        if st != 0:
            raise RuntimeError("copr_comm_create: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))

    def comm_destroy(self):
        self._lib.copr_comm_destroy(self._h)

    def merge_count(self, count: int) -> int:
        v = C.c_uint64(count)
        st = self._lib.copr_merge_count(self._h, C.byref(v))
        if st != 0:
            raise RuntimeError("copr_merge_count: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return v.value

    def merge_checksum(self, xor_val: int) -> int:
        v = C.c_uint64(xor_val)
        st = self._lib.copr_merge_checksum(self._h, C.byref(v))
        if st != 0:
            raise RuntimeError("copr_merge_checksum: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return v.value

    def merge_sum_i128(self, lo: int, hi: int):
        lov = C.c_uint64(lo)
        hiv = C.c_uint64(hi)
        st = self._lib.copr_merge_sum_i128(self._h, C.byref(lov), C.byref(hiv))
        if st != 0:
            raise RuntimeError("copr_merge_sum_i128: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return lov.value, hiv.value

    def merge_sum_f64(self, x: float) -> float:
        v = C.c_double(x)
        st = self._lib.copr_merge_sum_f64(self._h, C.byref(v))
        if st != 0:
            raise RuntimeError("copr_merge_sum_f64: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return v.value

    def checksum(self, regions):
        arr = (C.c_void_p * len(regions))(*[r._h for r in regions])
        cs = C.c_uint64()
        kvs = C.c_uint64()
        byts = C.c_uint64()
        st = self._lib.copr_checksum(self._h, arr, len(regions), C.byref(cs),
                                     C.byref(kvs), C.byref(byts))
        if st != 0:
            raise RuntimeError("copr_checksum: %d (%s)" %
                               (st, self._lib.copr_last_error().decode()))
        return cs.value, kvs.value, byts.value

    def close(self):
        if self._h:
            self._lib.copr_engine_destroy(self._h)
            self._h = None


class Region:
    def __init__(self, eng, h):
        self._eng = eng
        self._h = h

    def close(self):
        if self._h:
            self._eng._lib.copr_region_destroy(self._h)
            self._h = None
