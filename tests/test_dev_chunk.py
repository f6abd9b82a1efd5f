"""Device TypeChunk encoder (kernels.hip dev_chunk_encode).

For project responses whose output columns all chunk-encode as 8-byte
fixed values (ints, DOUBLE, DURATION), the engine builds the TypeChunk
wire bytes (chunk/column.rs:41-71,1052-1071) entirely on device. These
tests pin it three ways on the same requests:
  - device bytes == host-path bytes (COPR_DEV_CHUNK=0 forces the host
    datum->chunk re-encode, the previously parity-pinned path)
  - device bytes == oracle bytes (the reference restatement)
  - the ineligible shapes (varbytes/decimal outputs) still match the
    oracle via the automatic host fallback.
"""
import ctypes as C
import os

import pytest

import tikv_amd
from tikv_amd import _ffi as F

import test_topn_stream as T
from test_topn_stream import cell_int, cell_null, cell_real, row_key


def _orc():
    return T._orc()


def region_bytes(rows_vals):
    """rows_vals: list of per-row VALUE byte strings; keys are handles 0..n."""
    keys = b"".join(row_key(i) for i in range(len(rows_vals)))
    vals = b"".join(rows_vals)
    vo = [0]
    for v in rows_vals:
        vo.append(vo[-1] + len(v))
    ko = [19 * i for i in range(len(rows_vals) + 1)]
    kb = (C.c_uint8 * max(len(keys), 1)).from_buffer_copy(keys or b"\0")
    vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
    return (C.cast(kb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(ko))(*ko),
            C.cast(vb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(vo))(*vo),
            len(rows_vals), (kb, vb))


def run_three_ways(engine, req, raw):
    """device path vs forced host path vs oracle; returns the agreed bytes."""
    orc = _orc()
    k, ko, v, vo, n, keep = raw
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        os.environ.pop("COPR_DEV_CHUNK", None)
        d_dev, n_dev, _, r_dev = engine.dag_run(req, [rgn], with_resume=True)
        os.environ["COPR_DEV_CHUNK"] = "0"
        try:
            d_host, n_host, _, r_host = engine.dag_run(req, [rgn],
                                                       with_resume=True)
        finally:
            del os.environ["COPR_DEV_CHUNK"]
    finally:
        rgn.close()
    assert n_dev == n_host, (n_dev, n_host)
    assert r_dev == r_host
    assert d_dev == d_host
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    assert n_dev == orows
    assert d_dev == od
    return d_dev, n_dev


def mixed_int_rows(n):
    """v1 rows over cols 1..4 with nulls and missing columns:
    col1 always present (filter column), col2 sometimes missing (has a
    default), col3 nullable + sometimes missing, col4 unsigned."""
    rows = []
    for i in range(n):
        v = cell_int(1, (i * 37) % 100 - 50)
        if i % 3 != 0:
            v += cell_int(2, i * 11)
        if i % 5 == 0:
            v += cell_null(3)
        elif i % 5 < 3:
            v += cell_int(3, -i)
        v += cell_int(4, i * 7)
        rows.append(v)
    return region_bytes(rows)


def int_cols():
    return [tikv_amd.Col(1),
            tikv_amd.Col(2, default_val=b"\x08" + T.var_i64(7)),
            tikv_amd.Col(3),
            tikv_amd.Col(4, flag=F.FLAG_UNSIGNED),
            tikv_amd.Col(-1, pk_handle=True)]


@pytest.mark.gpu
def test_dev_chunk_int_cols(engine):
    raw = mixed_int_rows(90)
    sel = tikv_amd.cmp_col_const(0, F.SIG_GT_INT, -40)
    req = tikv_amd.DagSelect(int_cols()).where(sel).chunked().build()
    data, nrows = run_three_ways(engine, req, raw)
    assert nrows > 0 and data


@pytest.mark.gpu
def test_dev_chunk_limit_and_paging(engine):
    raw = mixed_int_rows(300)
    sel = tikv_amd.cmp_col_const(0, F.SIG_GT_INT, -40)
    req = (tikv_amd.DagSelect(int_cols()).where(sel).limit(41)
           .chunked().build())
    _, nrows = run_three_ways(engine, req, raw)
    assert nrows == 41
    req = (tikv_amd.DagSelect(int_cols()).where(sel).paging(25)
           .chunked().build())
    _, nrows = run_three_ways(engine, req, raw)
    assert nrows >= 25


@pytest.mark.gpu
def test_dev_chunk_paging_plus_limit(engine):
    """paging and LIMIT together: the ladder stops at the paging
    boundary AND caps at the remaining limit; resume rows agree."""
    raw = mixed_int_rows(400)
    sel = tikv_amd.cmp_col_const(0, F.SIG_GT_INT, -40)
    for limit, page in ((30, 100), (200, 21), (37, 37)):
        req = (tikv_amd.DagSelect(int_cols()).where(sel).limit(limit)
               .paging(page).chunked().build())
        data, nrows = run_three_ways(engine, req, raw)
        assert nrows <= limit


@pytest.mark.gpu
def test_dev_chunk_double_col(engine):
    rows = []
    for i in range(80):
        v = cell_int(1, i)
        v += cell_null(2) if i % 7 == 0 else cell_real(2, i * 0.5 - 3.25)
        rows.append(v)
    raw = region_bytes(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    data, nrows = run_three_ways(engine, req, raw)
    assert nrows == 80


@pytest.mark.gpu
def test_dev_chunk_v2_rows(engine):
    """row-v2 values (raw LE payloads re-encoded): mixed widths, a
    missing column (default fill), mixed with v1 rows in one region."""
    from test_oracle_exec import v2_row
    rows = []
    for i in range(64):
        if i % 2 == 0:
            rows.append(bytes(v2_row([(1, i), (2, i * 1000 + 7),
                                      (4, (i * 631) % (1 << 33))])))
        else:
            rows.append(cell_int(1, i) + cell_int(2, -i) + cell_int(4, i))
    raw = region_bytes(rows)
    cols = [tikv_amd.Col(1),
            tikv_amd.Col(2),
            tikv_amd.Col(3, default_val=b"\x08" + T.var_i64(-9)),
            tikv_amd.Col(4, flag=F.FLAG_UNSIGNED)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    data, nrows = run_three_ways(engine, req, raw)
    assert nrows == 64


@pytest.mark.gpu
def test_dev_chunk_index_project(engine):
    """chunked project over an INDEX scan: spans reference the key
    stream; device == host == oracle on key-form and value-form handle
    layouts."""
    orc = _orc()
    for layout in (0, 1):
        g = tikv_amd.GenRegion(config_index=4, n_rows=40000, table_id=1,
                               n_cols=layout)
        try:
            rgn = engine.region(g)
            try:
                cols = [tikv_amd.Col(1), tikv_amd.Col(2),
                        tikv_amd.Col(-1, pk_handle=True)]
                req = (tikv_amd.DagSelect(cols, index=True)
                       .chunked().build())
                os.environ.pop("COPR_DEV_CHUNK", None)
                d_dev, n_dev, _ = engine.dag_run(req, [rgn])
                os.environ["COPR_DEV_CHUNK"] = "0"
                try:
                    d_host, n_host, _ = engine.dag_run(req, [rgn])
                finally:
                    del os.environ["COPR_DEV_CHUNK"]
                assert (n_dev, d_dev) == (n_host, d_host), layout
                od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                        g.val_offs, g.n_kv)
                assert n_dev == orows == 40000, layout
                assert d_dev == od, layout
            finally:
                rgn.close()
        finally:
            g.close()


@pytest.mark.gpu
def test_dev_chunk_topn(engine):
    """chunked TopN: the sub-region project routes its decoded order
    column through the filter channel; all-int outputs take the device
    encoder. device == host == oracle."""
    raw = mixed_int_rows(500)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, default_val=b"\x08" + T.var_i64(7)),
            tikv_amd.Col(3), tikv_amd.Col(4, flag=F.FLAG_UNSIGNED)]
    for desc in (False, True):
        req = (tikv_amd.DagSelect(cols)
               .topn(tikv_amd.Expr().col(0), 63, desc=desc)
               .chunked().build())
        data, nrows = run_three_ways(engine, req, raw)
        assert nrows == 63, desc


@pytest.mark.gpu
def test_dev_chunk_fallback_varbytes(engine):
    """a varbytes output column is ineligible for the device encoder; the
    silent host fallback must still match the oracle bit-for-bit."""
    rows = [{1: i, 2: b"s%d" % (i % 5)} for i in range(50)]
    k, ko, v, vo, n, keep = T.region_of_mixed(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_VARCHAR)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    raw = (k, ko, v, vo, n, keep)
    run_three_ways(engine, req, raw)


@pytest.mark.gpu
def test_dev_chunk_large_ladder(engine):
    """150k generated rows, int column only via output offsets: many
    1024-row chunks; oracle + host + device agree."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=2, n_rows=150001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(1),
                    tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
                    tikv_amd.Col(3, tp=F.TP_VARCHAR)]
            sel = tikv_amd.cmp_col_const(0, F.SIG_LT_INT, -500000000)
            req = (tikv_amd.DagSelect(cols).where(sel).output([0])
                   .chunked().build())
            os.environ.pop("COPR_DEV_CHUNK", None)
            d_dev, n_dev, _ = engine.dag_run(req, [rgn])
            os.environ["COPR_DEV_CHUNK"] = "0"
            try:
                d_host, n_host, _ = engine.dag_run(req, [rgn])
            finally:
                del os.environ["COPR_DEV_CHUNK"]
            assert (n_dev, d_dev) == (n_host, d_host)
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert n_dev == orows
            assert d_dev == od
        finally:
            rgn.close()
    finally:
        g.close()
