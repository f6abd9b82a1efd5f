"""Oracle executor-pipeline semantics on hand-built regions, plus the
generator cross-checked against an independent Python restatement of its RNG
and row encoding (CPU)."""
import ctypes as C
import importlib.util
import os

import tikv_amd
from tikv_amd import _ffi as F

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _orc():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


# ---------- python restatements for fixture building ----------
def var_u64(v):
    out = bytearray()
    while v >= 0x80:
        out.append(0x80 | (v & 0x7F))
        v >>= 7
    out.append(v)
    return bytes(out)


def var_i64(v):
    uv = (v << 1) & (2**64 - 1)
    if v < 0:
        uv = (~((v << 1) & (2**64 - 1))) & (2**64 - 1)
    return var_u64(uv)


def int_handle_key(table_id, handle):
    def cmp64(x):
        return ((x ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big")
    return b"t" + cmp64(table_id) + b"_r" + cmp64(handle)


def row_v1(cells):
    """cells: list of (col_id, datum_bytes)"""
    out = bytearray()
    for cid, datum in cells:
        out += bytes([8]) + var_i64(cid) + datum
    if not cells:
        out += bytes([0])
    return bytes(out)


def d_varint(v):
    return bytes([8]) + var_i64(v)


D_NULL = bytes([0])


def d_int(v):
    return bytes([3]) + (((v ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big"))


def d_uint(v):
    return bytes([4]) + v.to_bytes(8, "big")


def make_region(rows):
    """rows: list of (key_bytes, value_bytes) -> ctypes arrays"""
    keys = b"".join(k for k, _ in rows)
    vals = b"".join(v for _, v in rows)
    key_offs = [0]
    val_offs = [0]
    for k, v in rows:
        key_offs.append(key_offs[-1] + len(k))
        val_offs.append(val_offs[-1] + len(v))
    kb = (C.c_uint8 * max(len(keys), 1)).from_buffer_copy(keys or b"\0")
    vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
    ko = (C.c_uint64 * len(key_offs))(*key_offs)
    vo = (C.c_uint64 * len(val_offs))(*val_offs)
    return kb, ko, vb, vo, len(rows)


def run_oracle(req, region):
    orc = _orc()
    kb, ko, vb, vo, n = region
    return orc.dag_run(req, C.cast(kb, C.POINTER(C.c_uint8)), ko,
                       C.cast(vb, C.POINTER(C.c_uint8)), vo, n)


# ---------- tests ----------
def test_scan_project_basic():
    """TableScan of 2 int columns + handle; NULLs, missing columns with
    defaults (table_scan_executor.rs:456-483 semantics)."""
    default_7 = d_varint(7)
    cols = [
        tikv_amd.Col(1),
        tikv_amd.Col(2, default_val=default_7),
        tikv_amd.Col(-1, pk_handle=True),  # id ignored for handle
    ]
    rows = [
        (int_handle_key(1, 10), row_v1([(1, d_varint(100)), (2, d_varint(200))])),
        (int_handle_key(1, 11), row_v1([(1, D_NULL)])),           # col2 missing -> default
        (int_handle_key(1, 12), row_v1([(2, d_varint(-5))])),     # col1 missing -> NULL
        (int_handle_key(1, 13), b""),                             # empty row
    ]
    req = tikv_amd.DagSelect(cols).build()
    data, n = run_oracle(req, make_region(rows))
    assert n == 4
    expect = (
        d_varint(100) + d_varint(200) + d_int(10) +
        D_NULL + default_7 + d_int(11) +
        D_NULL + d_varint(-5) + d_int(12) +
        D_NULL + default_7 + d_int(13)
    )
    assert data == expect


def test_selection_count():
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
    rows = []
    for i in range(100):
        rows.append((int_handle_key(1, i),
                     row_v1([(1, d_varint(i)), (2, d_varint(i * 2))])))
    # add a NULL row for col1: predicate NULL -> dropped
    rows.append((int_handle_key(1, 1000), row_v1([(1, D_NULL), (2, d_varint(1))])))
    sel = tikv_amd.cmp_col_const(0, F.SIG_LT_INT, 50)
    req = (tikv_amd.DagSelect(cols).where(sel)
           .simple_agg([tikv_amd.count_star()]).build())
    data, n = run_oracle(req, make_region(rows))
    assert n == 1
    assert data == d_int(50)  # count out_ft is signed LongLong here


def test_simple_agg_sum_avg():
    cols = [tikv_amd.Col(1)]
    rows = [(int_handle_key(1, i), row_v1([(1, d_varint(v))]))
            for i, v in enumerate([5, -3, 10, 0])]
    rows.append((int_handle_key(1, 99), row_v1([(1, D_NULL)])))
    req = tikv_amd.DagSelect(cols).simple_agg(
        [tikv_amd.count_col(0), tikv_amd.sum_col(0), tikv_amd.avg_col(0)]).build()
    data, n = run_oracle(req, make_region(rows))
    assert n == 1
    orc = _orc()
    lib = orc.load_lib()
    out = C.create_string_buffer(64)
    nd = lib.orc_test_dec_from_i64_encode(12, out)
    dec12 = bytes([6]) + out.raw[:nd]
    # count=4 (nulls skipped), sum=Dec(12), avg=(count 4, sum Dec(12))
    assert data == d_int(4) + dec12 + d_uint(4) + dec12


def test_hash_agg_groups():
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
    rows = []
    h = 0
    for g, v in [(1, 10), (2, 20), (1, 30), (3, 5), (2, -20), (1, 2)]:
        rows.append((int_handle_key(1, h),
                     row_v1([(1, d_varint(g)), (2, d_varint(v))])))
        h += 1
    # NULL group
    rows.append((int_handle_key(1, h), row_v1([(1, D_NULL), (2, d_varint(100))])))
    req = tikv_amd.DagSelect(cols).hash_agg(
        [tikv_amd.count_star(), tikv_amd.sum_col(1)],
        tikv_amd.Expr().col(0)).build()
    data, n = run_oracle(req, make_region(rows))
    assert n == 4
    # parse rows and compare as a set (group order is not part of parity)
    rows_out = split_datum_rows(data, 3)
    lib = _orc().load_lib()

    def dec(v):
        out = C.create_string_buffer(64)
        nd = lib.orc_test_dec_from_i64_encode(v, out)
        return bytes([6]) + out.raw[:nd]

    expect = {
        d_int(3) + dec(42) + d_varint_group(1),
        d_int(2) + dec(0) + d_varint_group(2),
        d_int(1) + dec(5) + d_varint_group(3),
        d_int(1) + dec(100) + D_NULL,
    }
    assert set(rows_out) == expect


def d_varint_group(v):
    """group key output: decoded Int column -> INT flag comparable."""
    return d_int(v)


def split_datum_rows(data, cols_per_row):
    """split a datum-encoded response into rows (knowing datum framing)."""
    rows = []
    i = 0
    cur = []
    ncol = 0
    while i < len(data):
        start = i
        flag = data[i]
        i += 1
        if flag == 0:
            pass
        elif flag in (3, 4, 5, 7):
            i += 8
        elif flag in (8, 9):
            while data[i] & 0x80:
                i += 1
            i += 1
        elif flag == 6:
            prec, frac = data[i], data[i + 1]
            D2B = [0, 1, 1, 2, 2, 3, 3, 4, 4, 4]
            ic = prec - frac
            size = (ic // 9) * 4 + D2B[ic % 9] + (frac // 9) * 4 + D2B[frac % 9]
            i += 2 + size
        elif flag == 2:
            ln = 0
            shift = 0
            while True:
                b = data[i]
                i += 1
                ln |= (b & 0x7F) << shift
                shift += 7
                if b < 0x80:
                    break
            ln = ln >> 1  # zigzag, lengths are non-negative
            i += ln
        else:
            raise AssertionError("flag %d" % flag)
        cur.append(data[start:i])
        ncol += 1
        if ncol == cols_per_row:
            rows.append(b"".join(cur))
            cur = []
            ncol = 0
    assert not cur
    return rows


# ---------- generator cross-check ----------
def _xoshiro_py(config_seed, row):
    MASK = 2**64 - 1

    def splitmix(x):
        z = (x + 0x9E3779B97F4A7C15) & MASK
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & MASK
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & MASK
        return (z ^ (z >> 31)), (x + 0x9E3779B97F4A7C15) & MASK

    x = ((config_seed * 0x9E3779B97F4A7C15) & MASK) ^ ((row + 0x243F6A8885A308D3) & MASK)
    s = []
    for _ in range(4):
        v, x = splitmix(x)
        s.append(v)

    def rotl(v, k):
        return ((v << k) | (v >> (64 - k))) & MASK

    def nxt():
        nonlocal s
        result = (rotl((s[0] + s[3]) & MASK, 23) + s[0]) & MASK
        t = (s[1] << 17) & MASK
        s[2] ^= s[0]
        s[3] ^= s[1]
        s[1] ^= s[2]
        s[0] ^= s[3]
        s[2] ^= t
        s[3] = rotl(s[3], 45)
        return result
    return nxt


def test_generator_matches_python_restatement():
    g = tikv_amd.GenRegion(config_index=1, n_rows=64, table_id=1, first_handle=5)
    try:
        assert g.n_kv == 64
        for i in [0, 1, 63]:
            handle = 5 + i
            ko, ko2 = g.key_offs[i], g.key_offs[i + 1]
            key = bytes(g.keys[ko:ko2])
            assert key == int_handle_key(1, handle)
            nxt = _xoshiro_py(0xC0FFEE + 1, handle)
            cells = []
            for cid in range(1, 17):
                v = (nxt() % 2000000001) - 1000000000
                cells.append((cid, d_varint(v)))
            vo, vo2 = g.val_offs[i], g.val_offs[i + 1]
            val = bytes(g.vals[vo:vo2])
            assert val == row_v1(cells), i
    finally:
        g.close()


def test_oracle_on_generated_cfg1():
    g = tikv_amd.GenRegion(config_index=0, n_rows=1000, table_id=1)
    try:
        cols = [tikv_amd.Col(i) for i in range(1, 5)]
        req = tikv_amd.DagSelect(cols).simple_agg(
            [tikv_amd.count_star(), tikv_amd.sum_col(0)]).build()
        orc = _orc()
        data, n = orc.dag_run(req, g.keys, g.key_offs, g.vals, g.val_offs, g.n_kv)
        assert n == 1
        # recompute expected sum with the python RNG
        total = 0
        for i in range(1000):
            nxt = _xoshiro_py(0xC0FFEE + 0, i)
            total += (nxt() % 2000000001) - 1000000000
        lib = orc.load_lib()
        out = C.create_string_buffer(64)
        nd = lib.orc_test_dec_from_i64_encode(total, out)
        assert data == d_int(1000) + bytes([6]) + out.raw[:nd]
    finally:
        g.close()


def test_oracle_paging_resume():
    cols = [tikv_amd.Col(1)]
    rows = [(int_handle_key(1, i), row_v1([(1, d_varint(i))])) for i in range(5000)]
    req = (tikv_amd.DagSelect(cols)
           .where(tikv_amd.cmp_col_const(0, F.SIG_GE_INT, 0))
           .paging(100).build())
    orc = _orc()
    kb, ko, vb, vo, n = make_region(rows)
    data, nrows, resume = orc.dag_run(
        req, C.cast(kb, C.POINTER(C.c_uint8)), ko,
        C.cast(vb, C.POINTER(C.c_uint8)), vo, n, with_resume=True)
    # ladder 32,64,128 -> 224 rows scanned/output at the stopping boundary
    assert nrows == 224 and resume == 224
    # drained case: paging larger than output
    req2 = (tikv_amd.DagSelect(cols)
            .where(tikv_amd.cmp_col_const(0, F.SIG_GE_INT, 0))
            .paging(10000).build())
    data, nrows, resume = orc.dag_run(
        req2, C.cast(kb, C.POINTER(C.c_uint8)), ko,
        C.cast(vb, C.POINTER(C.c_uint8)), vo, n, with_resume=True)
    assert nrows == 5000 and resume == 2**64 - 1


def test_oracle_row_v2_generated():
    """generator row_format=2 parses through the oracle scan pipeline."""
    g = tikv_amd.GenRegion(config_index=0, n_rows=3000, table_id=1, row_format=2)
    try:
        cols = [tikv_amd.Col(i) for i in range(1, 5)]
        req = tikv_amd.DagSelect(cols).simple_agg(
            [tikv_amd.count_star(), tikv_amd.count_col(0)]).build()
        orc = _orc()
        data, n = orc.dag_run(req, g.keys, g.key_offs, g.vals, g.val_offs, g.n_kv)
        assert n == 1
        total = int.from_bytes(data[1:9], "big") ^ (1 << 63)
        nonnull = int.from_bytes(data[10:18], "big") ^ (1 << 63)
        assert total == 3000
        assert 0 < nonnull <= 3000   # ~1/64 of col1 values are NULL
        assert nonnull < 3000        # with 3000 rows some NULLs occur w.h.p.
    finally:
        g.close()


def idx_key(table_id, index_id, datums, handle=None):
    def cmp64(x):
        return ((x ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big")
    k = b"t" + cmp64(table_id) + b"_i" + cmp64(index_id)
    for d in datums:
        k += d
    if handle is not None:
        k += bytes([3]) + cmp64(handle)
    return k


def test_oracle_index_scan():
    """non-unique index: positional comparable datums + handle from key."""
    cols = [tikv_amd.Col(1), tikv_amd.Col(2), tikv_amd.Col(-1, pk_handle=True)]
    rows = []
    for h, (a, b) in enumerate([(5, 100), (7, -3), (5, 42)]):
        rows.append((idx_key(1, 1, [d_int(a), d_int(b)], handle=h), b"0"))
    req = tikv_amd.DagSelect(cols, index=True).build()
    data, n = run_oracle(req, make_region(rows))
    assert n == 3
    expect = (d_int(5) + d_int(100) + d_int(0) +
              d_int(7) + d_int(-3) + d_int(1) +
              d_int(5) + d_int(42) + d_int(2))
    assert data == expect
    # unique index: handle from the 8-byte BE value
    rows_u = [(idx_key(1, 1, [d_int(9), d_int(1)]),
               (12345).to_bytes(8, "big"))]
    data, n = run_oracle(req, make_region(rows_u))
    assert n == 1
    assert data == d_int(9) + d_int(1) + d_int(12345)


def v2_row(cells):
    """tiny row-v2 (small layout) of (id, int) cells, ids ascending."""
    body = b""
    ends = []
    for _, v in cells:
        if 0 <= v <= 0xFF:
            body += bytes([v])
        elif 0 <= v <= 0xFFFF:
            body += v.to_bytes(2, "little")
        elif 0 <= v <= 0xFFFFFFFF:
            body += v.to_bytes(4, "little")
        else:
            body += (v & (2**64 - 1)).to_bytes(8, "little")
        ends.append(len(body))
    out = bytes([128, 0, len(cells), 0, 0, 0])
    out += bytes(cid for cid, _ in cells)
    for e in ends:
        out += e.to_bytes(2, "little")
    return out + body


def test_oracle_index_value_layouts():
    """new index value layouts (index_scan_executor.rs:322-371):
    unique v0 + V4 restore row; non-unique v1 + partition-id option."""
    cols = [tikv_amd.Col(1), tikv_amd.Col(2), tikv_amd.Col(-1, pk_handle=True)]
    req = tikv_amd.DagSelect(cols, index=True).build()

    # unique new-format v0: [TailLen=8] || restore row-v2 || BE handle;
    # columns come from the RESTORE row (V4), not the key datums
    val = bytes([8]) + v2_row([(1, 9), (2, 77)]) + (4242).to_bytes(8, "big")
    rows = [(idx_key(1, 1, [d_int(9), d_int(77)]), val)]
    data, n = run_oracle(req, make_region(rows))
    assert n == 1
    assert data == d_int(9) + d_int(77) + d_int(4242)

    # non-unique new-format v1: [0][125][1][126][pid 8B]; handle from key,
    # V5 restore absent, columns from key datums
    val = bytes([0, 125, 1, 126]) + (1).to_bytes(8, "big")
    rows = [(idx_key(1, 1, [d_int(3), d_int(4)], handle=17), val)]
    data, n = run_oracle(req, make_region(rows))
    assert n == 1
    assert data == d_int(3) + d_int(4) + d_int(17)

    # corrupted tail_len is a loud storage error
    import pytest
    bad = bytes([200]) + bytes(10)
    with pytest.raises(Exception):
        run_oracle(req, make_region([(idx_key(1, 1, [d_int(1), d_int(2)]),
                                      bad)]))


def test_oracle_index_agg():
    cols = [tikv_amd.Col(1), tikv_amd.Col(2), tikv_amd.Col(-1, pk_handle=True)]
    rows = []
    for h, (a, b) in enumerate([(5, 100), (7, -3), (5, 42), (7, 1)]):
        rows.append((idx_key(1, 1, [d_int(a), d_int(b)], handle=h), b"0"))
    req = (tikv_amd.DagSelect(cols, index=True)
           .where(tikv_amd.cmp_col_const(0, F.SIG_GT_INT, 0))
           .hash_agg([tikv_amd.count_star(), tikv_amd.sum_col(1)],
                     tikv_amd.Expr().col(0)).build())
    data, n = run_oracle(req, make_region(rows))
    assert n == 2
    lib = _orc().load_lib()

    def dec(v):
        out = C.create_string_buffer(64)
        nd = lib.orc_test_dec_from_i64_encode(v, out)
        return bytes([6]) + out.raw[:nd]

    got = set(split_datum_rows(data, 3))
    assert got == {d_int(2) + dec(142) + d_int(5), d_int(2) + dec(-2) + d_int(7)}
