"""TopN and StreamAgg executors.

Reference semantics pinned:
  - TopN (top_n_executor.rs): keep the n smallest rows under the order-by
    comparator (datum order: NULL < any value; desc reverses the whole
    order including NULL placement), emit in sorted order after the source
    drains. Tie order is unspecified in the reference (unstable heap);
    both the oracle and the engine emit ties in source-row order.
  - StreamAgg (stream_aggr_executor.rs:108-117): groups are contiguous
    runs of equal group-key values in input order; non-adjacent equal
    keys are NOT merged.

CPU tests pin the oracle against hand-computed answers; GPU tests compare
the engine bit-for-bit against the oracle on generated regions.
"""
import ctypes as C
import importlib.util
import os

import pytest

import tikv_amd
from tikv_amd import _ffi as F

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _orc():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


# ---- hand-built v1 rows: [VAR_INT col_id, datum]* ----------------------
def var_u64(v):
    out = bytearray()
    while v >= 0x80:
        out.append(0x80 | (v & 0x7F))
        v >>= 7
    out.append(v)
    return bytes(out)


def var_i64(v):
    uv = (v << 1) ^ (0xFFFFFFFFFFFFFFFF if v < 0 else 0)
    return var_u64(uv & (2**64 - 1))


def cell_int(col_id, v):
    return b"\x08" + var_i64(col_id) + b"\x08" + var_i64(v)


def cell_null(col_id):
    return b"\x08" + var_i64(col_id) + b"\x00"


def row_key(handle):
    return (b"t" + (5).to_bytes(8, "big") + b"_r"
            + ((handle ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big"))


def region_of(rows):
    """rows: list of dicts {col_id: int-or-None}"""
    keys = b"".join(row_key(i) for i in range(len(rows)))
    vals = b""
    vo = [0]
    for r in rows:
        v = b""
        for cid in sorted(r):
            v += cell_null(cid) if r[cid] is None else cell_int(cid, r[cid])
        vals += v
        vo.append(len(vals))
    ko = [19 * i for i in range(len(rows) + 1)]
    kb = (C.c_uint8 * max(len(keys), 1)).from_buffer_copy(keys or b"\0")
    vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
    return (C.cast(kb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(ko))(*ko),
            C.cast(vb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(vo))(*vo),
            len(rows), (kb, vb))


def dec_int(data, pos):
    """decode one INT_FLAG datum at pos -> (value, next_pos)"""
    assert data[pos] == 3
    u = int.from_bytes(data[pos + 1:pos + 9], "big") ^ (1 << 63)
    return u - (1 << 64) if u >= (1 << 63) else u, pos + 9


def test_topn_oracle_basic():
    orc = _orc()
    rows = [{1: v} for v in [5, -3, None, 9, -3, 0]]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1)]
    req = (tikv_amd.DagSelect(cols)
           .topn(tikv_amd.Expr().col(0), 3).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 3
    # NULL first, then -3 (row 1 before row 4: stable), i.e. NULL,-3,-3
    assert data[0] == 0          # NIL datum
    a, p = dec_int(data, 1)
    b, p = dec_int(data, p)
    assert (a, b) == (-3, -3) and p == len(data)


def test_topn_oracle_desc_and_filter():
    orc = _orc()
    rows = [{1: v} for v in [5, -3, None, 9, -3, 0]]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1)]
    sel = tikv_amd.cmp_col_const(0, F.SIG_NE_INT, 9)
    req = (tikv_amd.DagSelect(cols).where(sel)
           .topn(tikv_amd.Expr().col(0), 2, desc=True).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    # filter drops 9 and the NULL row (NULL != 9 is NULL -> not true);
    # desc: 5, 0
    assert nrows == 2
    a, p = dec_int(data, 0)
    b, p = dec_int(data, p)
    assert (a, b) == (5, 0) and p == len(data)


def test_stream_agg_oracle_runs():
    orc = _orc()
    # sorted-ish input with a non-adjacent repeat: runs must NOT merge
    seq = [1, 1, 2, 2, 2, 1, None, None, 3]
    rows = [{1: g, 2: 10} for g in seq]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
    req = (tikv_amd.DagSelect(cols)
           .stream_agg([tikv_amd.count_star()], tikv_amd.Expr().col(0))
           .build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    # runs: [1,1], [2,2,2], [1], [None,None], [3]
    assert nrows == 5
    got = []
    p = 0
    for _ in range(5):
        cnt, p = dec_int(data, p)
        if data[p] == 0:
            key, p = None, p + 1
        else:
            key, p = dec_int(data, p)
        got.append((cnt, key))
    assert got == [(2, 1), (3, 2), (1, 1), (2, None), (1, 3)]
    assert p == len(data)


@pytest.mark.gpu
def test_topn_gpu_parity(engine):
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=200001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            sel = tikv_amd.cmp_col_const(3, F.SIG_GT_INT, 0)
            for desc in (False, True):
                req = (tikv_amd.DagSelect(cols).where(sel)
                       .topn(tikv_amd.Expr().col(4), 100, desc=desc)
                       .output([0, 4, 7]).build())
                gd, gr, _ = engine.dag_run(req, [rgn])
                od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                        g.val_offs, g.n_kv)
                assert gr == orows
                assert gd == od
        finally:
            rgn.close()
    finally:
        g.close()


@pytest.mark.gpu
def test_stream_agg_gpu_parity(engine):
    orc = _orc()
    # handle-ordered scan of generated rows: col values are random, so this
    # exercises run detection heavily (many 1-row runs + occasional repeats)
    g = tikv_amd.GenRegion(config_index=0, n_rows=100001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 5)]
            sel = tikv_amd.cmp_col_const(2, F.SIG_GE_INT, -10**9)
            req = (tikv_amd.DagSelect(cols).where(sel)
                   .stream_agg([tikv_amd.count_star(),
                                tikv_amd.max_col(2)],
                               tikv_amd.Expr().col(0)).build())
            gd, gr, _ = engine.dag_run(req, [rgn])
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert gr == orows
            assert gd == od
        finally:
            rgn.close()
    finally:
        g.close()


# ---- bytes group keys (SlowHashAggregationImpl) ------------------------
def var_bytes_cell(col_id, payload):
    return (b"\x08" + var_i64(col_id) + b"\x02" + var_i64(len(payload))
            + payload)


def region_of_mixed(rows):
    """rows: list of dicts {col_id: int | bytes | None}"""
    keys = b"".join(row_key(i) for i in range(len(rows)))
    vals = b""
    vo = [0]
    for r in rows:
        v = b""
        for cid in sorted(r):
            x = r[cid]
            if x is None:
                v += cell_null(cid)
            elif isinstance(x, bytes):
                v += var_bytes_cell(cid, x)
            else:
                v += cell_int(cid, x)
        vals += v
        vo.append(len(vals))
    ko = [19 * i for i in range(len(rows) + 1)]
    kb = (C.c_uint8 * max(len(keys), 1)).from_buffer_copy(keys or b"\0")
    vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
    return (C.cast(kb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(ko))(*ko),
            C.cast(vb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(vo))(*vo),
            len(rows), (kb, vb))


def split_rows(data, n_cols):
    """split a datum-encoded response into per-row byte strings (ints,
    NILs and compact-bytes only)."""
    rows = []
    p = 0
    while p < len(data):
        start = p
        for _ in range(n_cols):
            f = data[p]
            if f == 0:
                p += 1
            elif f in (3, 4, 5):
                p += 9
            elif f == 2:
                q = p + 1
                uv, sh = 0, 0
                while True:
                    b = data[q]
                    q += 1
                    uv |= (b & 0x7F) << sh
                    sh += 7
                    if not (b & 0x80):
                        break
                n = uv >> 1
                p = q + n
            else:
                raise AssertionError("unexpected datum flag %d" % f)
        rows.append(bytes(data[start:p]))
    return rows


def test_bytes_group_oracle():
    orc = _orc()
    rows = [{1: 1, 2: b"aa"}, {1: 2, 2: b"bb"}, {1: 3, 2: b"aa"},
            {1: 4, 2: None}, {1: 5, 2: b"aa"}, {1: 6, 2: None},
            {1: 7, 2: b""}]
    k, ko, v, vo, n, keep = region_of_mixed(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_VARCHAR)]
    req = (tikv_amd.DagSelect(cols)
           .hash_agg([tikv_amd.count_star(), tikv_amd.max_col(0)],
                     tikv_amd.Expr().col(1)).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 4  # aa, bb, NULL, ""
    got = set(split_rows(data, 3))
    want = set()
    for cnt, mx, key in ((3, 5, b"aa"), (1, 2, b"bb"), (2, 6, None),
                         (1, 7, b"")):
        row = b"\x03" + ((cnt ^ (1 << 63)).to_bytes(8, "big"))
        row += b"\x03" + ((mx ^ (1 << 63)).to_bytes(8, "big"))
        if key is None:
            row += b"\x00"
        else:
            row += b"\x02" + var_i64(len(key)) + key
        want.add(row)
    assert got == want


@pytest.mark.gpu
def test_bytes_group_gpu_parity(engine):
    orc = _orc()
    # hand-built low-cardinality case (repeats + NULL + empty)
    rows = []
    keys = [b"k1", b"key-two", b"", None, b"k1", b"x" * 40]
    for i in range(3000):
        rows.append({1: i, 2: keys[i % len(keys)]})
    k, ko, v, vo, n, keep = region_of_mixed(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_VARCHAR)]
    req = (tikv_amd.DagSelect(cols)
           .hash_agg([tikv_amd.count_star(), tikv_amd.max_col(0)],
                     tikv_amd.Expr().col(1)).build())
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    import ctypes as CT
    kb = CT.cast((CT.c_uint8 * max(ko[-1], 1)).from_buffer_copy(
        CT.string_at(k, ko[-1]) or b"\0"), CT.POINTER(CT.c_uint8))
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        gd, grows, _ = engine.dag_run(req, [rgn])
        assert grows == orows
        assert set(split_rows(gd, 3)) == set(split_rows(od, 3))
    finally:
        rgn.close()
    # generated cfg2 region: random varbytes (mostly distinct keys) + filter
    g = tikv_amd.GenRegion(config_index=2, n_rows=60001, table_id=5)
    try:
        rgn2 = engine.region(g)
        try:
            cols2 = [tikv_amd.Col(1),
                     tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
                     tikv_amd.Col(3, tp=F.TP_VARCHAR)]
            sel = tikv_amd.cmp_col_const(0, F.SIG_GT_INT, 0)
            req2 = (tikv_amd.DagSelect(cols2).where(sel)
                    .hash_agg([tikv_amd.count_star(), tikv_amd.max_col(0)],
                              tikv_amd.Expr().col(2)).build())
            gd2, gr2, _ = engine.dag_run(req2, [rgn2])
            od2, or2 = orc.dag_run(req2, g.keys, g.key_offs, g.vals,
                                   g.val_offs, g.n_kv)
            assert gr2 == or2
            assert set(split_rows(gd2, 3)) == set(split_rows(od2, 3))
        finally:
            rgn2.close()
    finally:
        g.close()


# ---- TypeChunk response encoding ---------------------------------------
def decode_chunks(data, col_kinds):
    """decode chunk-encoded response: col_kinds per output column in
    ('i64', 'u64', 'dec', 'bytes'). Returns list of row tuples."""
    rows = []
    p = 0
    while p < len(data):
        cols = []
        for kind in col_kinds:
            length = int.from_bytes(data[p:p + 4], "little")
            null_cnt = int.from_bytes(data[p + 4:p + 8], "little")
            p += 8
            bitmap = None
            if null_cnt:
                nb = (length + 7) // 8
                bitmap = data[p:p + nb]
                p += nb
            offs = None
            if kind == "bytes":
                offs = [int.from_bytes(data[p + 8 * i:p + 8 * i + 8],
                                       "little") for i in range(length + 1)]
                p += 8 * (length + 1)
            vals = []
            for r in range(length):
                null = bitmap is not None and not (bitmap[r >> 3] >> (r & 7)) & 1
                if kind == "bytes":
                    v = None if null else bytes(data[p + offs[r]:p + offs[r + 1]])
                elif kind == "dec":
                    raw = data[p + 40 * r:p + 40 * (r + 1)]
                    v = None if null else raw
                else:
                    u = int.from_bytes(data[p + 8 * r:p + 8 * (r + 1)], "little")
                    if kind == "i64" and u >= 1 << 63:
                        u -= 1 << 64
                    v = None if null else u
                vals.append(v)
            if kind == "bytes":
                p += offs[length]
            elif kind == "dec":
                p += 40 * length
            else:
                p += 8 * length
            cols.append(vals)
        rows.extend(zip(*cols))
    return rows


def test_chunked_oracle_project():
    orc = _orc()
    rows = [{1: 5, 2: b"aa"}, {1: None, 2: b"xyz"}, {1: -7, 2: None},
            {1: 1 << 40, 2: b""}]
    k, ko, v, vo, n, keep = region_of_mixed(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_VARCHAR)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 4
    got = decode_chunks(data, ["i64", "bytes"])
    assert got == [(5, b"aa"), (None, b"xyz"), (-7, None), (1 << 40, b"")]


def test_chunked_oracle_batching():
    """chunks follow the 32 -> x2 -> 1024 source-batch ladder."""
    orc = _orc()
    rows = [{1: i} for i in range(100)]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 100
    # chunk lengths: 32, 64, 4
    lens = []
    p = 0
    while p < len(data):
        length = int.from_bytes(data[p:p + 4], "little")
        lens.append(length)
        p += 8 + 8 * length    # no nulls, fixed col
    assert lens == [32, 64, 4]


def test_chunked_oracle_agg_decimal():
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=2, n_rows=3000, table_id=5)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
            tikv_amd.Col(3, tp=F.TP_VARCHAR)]
    req = (tikv_amd.DagSelect(cols)
           .simple_agg([tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=2)])
           .chunked().build())
    data, nrows = orc.dag_run(req, g.keys, g.key_offs, g.vals, g.val_offs,
                              g.n_kv)
    assert nrows == 1
    got = decode_chunks(data, ["i64", "dec"])
    assert len(got) == 1 and got[0][0] == 3000
    assert isinstance(got[0][1], (bytes, bytearray)) and len(got[0][1]) == 40
    g.close()


@pytest.mark.gpu
def test_chunked_gpu_parity(engine):
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=2, n_rows=150001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(1),
                    tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
                    tikv_amd.Col(3, tp=F.TP_VARCHAR)]
            reqs = []
            # simple agg (count + decimal sum)
            reqs.append(tikv_amd.DagSelect(cols).simple_agg(
                [tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=2)])
                .chunked().build())
            # filtered project incl. varbytes + decoded filter column
            sel = tikv_amd.cmp_col_const(0, F.SIG_LT_INT, -880000000)
            reqs.append(tikv_amd.DagSelect(cols).where(sel).chunked().build())
            # topn by int col
            reqs.append(tikv_amd.DagSelect(cols)
                        .topn(tikv_amd.Expr().col(0), 77).chunked().build())
            for req in reqs:
                gd, gr, _ = engine.dag_run(req, [rgn])
                od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                        g.val_offs, g.n_kv)
                assert gr == orows
                assert gd == od
        finally:
            rgn.close()
    finally:
        g.close()


# ---- FIRST aggregate ----------------------------------------------------
def test_first_oracle():
    orc = _orc()
    seq = [(1, 10), (1, None), (2, None), (2, 7), (None, 3)]
    rows = [{1: g, 2: v} for g, v in seq]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
    req = (tikv_amd.DagSelect(cols)
           .stream_agg([tikv_amd.first_col(1), tikv_amd.count_star()],
                       tikv_amd.Expr().col(0)).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 3   # runs: [1,1], [2,2], [None]
    got = []
    p = 0
    for _ in range(3):
        if data[p] == 0:
            fv, p = None, p + 1
        else:
            fv, p = dec_int(data, p)
        cnt, p = dec_int(data, p)
        if data[p] == 0:
            key, p = None, p + 1
        else:
            key, p = dec_int(data, p)
        got.append((fv, cnt, key))
    # first of run 1 = 10; run 2 first value is NULL; NULL-key run first = 3
    assert got == [(10, 2, 1), (None, 2, 2), (3, 1, None)]


def test_first_simple_oracle():
    orc = _orc()
    rows = [{1: v} for v in [None, 5, 6]]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1)]
    req = (tikv_amd.DagSelect(cols)
           .simple_agg([tikv_amd.first_col(0), tikv_amd.count_star()])
           .build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 1
    assert data[0] == 0          # first row's value is NULL
    cnt, p = dec_int(data, 1)
    assert cnt == 3 and p == len(data)


@pytest.mark.gpu
def test_first_gpu_parity(engine):
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=0, n_rows=50001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 5)]
            # stream agg with FIRST
            req = (tikv_amd.DagSelect(cols)
                   .stream_agg([tikv_amd.first_col(1), tikv_amd.count_star()],
                               tikv_amd.Expr().col(0)).build())
            gd, gr, _ = engine.dag_run(req, [rgn])
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert (gr, gd) == (orows, od)
            # simple agg with FIRST (reroutes through the one-run stream)
            sel = tikv_amd.cmp_col_const(2, F.SIG_GT_INT, 0)
            req2 = (tikv_amd.DagSelect(cols).where(sel)
                    .simple_agg([tikv_amd.first_col(1), tikv_amd.count_star(),
                                 tikv_amd.max_col(3)]).build())
            gd2, gr2, _ = engine.dag_run(req2, [rgn])
            od2, or2 = orc.dag_run(req2, g.keys, g.key_offs, g.vals,
                                   g.val_offs, g.n_kv)
            assert (gr2, gd2) == (or2, od2)
        finally:
            rgn.close()
    finally:
        g.close()


@pytest.mark.gpu
def test_project_row_v2_gpu_parity(engine):
    """project mode over row-v2 values: raw v2 cells re-encode as datums
    (compat_v1.rs:28-126); NULL cells and the decoded filter column too."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=20001, table_id=5,
                           row_format=2)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            sel = tikv_amd.cmp_col_const(3, F.SIG_LT_INT, -500000000)
            req = (tikv_amd.DagSelect(cols).where(sel)
                   .output([0, 3, 9, 15]).build())
            gd, gr, _ = engine.dag_run(req, [rgn])
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert gr == orows
            assert gd == od
            # chunked variant exercises the v2 datum -> chunk path too
            req2 = (tikv_amd.DagSelect(cols).where(sel)
                    .output([0, 3, 9, 15]).chunked().build())
            gd2, gr2, _ = engine.dag_run(req2, [rgn])
            od2, or2 = orc.dag_run(req2, g.keys, g.key_offs, g.vals,
                                   g.val_offs, g.n_kv)
            assert (gr2, gd2) == (or2, od2)
        finally:
            rgn.close()
    finally:
        g.close()


@pytest.mark.gpu
def test_topn_filter_col_in_output_gpu(engine):
    """selection + TopN where BOTH the filter column and the order column
    appear in the output: both must encode in decoded form."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=80001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            sel = tikv_amd.cmp_col_const(3, F.SIG_GT_INT, -900000000)
            for desc in (False, True):
                req = (tikv_amd.DagSelect(cols).where(sel)
                       .topn(tikv_amd.Expr().col(6), 64, desc=desc)
                       .output([3, 6, 11]).build())
                gd, gr, _ = engine.dag_run(req, [rgn])
                od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                        g.val_offs, g.n_kv)
                assert gr == orows
                assert gd == od
        finally:
            rgn.close()
    finally:
        g.close()


@pytest.mark.gpu
def test_chunked_paging_and_misc_aggs_gpu(engine):
    """paging + TypeChunk coexist (chunks follow the ladder up to the
    resume point); avg/bit aggregates through the chunk encoder."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=70001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            sel = tikv_amd.cmp_col_const(3, F.SIG_GT_INT, 0)
            req = (tikv_amd.DagSelect(cols).where(sel).paging(500)
                   .output([1, 5]).chunked().build())
            gd, gr, _, gresume = engine.dag_run(req, [rgn], with_resume=True)
            od, orows, oresume = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                             g.val_offs, g.n_kv,
                                             with_resume=True)
            assert (gr, gresume) == (orows, oresume)
            assert gd == od
            # avg + bit ops, chunked simple agg
            req2 = (tikv_amd.DagSelect(cols)
                    .simple_agg([tikv_amd.avg_col(2),
                                 tikv_amd.bit_op(F.AGG_BIT_XOR, 4),
                                 tikv_amd.bit_op(F.AGG_BIT_AND, 4)])
                    .chunked().build())
            gd2, gr2, _ = engine.dag_run(req2, [rgn])
            od2, or2 = orc.dag_run(req2, g.keys, g.key_offs, g.vals,
                                   g.val_offs, g.n_kv)
            assert (gr2, gd2) == (or2, od2)
        finally:
            rgn.close()
    finally:
        g.close()


# ---- two ANDed selection conditions ------------------------------------
def test_two_conditions_oracle():
    orc = _orc()
    rows = [{1: a, 2: b} for a, b in
            [(5, 1), (15, 1), (5, 9), (None, 1), (7, None), (12, 3)]]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
    req = (tikv_amd.DagSelect(cols)
           .where(tikv_amd.cmp_col_const(0, F.SIG_LT_INT, 10),
                  tikv_amd.cmp_col_const(1, F.SIG_LT_INT, 5))
           .simple_agg([tikv_amd.count_star()]).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    cnt, p = dec_int(data, 0)
    assert cnt == 1 and nrows == 1   # only (5,1)


@pytest.mark.gpu
def test_two_conditions_gpu_parity(engine):
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=120001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            c1 = tikv_amd.cmp_col_const(3, F.SIG_GT_INT, -500000000)
            c2 = tikv_amd.cmp_col_const(7, F.SIG_LE_INT, 250000000)
            # count with both conjuncts (generic collect path)
            req = (tikv_amd.DagSelect(cols).where(c1, c2)
                   .simple_agg([tikv_amd.count_star(),
                                tikv_amd.sum_col(1)]).build())
            gd, gr, _ = engine.dag_run(req, [rgn])
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert (gr, gd) == (orows, od)
            # project: BOTH predicate columns in the output, decoded form
            req2 = (tikv_amd.DagSelect(cols).where(c1, c2)
                    .output([3, 7, 10]).build())
            gd2, gr2, _ = engine.dag_run(req2, [rgn])
            od2, or2 = orc.dag_run(req2, g.keys, g.key_offs, g.vals,
                                   g.val_offs, g.n_kv)
            assert (gr2, gd2) == (or2, od2)
            # hash agg with two conjuncts
            req3 = (tikv_amd.DagSelect(cols).where(c1, c2)
                    .hash_agg([tikv_amd.count_star()],
                              tikv_amd.Expr().col(5)).build())
            gd3, gr3, _ = engine.dag_run(req3, [rgn])
            od3, or3 = orc.dag_run(req3, g.keys, g.key_offs, g.vals,
                                   g.val_offs, g.n_kv)
            assert gr3 == or3
            assert set(split_rows(gd3, 2)) == set(split_rows(od3, 2))
        finally:
            rgn.close()
    finally:
        g.close()


def test_chunk_wire_format_golden():
    """hand-computed TypeChunk bytes (chunk/column.rs:1052-1071): pins the
    wire layout independently of the python decoder used elsewhere."""
    orc = _orc()
    rows = [{1: 5, 2: b"ab"}, {1: None, 2: b"c"}, {1: -2, 2: b"ab"}]
    k, ko, v, vo, n, keep = region_of_mixed(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_VARCHAR)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 3
    # one chunk (first ladder batch of 32 covers all 3 rows), two columns
    exp = bytearray()
    # col 0: i64 fixed; row1 NULL -> bitmap present, bit set = NOT null
    exp += (3).to_bytes(4, "little") + (1).to_bytes(4, "little")
    exp.append(0b00000101)
    exp += (5).to_bytes(8, "little", signed=True)
    exp += (0).to_bytes(8, "little")
    exp += (-2).to_bytes(8, "little", signed=True)
    # col 1: var-size, no nulls -> no bitmap; offsets i64le x4; payloads
    exp += (3).to_bytes(4, "little") + (0).to_bytes(4, "little")
    for off in (0, 2, 3, 5):
        exp += off.to_bytes(8, "little")
    exp += b"abcab"
    assert data == bytes(exp)


@pytest.mark.gpu
def test_executor_edge_cases_gpu(engine):
    """empty-result edges across the new executors."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=5001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            never = tikv_amd.cmp_col_const(3, F.SIG_GT_INT, 2 * 10**9)
            reqs = [
                # topn with nothing surviving
                (tikv_amd.DagSelect(cols).where(never)
                 .topn(tikv_amd.Expr().col(4), 10).build()),
                # stream agg with nothing surviving
                (tikv_amd.DagSelect(cols).where(never)
                 .stream_agg([tikv_amd.count_star()],
                             tikv_amd.Expr().col(0)).build()),
                # simple agg + FIRST with nothing surviving (one zero row)
                (tikv_amd.DagSelect(cols).where(never)
                 .simple_agg([tikv_amd.first_col(1),
                              tikv_amd.count_star()]).build()),
                # topn n larger than the row count
                (tikv_amd.DagSelect(cols)
                 .topn(tikv_amd.Expr().col(4), 100000).output([2]).build()),
                # limit 0
                (tikv_amd.DagSelect(cols).limit(0).build()),
            ]
            for req in reqs:
                gd, gr, _ = engine.dag_run(req, [rgn])
                od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                        g.val_offs, g.n_kv)
                assert (gr, gd) == (orows, od)
        finally:
            rgn.close()
    finally:
        g.close()


# ---- general RPN selection predicates ----------------------------------
def test_rpn_oracle_shapes():
    orc = _orc()
    rows = [{1: a, 2: b} for a, b in
            [(5, 3), (10, -4), (None, 7), (2, None), (-6, -6), (0, 0)]]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]

    def count_with(sel):
        req = (tikv_amd.DagSelect(cols).where(sel)
               .simple_agg([tikv_amd.count_star()]).build())
        data, nrows = orc.dag_run(req, k, ko, v, vo, n)
        return dec_int(data, 0)[0]

    # col0 + col1 > 0
    e = (tikv_amd.Expr().col(0).col(1).func(F.SIG_PLUS_INT)
         .const_int(0).func(F.SIG_GT_INT))
    assert count_with(e) == 2       # (5,3)=8, (10,-4)=6
    # NOT(col0 < col1)
    e = (tikv_amd.Expr().col(0).col(1).func(F.SIG_LT_INT)
         .func(F.SIG_UNARY_NOT, 1))
    assert count_with(e) == 4       # (5,3), (10,-4), (-6,-6), (0,0)
    # col0 IS NULL
    e = tikv_amd.Expr().col(0).func(F.SIG_INT_IS_NULL, 1)
    assert count_with(e) == 1
    # col0 > 0 OR col1 > 0
    e = (tikv_amd.Expr().col(0).const_int(0).func(F.SIG_GT_INT)
         .col(1).const_int(0).func(F.SIG_GT_INT).func(F.SIG_LOGICAL_OR))
    assert count_with(e) == 4       # (5,3), (10,-4), (None,7), (2,None)


@pytest.mark.gpu
def test_rpn_gpu_parity(engine):
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=90001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            sels = [
                # col3 + col7 > 0 (two columns, arithmetic)
                (tikv_amd.Expr().col(2).col(6).func(F.SIG_PLUS_INT)
                 .const_int(0).func(F.SIG_GT_INT)),
                # NOT(col3 < col7) (column-vs-column compare)
                (tikv_amd.Expr().col(2).col(6).func(F.SIG_LT_INT)
                 .func(F.SIG_UNARY_NOT, 1)),
                # range on ONE column through the RPN path: a < col3 AND
                # col3 < b (same column twice -> one capture slot)
                (tikv_amd.Expr()
                 .const_int(-600000000).col(2).func(F.SIG_LT_INT)
                 .col(2).const_int(600000000).func(F.SIG_LT_INT)
                 .func(F.SIG_LOGICAL_AND)),
                # col3 IS TRUE
                tikv_amd.Expr().col(2).func(F.SIG_INT_IS_TRUE, 1),
            ]
            for sel in sels:
                req = (tikv_amd.DagSelect(cols).where(sel)
                       .simple_agg([tikv_amd.count_star(),
                                    tikv_amd.sum_col(5)]).build())
                gd, gr, _ = engine.dag_run(req, [rgn])
                od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                        g.val_offs, g.n_kv)
                assert (gr, gd) == (orows, od)
            # project with an RPN predicate: both referenced columns in the
            # output, decoded
            sel = (tikv_amd.Expr().col(2).col(6).func(F.SIG_PLUS_INT)
                   .const_int(0).func(F.SIG_GT_INT))
            req = (tikv_amd.DagSelect(cols).where(sel)
                   .output([2, 6, 12]).build())
            gd, gr, _ = engine.dag_run(req, [rgn])
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert (gr, gd) == (orows, od)
        finally:
            rgn.close()
    finally:
        g.close()


def test_chunk_decimal_struct_golden():
    """40-byte decimal struct dump layout (decimal.rs:2135-2142 dump of the
    read_decimal result), hand-computed for 123.45 and -0.07."""
    orc = _orc()
    lib = orc.load_lib()
    # datum-encode 123.45 via the oracle codec test hook: scaled 12345,frac 2
    import ctypes as CT
    # build rows with decimal cells through the encode hook
    buf = CT.create_string_buffer(48)
    st = lib.orc_test_dec_from_i64_encode(123, buf)   # 123 as decimal
    assert st >= 0
    # instead pin via a full pipeline: single row {1: decimal}, project
    # hand-encode the datum: [prec=5][frac=2] comparable words
    # 123.45: int word 123 (1 leading word of 3 digits -> 2 bytes), frac 45
    # (2 trailing digits -> 1 byte); positive -> first byte |= 0x80
    enc = bytes([5, 2, 0x80, 123, 45])
    row = b"\x08" + var_i64(1) + b"\x06" + enc
    keys = row_key(0)
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * len(row)).from_buffer_copy(row)
    ko = (C.c_uint64 * 2)(0, len(keys))
    vo = (C.c_uint64 * 2)(0, len(row))
    cols = [tikv_amd.Col(1, tp=F.TP_NEWDECIMAL, decimal=2)]
    req = tikv_amd.DagSelect(cols).chunked().build()
    data, nrows = orc.dag_run(req, C.cast(kb, C.POINTER(C.c_uint8)), ko,
                              C.cast(vb, C.POINTER(C.c_uint8)), vo, 1)
    assert nrows == 1
    exp = bytearray()
    exp += (1).to_bytes(4, "little") + (0).to_bytes(4, "little")
    exp += bytes([3, 2, 2, 0])            # int_cnt, frac_cnt, result_frac, neg
    words = [123, 450000000] + [0] * 7    # 45 scaled to a full 9-digit word
    for w in words:
        exp += w.to_bytes(4, "little")
    assert data == bytes(exp)


def test_rpn_overflow_errors():
    """BIGINT overflow in a predicate errors the request (both sides)."""
    orc = _orc()
    rows = [{1: 2**62}]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1)]
    sel = (tikv_amd.Expr().col(0).const_int(2**62).func(F.SIG_PLUS_INT)
           .const_int(0).func(F.SIG_GT_INT))
    req = (tikv_amd.DagSelect(cols).where(sel)
           .simple_agg([tikv_amd.count_star()]).build())
    import pytest as _pytest
    with _pytest.raises(RuntimeError):
        orc.dag_run(req, k, ko, v, vo, n)


@pytest.mark.gpu
def test_rpn_overflow_errors_gpu(engine):
    rows = [{1: 2**62}]
    k, ko, v, vo, n, keep = region_of(rows)
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        cols = [tikv_amd.Col(1)]
        sel = (tikv_amd.Expr().col(0).const_int(2**62).func(F.SIG_PLUS_INT)
               .const_int(0).func(F.SIG_GT_INT))
        req = (tikv_amd.DagSelect(cols).where(sel)
               .simple_agg([tikv_amd.count_star()]).build())
        import pytest as _pytest
        with _pytest.raises(RuntimeError):
            engine.dag_run(req, [rgn])
    finally:
        rgn.close()


@pytest.mark.gpu
def test_first_over_hash_groups_gpu(engine):
    """FIRST in a FastHash (int group) request routes through the sorted
    pipeline; groups compare order-insensitively against the oracle."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=0, n_rows=60001, table_id=5)
    try:
        rgn = engine.region(g)
        try:
            cols = [tikv_amd.Col(i) for i in range(1, 5)]
            req = (tikv_amd.DagSelect(cols)
                   .hash_agg([tikv_amd.first_col(1), tikv_amd.count_star(),
                              tikv_amd.max_col(3)],
                             tikv_amd.Expr().col(0)).build())
            gd, gr, _ = engine.dag_run(req, [rgn])
            od, orows = orc.dag_run(req, g.keys, g.key_offs, g.vals,
                                    g.val_offs, g.n_kv)
            assert gr == orows
            assert set(split_rows(gd, 4)) == set(split_rows(od, 4))
        finally:
            rgn.close()
    finally:
        g.close()


# ---- Real (f64) aggregates ---------------------------------------------
import struct as _struct


def cell_real(col_id, x):
    bits = _struct.unpack("<Q", _struct.pack("<d", x))[0]
    u = (bits | (1 << 63)) if not (bits >> 63) else (~bits) & (2**64 - 1)
    return b"\x08" + var_i64(col_id) + b"\x05" + u.to_bytes(8, "big")


def region_real(vals_list):
    rows = []
    keys = b""
    vbytes = b""
    vo = [0]
    for i, x in enumerate(vals_list):
        keys += row_key(i)
        v = cell_int(1, i)
        v += cell_null(2) if x is None else cell_real(2, x)
        vbytes += v
        vo.append(len(vbytes))
    ko = [19 * i for i in range(len(vals_list) + 1)]
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * len(vbytes)).from_buffer_copy(vbytes)
    return (C.cast(kb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(ko))(*ko),
            C.cast(vb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(vo))(*vo),
            len(vals_list), (kb, vb))


def dec_uint(data, pos):
    assert data[pos] == 4
    return int.from_bytes(data[pos + 1:pos + 9], "big"), pos + 9


def dec_real(data, pos):
    assert data[pos] == 5
    u = int.from_bytes(data[pos + 1:pos + 9], "big")
    if u >> 63:
        u &= (1 << 63) - 1
    else:
        u = (~u) & (2**64 - 1)
    return _struct.unpack("<d", _struct.pack("<Q", u))[0], pos + 9


def test_real_aggs_oracle():
    orc = _orc()
    xs = [1.5, -2.25, None, 10.0, 0.5]
    k, ko, v, vo, n, keep = region_real(xs)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
    req = (tikv_amd.DagSelect(cols)
           .simple_agg([tikv_amd.sum_real(1), tikv_amd.avg_real(1),
                        tikv_amd.max_col(1, tp=F.TP_DOUBLE),
                        tikv_amd.min_col(1, tp=F.TP_DOUBLE),
                        tikv_amd.first_col(1, tp=F.TP_DOUBLE)]).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 1
    s, p = dec_real(data, 0)                      # sum
    cnt, p = dec_uint(data, p)                    # avg count (UNSIGNED)
    av, p = dec_real(data, p)                     # avg sum
    mx, p = dec_real(data, p)
    mn, p = dec_real(data, p)
    fv, p = dec_real(data, p)
    assert (s, cnt, av, mx, mn, fv) == (9.75, 4, 9.75, 10.0, -2.25, 1.5)
    assert p == len(data)


@pytest.mark.gpu
def test_real_aggs_gpu_parity(engine):
    import random
    rng = random.Random(7)
    xs = [None if rng.random() < 0.05 else rng.uniform(-1e6, 1e6)
          for _ in range(30000)]
    k, ko, v, vo, n, keep = region_real(xs)
    orc = _orc()
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
        sel = tikv_amd.cmp_col_const(0, F.SIG_GE_INT, 1000)
        req = (tikv_amd.DagSelect(cols).where(sel)
               .simple_agg([tikv_amd.count_col(1),
                            tikv_amd.sum_real(1),
                            tikv_amd.max_col(1, tp=F.TP_DOUBLE),
                            tikv_amd.min_col(1, tp=F.TP_DOUBLE),
                            tikv_amd.first_col(1, tp=F.TP_DOUBLE)]).build())
        gd, gr, _ = engine.dag_run(req, [rgn])
        od, orows = orc.dag_run(req, k, ko, v, vo, n)
        assert gr == orows == 1
        gc, p = dec_int(gd, 0)
        gs, p = dec_real(gd, p)
        gmx, p = dec_real(gd, p)
        gmn, p = dec_real(gd, p)
        gf, p = dec_real(gd, p)
        oc_, q = dec_int(od, 0)
        os_, q = dec_real(od, q)
        omx, q = dec_real(od, q)
        omn, q = dec_real(od, q)
        of_, q = dec_real(od, q)
        # count / max / min / first: bit-exact; sum: parallel order -> ULPs
        assert (gc, gmx, gmn, gf) == (oc_, omx, omn, of_)
        assert abs(gs - os_) <= 1e-9 * max(1.0, abs(os_))
    finally:
        rgn.close()


def test_real_aggs_grouped_oracle():
    """real sums through grouped (stream) aggregation."""
    orc = _orc()
    rows = []
    seq = [(1, 1.5), (1, 2.5), (2, -1.0), (2, None), (2, 4.0)]
    for i, (g_, x) in enumerate(seq):
        rows.append((g_, x))
    keys = b""
    vbytes = b""
    vo = [0]
    for i, (g_, x) in enumerate(rows):
        keys += row_key(i)
        v = cell_int(1, g_)
        v += cell_null(2) if x is None else cell_real(2, x)
        vbytes += v
        vo.append(len(vbytes))
    ko = [19 * i for i in range(len(rows) + 1)]
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * len(vbytes)).from_buffer_copy(vbytes)
    k = C.cast(kb, C.POINTER(C.c_uint8))
    v = C.cast(vb, C.POINTER(C.c_uint8))
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
    req = (tikv_amd.DagSelect(cols)
           .stream_agg([tikv_amd.sum_real(1),
                        tikv_amd.max_col(1, tp=F.TP_DOUBLE)],
                       tikv_amd.Expr().col(0)).build())
    data, nrows = orc.dag_run(req, k, (C.c_uint64 * len(ko))(*ko), v,
                              (C.c_uint64 * len(vo))(*vo), len(rows))
    assert nrows == 2
    s1, p = dec_real(data, 0)
    m1, p = dec_real(data, p)
    g1, p = dec_int(data, p)
    s2, p = dec_real(data, p)
    m2, p = dec_real(data, p)
    g2, p = dec_int(data, p)
    assert (s1, m1, g1) == (4.0, 2.5, 1)
    assert (s2, m2, g2) == (3.0, 4.0, 2)


@pytest.mark.gpu
def test_real_aggs_grouped_gpu(engine):
    import random
    rng = random.Random(11)
    rows = [(rng.randrange(8), rng.uniform(-100, 100)) for _ in range(20000)]
    keys = b""
    vbytes = b""
    vo = [0]
    for i, (g_, x) in enumerate(rows):
        keys += row_key(i)
        v = cell_int(1, g_) + cell_real(2, x)
        vbytes += v
        vo.append(len(vbytes))
    ko = [19 * i for i in range(len(rows) + 1)]
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * len(vbytes)).from_buffer_copy(vbytes)
    k = C.cast(kb, C.POINTER(C.c_uint8))
    v = C.cast(vb, C.POINTER(C.c_uint8))
    koa = (C.c_uint64 * len(ko))(*ko)
    voa = (C.c_uint64 * len(vo))(*vo)
    orc = _orc()
    rgn = engine.region_raw(k, koa, v, voa, len(rows))
    try:
        cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
        req = (tikv_amd.DagSelect(cols)
               .hash_agg([tikv_amd.count_star(), tikv_amd.sum_real(1),
                          tikv_amd.max_col(1, tp=F.TP_DOUBLE)],
                         tikv_amd.Expr().col(0)).build())
        gd, gr, _ = engine.dag_run(req, [rgn])
        od, orows = orc.dag_run(req, k, koa, v, voa, len(rows))
        assert gr == orows == 8

        def rows_of(data):
            out = {}
            p = 0
            for _ in range(8):
                cnt, p = dec_int(data, p)
                s, p = dec_real(data, p)
                m, p = dec_real(data, p)
                g_, p = dec_int(data, p)
                out[g_] = (cnt, s, m)
            return out

        gm, om = rows_of(gd), rows_of(od)
        assert set(gm) == set(om)
        for g_ in gm:
            assert gm[g_][0] == om[g_][0]
            assert gm[g_][2] == om[g_][2]
            assert abs(gm[g_][1] - om[g_][1]) <= 1e-9 * max(1.0, abs(om[g_][1]))
    finally:
        rgn.close()


def test_unique_index_value_handle_oracle():
    """old-format unique index: key holds only the column datums, the PK
    int handle is the value's plain BE u64 (index_scan_executor.rs:416-422
    -- NOT number::decode_i64, no sign flip)."""
    orc = _orc()
    # index keys: prefix 't'+tid+'_i'+idx + one INT datum; value = 8B handle
    keys = b""
    vals = b""
    ko = [0]
    vo = [0]
    handles = [7, -3, 2**40]
    for i, h in enumerate(handles):
        k = (b"t" + (9).to_bytes(8, "big") + b"_i" + (1).to_bytes(8, "big")
             + b"\x03" + (((100 + i) ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big"))
        keys += k
        ko.append(len(keys))
        vals += (h & (2**64 - 1)).to_bytes(8, "big")
        vo.append(len(vals))
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * len(vals)).from_buffer_copy(vals)
    cols = [tikv_amd.Col(0), tikv_amd.Col(-1, pk_handle=True)]
    req = (tikv_amd.DagSelect(cols, index=True).build())
    data, nrows = orc.dag_run(req, C.cast(kb, C.POINTER(C.c_uint8)),
                              (C.c_uint64 * len(ko))(*ko),
                              C.cast(vb, C.POINTER(C.c_uint8)),
                              (C.c_uint64 * len(vo))(*vo), len(handles))
    assert nrows == 3
    p = 0
    got = []
    for i in range(3):
        cv, p = dec_int(data, p)       # index column datum
        hv, p = dec_int(data, p)       # handle (decoded -> INT datum)
        got.append((cv, hv))
    assert got == [(100, 7), (101, -3), (102, 2**40)]
    assert p == len(data)


def test_chunked_grouped_outputs_oracle():
    """TypeChunk over grouped results: bytes group datums ride the var-size
    column container; drains split into 1024-row chunks."""
    orc = _orc()
    # 2500 distinct int groups -> chunk lengths 1024, 1024, 452
    rows = [{1: i, 2: 1} for i in range(2500)]
    k, ko, v, vo, n, keep = region_of(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
    req = (tikv_amd.DagSelect(cols)
           .stream_agg([tikv_amd.count_star()], tikv_amd.Expr().col(0))
           .chunked().build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 2500
    lens = []
    p = 0
    while p < len(data):
        # two fixed i64 columns per chunk (count, group), no nulls
        ln = int.from_bytes(data[p:p + 4], "little")
        lens.append(ln)
        p += (8 + 8 * ln) * 2
    assert lens == [1024, 1024, 452]
    # bytes group keys through the chunk encoder
    rows2 = [{1: 5, 2: b"aa"}, {1: 6, 2: b"aa"}, {1: 7, 2: None},
             {1: 8, 2: b"zz"}]
    k2, ko2, v2, vo2, n2, keep2 = region_of_mixed(rows2)
    cols2 = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_VARCHAR)]
    req2 = (tikv_amd.DagSelect(cols2)
            .hash_agg([tikv_amd.count_star(), tikv_amd.max_col(0)],
                      tikv_amd.Expr().col(1)).chunked().build())
    d2, nr2 = orc.dag_run(req2, k2, ko2, v2, vo2, n2)
    assert nr2 == 3
    got = decode_chunks(bytes(d2), ["i64", "i64", "bytes"])
    assert sorted(got, key=lambda t: (t[2] is None, t[2])) == [
        (2, 6, b"aa"), (1, 8, b"zz"), (1, 7, None)]


# ---- Real comparers (impl_compare.rs:66-160 Real path) -----------------
def test_real_filter_oracle():
    orc = _orc()
    xs = [1.5, -2.25, None, 10.0, 0.5, 3.0]
    k, ko, v, vo, n, keep = region_real(xs)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
    sel = (tikv_amd.Expr().col(1).const_real(1.0)
           .func(F.SIG_GT_REAL, 2))
    req = (tikv_amd.DagSelect(cols).where(sel)
           .simple_agg([tikv_amd.count_star()]).build())
    data, nrows = orc.dag_run(req, k, ko, v, vo, n)
    assert nrows == 1
    assert data[0] == 3 and int.from_bytes(data[1:9], "big") ^ (1 << 63) == 3


@pytest.mark.gpu
@pytest.mark.parametrize("sig,name", [
    (F.SIG_LT_REAL, "lt"), (F.SIG_LE_REAL, "le"), (F.SIG_GT_REAL, "gt"),
    (F.SIG_GE_REAL, "ge"), (F.SIG_EQ_REAL, "eq"), (F.SIG_NE_REAL, "ne")])
def test_real_filter_gpu_parity(engine, sig, name):
    """Real comparer fast path on device, vs the oracle, incl. NULLs and a
    sum over the surviving rows."""
    import random
    rng = random.Random(11)
    xs = [None if rng.random() < 0.05 else rng.uniform(-100.0, 100.0)
          for _ in range(40000)]
    xs[17] = 25.5            # exact-match row for EQ
    k, ko, v, vo, n, keep = region_real(xs)
    orc = _orc()
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
        sel = tikv_amd.Expr().col(1).const_real(25.5).func(sig, 2)
        req = (tikv_amd.DagSelect(cols).where(sel)
               .simple_agg([tikv_amd.count_star(),
                            tikv_amd.sum_real(1)]).build())
        gd, gr, _ = engine.dag_run(req, [rgn])
        od, orows = orc.dag_run(req, k, ko, v, vo, n)
        assert orows == gr == 1
        # count bit-exact; f64 sum within the 1-ULP-class budget
        assert od[0:9] == gd[0:9]
        os_, _ = dec_real(od, 9)
        gs_, _ = dec_real(gd, 9)
        assert os_ == gs_ or abs(os_ - gs_) <= 1e-9 * max(1.0, abs(os_))
    finally:
        rgn.close()


@pytest.mark.gpu
def test_real_filter_project_gpu(engine):
    """project mode with a Real predicate: the filter column outputs in
    DECODED datum form (lazy_column.rs:165,242)."""
    xs = [1.5, -2.25, None, 10.0, 0.5]
    k, ko, v, vo, n, keep = region_real(xs)
    orc = _orc()
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_DOUBLE)]
        sel = tikv_amd.Expr().col(1).const_real(0.0).func(F.SIG_GE_REAL, 2)
        req = tikv_amd.DagSelect(cols).where(sel).build()
        gd, gr, _ = engine.dag_run(req, [rgn])
        od, orows = orc.dag_run(req, k, ko, v, vo, n)
        assert orows == gr == 3
        assert od == gd
    finally:
        rgn.close()


# ---- selection over >2 distinct columns (selection_executor.rs:86) -----
def _four_col_region(n=40000):
    import random
    rng = random.Random(23)
    rows = []
    for i in range(n):
        rows.append(b"".join(
            cell_int(c, rng.randrange(-1000, 1000)) for c in range(1, 5)))
    keys = b"".join(row_key(i) for i in range(n))
    ko = [19 * i for i in range(n + 1)]
    vals = b"".join(rows)
    vo = [0]
    for r in rows:
        vo.append(vo[-1] + len(r))
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * len(vals)).from_buffer_copy(vals)
    return (C.cast(kb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(ko))(*ko),
            C.cast(vb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(vo))(*vo),
            n, (kb, vb))


def _four_col_req(n_conds):
    cols = [tikv_amd.Col(i) for i in range(1, 5)]
    sels = [tikv_amd.cmp_col_const(0, F.SIG_GT_INT, -500),
            tikv_amd.cmp_col_const(1, F.SIG_LT_INT, 500),
            tikv_amd.cmp_col_const(2, F.SIG_NE_INT, 0),
            tikv_amd.cmp_col_const(3, F.SIG_GE_INT, -900)][:n_conds]
    return (tikv_amd.DagSelect(cols).where(*sels)
            .simple_agg([tikv_amd.count_star(), tikv_amd.sum_col(0),
                         tikv_amd.max_col(3)]).build())


def test_four_col_selection_oracle():
    orc = _orc()
    k, ko, v, vo, n, keep = _four_col_region(5000)
    for nc in (3, 4):
        data, nrows = orc.dag_run(_four_col_req(nc), k, ko, v, vo, n)
        assert nrows == 1


@pytest.mark.gpu
@pytest.mark.parametrize("n_conds", [3, 4])
def test_four_col_selection_gpu_parity(engine, n_conds):
    """ANDed conjuncts over 3-4 DISTINCT columns ride the capture-only
    pseudo-agg channels (DAGG_XCAP)."""
    orc = _orc()
    k, ko, v, vo, n, keep = _four_col_region()
    req = _four_col_req(n_conds)
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        gd, gr, _ = engine.dag_run(req, [rgn])
    finally:
        rgn.close()
    assert orows == gr == 1
    assert od == gd


@pytest.mark.gpu
def test_four_col_selection_hash_gpu(engine):
    orc = _orc()
    k, ko, v, vo, n, keep = _four_col_region()
    cols = [tikv_amd.Col(i) for i in range(1, 5)]
    sels = [tikv_amd.cmp_col_const(1, F.SIG_GT_INT, -800),
            tikv_amd.cmp_col_const(2, F.SIG_LT_INT, 800),
            tikv_amd.cmp_col_const(3, F.SIG_NE_INT, 7)]
    # int-valued aggregates only: this file's split_rows has no decimal arm
    req = (tikv_amd.DagSelect(cols).where(*sels)
           .hash_agg([tikv_amd.count_star(), tikv_amd.max_col(1)],
                     tikv_amd.Expr().col(0)).build())
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        gd, gr, _ = engine.dag_run(req, [rgn])
    finally:
        rgn.close()
    assert orows == gr
    assert sorted(split_rows(od, 3)) == sorted(split_rows(gd, 3))
