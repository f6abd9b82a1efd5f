"""MVCC write-CF version-filter semantics (forward.rs:440-515,
write.rs:296-361,425-442; types.rs:152-161,721-731).

CPU tests pin the oracle's restatement on hand-built write-CF entries;
the GPU test compares the device filter's materialized region bit-for-bit
against the oracle over generated multi-version regions."""
import ctypes as C
import importlib.util
import os

import pytest

import tikv_amd

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _orc():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


# ---------- python builders ----------
def memcmp_enc(b):
    out = bytearray()
    full = len(b) // 8
    for g in range(full):
        out += b[g * 8:(g + 1) * 8]
        out.append(0xFF)
    rem = b[full * 8:]
    out += rem + bytes(8 - len(rem))
    out.append(0xFF - (8 - len(rem)))
    return bytes(out)


def var_u64(v):
    out = bytearray()
    while v >= 0x80:
        out.append(0x80 | (v & 0x7F))
        v >>= 7
    out.append(v)
    return bytes(out)


def wkey(user_raw, commit_ts):
    return memcmp_enc(user_raw) + ((~commit_ts) & (2**64 - 1)).to_bytes(8, "big")


def wval(tp, start_ts, short_value=None, gc_fence=None, last_change=None):
    out = bytearray(tp.encode())
    out += var_u64(start_ts)
    if short_value is not None:
        out += b"v" + bytes([len(short_value)]) + short_value
    if gc_fence is not None:
        out += b"F" + gc_fence.to_bytes(8, "big")
    if last_change is not None:
        ts, vers = last_change
        out += b"l" + ts.to_bytes(8, "big") + var_u64(vers)
    return bytes(out)


def make_arrays(entries):
    keys = b"".join(k for k, _ in entries)
    vals = b"".join(v for _, v in entries)
    ko = [0]
    vo = [0]
    for k, v in entries:
        ko.append(ko[-1] + len(k))
        vo.append(vo[-1] + len(v))
    kb = (C.c_uint8 * max(len(keys), 1)).from_buffer_copy(keys or b"\0")
    vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
    koa = (C.c_uint64 * len(ko))(*ko)
    voa = (C.c_uint64 * len(vo))(*vo)
    return (C.cast(kb, C.POINTER(C.c_uint8)), koa,
            C.cast(vb, C.POINTER(C.c_uint8)), voa, len(entries), (kb, vb))


UK_A = b"t" + bytes(7) + b"\x01_rAAAAAAAA"   # 19-byte-ish raw keys
UK_B = b"t" + bytes(7) + b"\x01_rBBBBBBBB"


def run_filter(entries, read_ts):
    orc = _orc()
    k, ko, v, vo, n, keep = make_arrays(entries)
    return orc.mvcc_filter(k, ko, v, vo, n, read_ts)


def test_simple_put_visible():
    e = [(wkey(UK_A, 100), wval("P", 99, b"rowA"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 1
    assert keys == UK_A and vals == b"rowA"


def test_newer_than_read_ts_skipped():
    e = [(wkey(UK_A, 2000), wval("P", 1999, b"new")),
         (wkey(UK_A, 100), wval("P", 99, b"old"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 1 and vals == b"old"


def test_delete_hides_key():
    e = [(wkey(UK_A, 500), wval("D", 499)),
         (wkey(UK_A, 100), wval("P", 99, b"old")),
         (wkey(UK_B, 100), wval("P", 99, b"b"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 1 and vals == b"b" and keys == UK_B


def test_rollback_and_lock_skip_to_older():
    e = [(wkey(UK_A, 800), wval("R", 799, b"p")),     # protected rollback
         (wkey(UK_A, 700), wval("L", 699)),
         (wkey(UK_A, 100), wval("P", 99, b"old"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 1 and vals == b"old"


def test_last_change_not_exist():
    e = [(wkey(UK_A, 800), wval("L", 799, last_change=(0, 1))),
         (wkey(UK_A, 100), wval("P", 99, b"old"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 0  # LastChange::NotExist -> key invisible


def test_gc_fence_invalidates():
    e = [(wkey(UK_A, 800), wval("P", 799, b"x", gc_fence=500)),
         (wkey(UK_A, 100), wval("P", 99, b"old"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 0  # fence in (0, read_ts] -> invisible (write.rs:425-442)
    # fence beyond read_ts stays valid
    e2 = [(wkey(UK_A, 800), wval("P", 799, b"x", gc_fence=5000))]
    keys, ko, vals, vo, n = run_filter(e2, 1000)
    assert n == 1 and vals == b"x"


def test_read_ts_inclusive():
    e = [(wkey(UK_A, 1000), wval("P", 999, b"edge"))]
    keys, ko, vals, vo, n = run_filter(e, 1000)
    assert n == 1 and vals == b"edge"


def test_generated_mvcc_region_oracle():
    """generator row_format=3: the filtered stream scans + counts cleanly."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=2000, table_id=1, row_format=3)
    try:
        orc = _orc()
        keys, ko, vals, vo, n = orc.mvcc_filter(
            g.keys, g.key_offs, g.vals, g.val_offs, g.n_kv, 1000)
        assert 0 < n <= 2000
        # every visible key is a 19-byte record key; values parse as rows:
        # run the oracle scan pipeline over the filtered arrays
        from tikv_amd import _ffi as F
        cols = [tikv_amd.Col(i) for i in range(1, 17)]
        req = tikv_amd.DagSelect(cols).simple_agg([tikv_amd.count_star()]).build()
        kb = (C.c_uint8 * max(len(keys), 1)).from_buffer_copy(keys or b"\0")
        vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
        koa = (C.c_uint64 * len(ko))(*ko)
        voa = (C.c_uint64 * len(vo))(*vo)
        data, rows = orc.dag_run(req, C.cast(kb, C.POINTER(C.c_uint8)), koa,
                                 C.cast(vb, C.POINTER(C.c_uint8)), voa, n)
        assert rows == 1
        cnt = int.from_bytes(data[1:9], "big") ^ (1 << 63)
        assert cnt == n
    finally:
        g.close()


@pytest.mark.gpu
def test_mvcc_device_filter_parity(engine):
    """device version filter vs oracle, bit-for-bit, then a count query."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=60000, table_id=1, row_format=3)
    try:
        orc = _orc()
        o_keys, o_ko, o_vals, o_vo, o_n = orc.mvcc_filter(
            g.keys, g.key_offs, g.vals, g.val_offs, g.n_kv, 1000)
        rgn = engine.region_mvcc(g, 1000)
        try:
            d_keys, d_ko, d_vals, d_vo, d_n = engine.dump_region(rgn)
            assert d_n == o_n
            assert d_keys == o_keys and list(d_ko) == list(o_ko)
            assert d_vals == o_vals and list(d_vo) == list(o_vo)
            # the filtered region is a first-class scan source
            from tikv_amd import _ffi as F
            cols = [tikv_amd.Col(i) for i in range(1, 17)]
            sel = tikv_amd.cmp_col_const(3, F.SIG_LT_INT, 0)
            req = (tikv_amd.DagSelect(cols).where(sel)
                   .simple_agg([tikv_amd.count_star()]).build())
            g_data, g_rows, _ = engine.dag_run(req, [rgn])
            kb = (C.c_uint8 * max(len(o_keys), 1)).from_buffer_copy(o_keys or b"\0")
            vb = (C.c_uint8 * max(len(o_vals), 1)).from_buffer_copy(o_vals or b"\0")
            koa = (C.c_uint64 * len(o_ko))(*o_ko)
            voa = (C.c_uint64 * len(o_vo))(*o_vo)
            o_data, o_rows = orc.dag_run(req, C.cast(kb, C.POINTER(C.c_uint8)),
                                         koa, C.cast(vb, C.POINTER(C.c_uint8)),
                                         voa, o_n)
            assert o_rows == g_rows == 1
            assert o_data == g_data
        finally:
            rgn.close()
    finally:
        g.close()


def _py_visible(entries_by_key, read_ts):
    """Independent python model of forward.rs:440-515 + write.rs:425-442:
    newest version with commit_ts <= read_ts decides; Put emits unless its
    gc_fence is in (0, read_ts]; Delete hides; Lock/Rollback defer to the
    next older version, except LastChange::NotExist which hides."""
    out = {}
    for uk, versions in entries_by_key.items():
        for commit_ts, (tp, short, fence, last_change) in sorted(
                versions, reverse=True):
            if commit_ts > read_ts:
                continue
            if tp == "P":
                if fence is not None and 0 < fence <= read_ts:
                    break
                out[uk] = short
                break
            if tp == "D":
                break
            # Lock / Rollback
            if last_change is not None and last_change[0] == 0:
                break  # LastChange::NotExist
            continue
        # fallthrough: no visible version
    return out


def test_mvcc_random_chains_vs_oracle():
    """random multi-version chains: oracle == independent python model."""
    import random
    rng = random.Random(42)
    for trial in range(20):
        entries_by_key = {}
        flat = []
        n_keys = rng.randrange(1, 12)
        for ki in range(n_keys):
            uk = b"t" + bytes(7) + bytes([1]) + b"_r" + ki.to_bytes(8, "big")
            n_vers = rng.randrange(1, 5)
            used_ts = rng.sample(range(10, 3000), n_vers)
            versions = []
            for ts in used_ts:
                tp = rng.choice(["P", "P", "P", "D", "L", "R"])
                short = (b"v%d" % ts) if tp in ("P", "R") else None
                if tp == "R":
                    short = b"p"   # protected rollback payload
                fence = None
                if tp == "P" and rng.random() < 0.3:
                    fence = rng.choice([0, 400, 5000])
                    if fence == 0:
                        fence = None
                last_change = None
                if tp in ("L", "R") and rng.random() < 0.3:
                    last_change = (0, 1) if rng.random() < 0.5 else (5, 2)
                versions.append((ts, (tp, short if tp == "P" else None,
                                      fence, last_change)))
                flat.append((uk, ts, tp, short, fence, last_change))
            entries_by_key[uk] = versions
        # write-CF stream: key asc, commit_ts desc
        flat.sort(key=lambda e: (e[0], -e[1]))
        ents = []
        for uk, ts, tp, short, fence, last_change in flat:
            ents.append((wkey(uk, ts),
                         wval(tp, ts - 1, short_value=short, gc_fence=fence,
                              last_change=last_change)))
        read_ts = rng.choice([5, 500, 1500, 3500])
        keys, ko, vals, vo, n = run_filter(ents, read_ts)
        got = {}
        for i in range(n):
            got[keys[ko[i]:ko[i + 1]]] = vals[vo[i]:vo[i + 1]]
        want = _py_visible(entries_by_key, read_ts)
        assert got == want, (trial, read_ts)


# ---------- default-CF lookup (forward.rs:433-515; write.rs:296) ----------
def dkey(user_raw, start_ts):
    """default-CF key = memcomparable(user_key) || BE(~start_ts) — the same
    append_ts encoding as write keys but with the START ts."""
    return memcmp_enc(user_raw) + ((~start_ts) & (2**64 - 1)).to_bytes(8, "big")


def test_default_cf_lookup_oracle():
    long_a = b"A" * 300            # >255 B: cannot be a short value
    long_b = b"B" * 400
    wents = [(wkey(UK_A, 100), wval("P", 99)),          # no short value
             (wkey(UK_B, 200), wval("P", 150))]
    dents = [(dkey(UK_A, 99), long_a), (dkey(UK_B, 150), long_b)]
    wents.sort(key=lambda e: e[0])
    dents.sort(key=lambda e: e[0])
    orc = _orc()
    k, ko, v, vo, n, keep = make_arrays(wents)
    dk, dko, dv, dvo, dn, keep2 = make_arrays(dents)
    keys, kof, vals, vof, nv = orc.mvcc_filter(
        k, ko, v, vo, n, 1000, default_cf=(dk, dko, dv, dvo, dn))
    assert nv == 2
    got = {keys[kof[i]:kof[i + 1]]: vals[vof[i]:vof[i + 1]] for i in range(nv)}
    assert got == {UK_A: long_a, UK_B: long_b}

    # mixed: one short value, one default-CF value
    wents = [(wkey(UK_A, 100), wval("P", 99)),
             (wkey(UK_B, 200), wval("P", 150, b"short"))]
    wents.sort(key=lambda e: e[0])
    k, ko, v, vo, n, keep = make_arrays(wents)
    keys, kof, vals, vof, nv = orc.mvcc_filter(
        k, ko, v, vo, n, 1000, default_cf=(dk, dko, dv, dvo, dn))
    got = {keys[kof[i]:kof[i + 1]]: vals[vof[i]:vof[i + 1]] for i in range(nv)}
    assert got == {UK_A: long_a, UK_B: b"short"}

    # missing default entry = corruption (DEFAULT_NOT_FOUND), loud
    wents = [(wkey(UK_A, 100), wval("P", 77))]          # start_ts 77 absent
    k, ko, v, vo, n, keep = make_arrays(wents)
    import pytest
    with pytest.raises(RuntimeError):
        orc.mvcc_filter(k, ko, v, vo, n, 1000,
                        default_cf=(dk, dko, dv, dvo, dn))
    # and with NO default stream it stays loudly unsupported
    with pytest.raises(RuntimeError):
        orc.mvcc_filter(k, ko, v, vo, n, 1000)


@pytest.mark.gpu
def test_default_cf_lookup_device(engine):
    """device MVCC filter with default-CF stream == oracle, and the
    resulting region (long row-v1 values) scans correctly end to end."""
    import tikv_amd

    def zig(v):
        return (v << 1) if v >= 0 else (((-v) << 1) - 1)

    def row_v1(cells):
        """[VAR_INT flag, zigzag col_id, VAR_INT flag, zigzag value]* — the
        encode_row layout (codec/table.rs:166-184)."""
        out = bytearray()
        for cid, val in cells:
            out.append(8)
            out += var_u64(zig(cid))
            out.append(8)
            out += var_u64(zig(val))
        return bytes(out)

    rng = __import__("random").Random(7)
    wents, dents = [], []
    expect_count = 0
    for h in range(2000):
        uk = b"t" + bytes(7) + bytes([1]) + b"_r" + h.to_bytes(8, "big")
        val = row_v1([(1, h), (2, rng.randrange(0, 100))])
        # pad with a big bytes cell so the value exceeds 255 B for half
        if h % 2 == 0:
            pad = bytes([97 + (h % 26)]) * 300
            val += bytes([8]) + var_u64(zig(3)) + bytes([2]) + \
                var_u64(zig(len(pad))) + pad
            wents.append((wkey(uk, 100 + h), wval("P", 99)))
            dents.append((dkey(uk, 99), val))
        else:
            wents.append((wkey(uk, 100 + h), wval("P", 99, val)))
        expect_count += 1
    wents.sort(key=lambda e: e[0])
    dents.sort(key=lambda e: e[0])
    k, ko, v, vo, n, keep = make_arrays(wents)
    dk, dko, dv, dvo, dn, keep2 = make_arrays(dents)
    rgn = engine.region_mvcc_with_default(k, ko, v, vo, n,
                                          dk, dko, dv, dvo, dn, 1 << 40)
    try:
        from tikv_amd import _ffi as F
        cols = [tikv_amd.Col(1), tikv_amd.Col(2)]
        sel = tikv_amd.cmp_col_const(1, F.SIG_GE_INT, 0)
        req = (tikv_amd.DagSelect(cols).where(sel)
               .simple_agg([tikv_amd.count_star(), tikv_amd.sum_col(0)]).build())
        g_data, g_n, _ = engine.dag_run(req, [rgn])
    finally:
        rgn.close()
    # oracle over the same streams
    orc = _orc()
    okeys, okof, ovals, ovof, onv = orc.mvcc_filter(
        k, ko, v, vo, n, 1 << 40, default_cf=(dk, dko, dv, dvo, dn))
    assert onv == expect_count
    o_data, o_n = orc.dag_run(req, C.cast(
        (C.c_uint8 * len(okeys)).from_buffer_copy(okeys), C.POINTER(C.c_uint8)),
        (C.c_uint64 * len(okof))(*okof),
        C.cast((C.c_uint8 * len(ovals)).from_buffer_copy(ovals),
               C.POINTER(C.c_uint8)),
        (C.c_uint64 * len(ovof))(*ovof), onv)
    assert o_n == g_n == 1
    assert o_data == g_data
