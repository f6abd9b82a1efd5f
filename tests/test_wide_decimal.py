"""Wide Decimal sums: values of 19..38 digits (scaled i128) and sums past
i128 (256-bit accumulators), vs the oracle's full word_buf arithmetic
(decimal.rs:927-942; read_decimal :2204-2289; Summable do_add)."""
import ctypes as C
import importlib.util
import os

import pytest

import tikv_amd
from tikv_amd import _ffi as F

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
D2B = [0, 1, 1, 2, 2, 3, 3, 4, 4, 4]


def _orc():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def enc_decimal(scaled, frac):
    """binary decimal: [prec][frac] + sign-masked BE digit
    groups; prec = digit count of |scaled| (min frac+1)."""
    neg = scaled < 0
    mag = -scaled if neg else scaled
    digits = str(mag)
    if len(digits) <= frac:
        digits = "0" * (frac + 1 - len(digits)) + digits
    prec = len(digits)
    int_digits = digits[:prec - frac]
    frac_digits = digits[prec - frac:]
    int_cnt = prec - frac
    iw, ld = divmod(int_cnt, 9)
    fw, td = divmod(frac, 9)
    int_digits = int_digits.rjust(ld + iw * 9, "0")
    payload = bytearray()
    pos = 0
    if ld:
        payload += int(int_digits[:ld]).to_bytes(D2B[ld], "big")
        pos = ld
    for w in range(iw):
        payload += int(int_digits[pos:pos + 9]).to_bytes(4, "big")
        pos += 9
    pos = 0
    for w in range(fw):
        payload += int(frac_digits[pos:pos + 9]).to_bytes(4, "big")
        pos += 9
    if td:
        payload += int(frac_digits[pos:pos + td]).to_bytes(D2B[td], "big")
    payload[0] ^= 0x80
    if neg:
        payload = bytearray(b ^ 0xFF for b in payload)
    return bytes([6, prec, frac]) + bytes(payload)


def var_i64(v):
    u = (v << 1) if v >= 0 else (((-v) << 1) - 1)
    out = bytearray()
    while u >= 0x80:
        out.append(0x80 | (u & 0x7F))
        u >>= 7
    out.append(u)
    return bytes(out)


def cell(col_id, datum):
    return b"\x08" + var_i64(col_id) + datum


def cell_int(col_id, v):
    return cell(col_id, b"\x08" + var_i64(v))


def row_key(h):
    return (b"t" + ((1 ^ (1 << 63)).to_bytes(8, "big")) + b"_r" +
            ((h ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big"))


def region(rows):
    keys = b"".join(row_key(i) for i in range(len(rows)))
    ko = [19 * i for i in range(len(rows) + 1)]
    vals = b"".join(rows)
    vo = [0]
    for r in rows:
        vo.append(vo[-1] + len(r))
    kb = (C.c_uint8 * len(keys)).from_buffer_copy(keys)
    vb = (C.c_uint8 * max(len(vals), 1)).from_buffer_copy(vals or b"\0")
    return (C.cast(kb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(ko))(*ko),
            C.cast(vb, C.POINTER(C.c_uint8)), (C.c_uint64 * len(vo))(*vo),
            len(rows), (kb, vb))


# cell fracs <= the sum's target frac (the engine scales values UP to the
# output type's frac; a cell with more fraction digits than the target is a
# loud error by design)
WIDE_VALS = [
    (10**37 - 1, 4),            # 37 digits
    (-(10**30 + 12345), 2),
    (98765432109876543210987654321, 4),   # 29 digits
    (5, 2),                     # tiny narrow among wide
    (-(10**19), 0),
]


def build_req(group=False):
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=4)]
    aggs = [tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=4)]
    if group:
        return tikv_amd.DagSelect(cols).hash_agg(
            aggs, tikv_amd.Expr().col(0)).build()
    return tikv_amd.DagSelect(cols).simple_agg(aggs).build()


def make_rows(n=2000):
    rows = []
    for i in range(n):
        sc, fr = WIDE_VALS[i % len(WIDE_VALS)]
        rows.append(cell_int(1, i % 4) + cell(2, enc_decimal(sc, fr)))
    return rows


def test_oracle_wide_sum():
    orc = _orc()
    k, ko, v, vo, n, keep = region(make_rows(50))
    data, nrows = orc.dag_run(build_req(), k, ko, v, vo, n)
    assert nrows == 1
    # the count datum then the decimal sum; the exact value is checked in
    # the GPU parity test — here assert the oracle accepted wide inputs
    assert data[0] == 3


@pytest.mark.gpu
def test_wide_decimal_simple_parity(engine):
    orc = _orc()
    k, ko, v, vo, n, keep = region(make_rows(30000))
    req = build_req()
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        gd, gr, _ = engine.dag_run(req, [rgn])
    finally:
        rgn.close()
    assert orows == gr == 1
    assert od == gd


@pytest.mark.gpu
def test_wide_decimal_hash_parity(engine):
    orc = _orc()
    k, ko, v, vo, n, keep = region(make_rows(30000))
    req = build_req(group=True)
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        gd, gr, _ = engine.dag_run(req, [rgn])
    finally:
        rgn.close()
    assert orows == gr == 4
    # order-insensitive: count, sum, group datums per row
    def rows_of(d):
        out, i = [], 0
        while i < len(d):
            row = bytearray()
            for _ in range(3):
                if d[i] == 6:
                    prec, frac = d[i + 1], d[i + 2]
                    ic = prec - frac
                    ln = 3 + (ic // 9) * 4 + D2B[ic % 9] + \
                        (frac // 9) * 4 + D2B[frac % 9]
                elif d[i] in (3, 4):
                    ln = 9
                elif d[i] == 8:
                    ln = 2
                    while d[i + ln - 1] & 0x80:
                        ln += 1
                else:
                    raise AssertionError(d[i])
                row += d[i:i + ln]
                i += ln
            out.append(bytes(row))
        return sorted(out)
    assert rows_of(od) == rows_of(gd)


@pytest.mark.gpu
def test_wide_decimal_sum_past_i128(engine):
    """sums that exceed i128 exercise the 256-bit ext limbs end to end."""
    orc = _orc()
    rows = [cell_int(1, 0) + cell(2, enc_decimal(10**37, 0))
            for _ in range(400)]        # total 4e39 > 2^127
    k, ko, v, vo, n, keep = region(rows)
    cols = [tikv_amd.Col(1), tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=0)]
    req = tikv_amd.DagSelect(cols).simple_agg(
        [tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=0)]).build()
    od, orows = orc.dag_run(req, k, ko, v, vo, n)
    rgn = engine.region_raw(k, ko, v, vo, n)
    try:
        gd, gr, _ = engine.dag_run(req, [rgn])
    finally:
        rgn.close()
    assert orows == gr == 1
    assert od == gd
