"""Oracle codecs vs the reference's golden vectors (CPU)."""
import ctypes as C

from golden import vectors as GV
import importlib.util
import os

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _orc():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def test_crc64_kat():
    lib = _orc().load_lib()
    data, expect = GV.CRC64_CHECK
    assert lib.orc_crc64_xz(data, len(data)) == expect
    # incremental == one-shot on a longer buffer
    buf = bytes(range(256)) * 5
    assert lib.orc_crc64_xz(buf, len(buf)) == lib.orc_crc64_xz(buf, len(buf))


def test_memcmp_golden():
    lib = _orc().load_lib()
    for src, asc, desc in GV.MEMCMP_CASES:
        out = C.create_string_buffer(80)
        n = lib.orc_test_memcmp_encode(src, len(src), 0, out)
        assert out.raw[:n] == asc, src
        n = lib.orc_test_memcmp_encode(src, len(src), 1, out)
        assert out.raw[:n] == desc, src
        dec = C.create_string_buffer(80)
        dlen = C.c_uint64()
        c = lib.orc_test_memcmp_decode(asc, len(asc), dec, C.byref(dlen))
        assert c == len(asc) and dec.raw[:dlen.value] == src


def test_varint_roundtrip():
    lib = _orc().load_lib()
    for v in [0, 1, -1, 2, -2, 63, 64, -64, -65, 127, -128, 300, -300,
              2**31, -(2**31), 2**62, -(2**62), 2**63 - 1, -(2**63)]:
        out = C.create_string_buffer(10)
        n = lib.orc_test_var_i64_encode(v, out)
        dv = C.c_int64()
        c = C.c_uint64()
        assert lib.orc_test_var_i64_decode(out.raw, n, C.byref(dv), C.byref(c)) == 0
        assert dv.value == v and c.value == n


def test_row_key():
    lib = _orc().load_lib()
    for h in [-(2**63), 2**63 - 1, -1, 0, 2, 3, 1024]:  # table.rs:751-757
        k = C.create_string_buffer(19)
        lib.orc_test_row_key(1, h, k)
        hv = C.c_int64()
        assert lib.orc_test_int_handle(k.raw, 19, C.byref(hv)) == 0
        assert hv.value == h
    # layout: 't' + BE(1^sign) + '_r' (table.rs:26-34)
    k = C.create_string_buffer(19)
    lib.orc_test_row_key(1, 2, k)
    assert k.raw[:11] == b"t" + bytes([0x80, 0, 0, 0, 0, 0, 0, 1]) + b"_r"
    assert k.raw[11:19] == bytes([0x80, 0, 0, 0, 0, 0, 0, 2])


def _int_datum(x):
    return bytes([3]) + (((x ^ (1 << 63)) & (2**64 - 1)).to_bytes(8, "big"))


def test_row_v2_golden():
    lib = _orc().load_lib()
    out = C.create_string_buffer(80)
    isn = C.c_int()
    LL, UNS = 8, 1 << 5
    v = GV.ROW_V2_UNSIGNED
    n = lib.orc_test_row_v2_col(v, len(v), 1, LL, UNS, out, C.byref(isn))
    assert n == 9 and out.raw[:9] == bytes([4]) + b"\xff" * 8
    n = lib.orc_test_row_v2_col(v, len(v), 2, LL, 0, out, C.byref(isn))
    assert n == 9 and out.raw[:9] == _int_datum(-1)
    v = GV.ROW_V2_BIG
    for cid, exp in [(1, 1000), (12, 2), (3, 3), (8, 32767)]:
        n = lib.orc_test_row_v2_col(v, len(v), cid, LL, 0, out, C.byref(isn))
        assert n == 9 and out.raw[:9] == _int_datum(exp), cid
    n = lib.orc_test_row_v2_col(v, len(v), 335, LL, 0, out, C.byref(isn))
    assert n == 0 and isn.value == 1
    n = lib.orc_test_row_v2_col(v, len(v), 99, LL, 0, out, C.byref(isn))
    assert n == 0 and isn.value == 0  # absent entirely


def test_decimal_add_small():
    lib = _orc().load_lib()
    out = C.create_string_buffer(64)
    b2 = C.create_string_buffer(64)
    cases = [(1, 1), (0, 5), (-3, 3), (999999999, 1), (10**17, 10**17),
             (-(10**15), 7), (123456789, -987654321)]
    for a, b in cases:
        na = lib.orc_test_dec_from_i64_encode(a, out)
        ea = out.raw[:na]
        nb = lib.orc_test_dec_from_i64_encode(b, b2)
        eb = b2.raw[:nb]
        s = C.create_string_buffer(64)
        ns = lib.orc_test_dec_add_encode(ea, na, eb, nb, s)
        ne = lib.orc_test_dec_from_i64_encode(a + b, out)
        assert s.raw[:ns] == out.raw[:ne], (a, b)
