"""Whole-SST ingestion (the file layer).

BlockBasedTable file = data blocks + metaindex + index block + footer
(RocksDB format.cc public format; TiKV reads SSTs via rust-rocksdb,
engine_iterator.rs:12). Every block carries a 5-byte trailer
[compression u8][checksum u32le]; checksum_type 1 = RocksDB-masked
crc32c(contents || compression byte). The fixture writer here builds
real files (format_version 2, kBinarySearch index, restart_interval 1),
and the tests pin the oracle walk and the engine ingestion against the
same underlying KV stream, including corrupt-file rejection.
"""
import ctypes as C
import importlib.util
import os

import pytest

import tikv_amd

from test_blocks import hand_block, varint32, as_bufs

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _orc():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


# ---- crc32c (Castagnoli, reflected) + RocksDB mask --------------------
_CRC_TAB = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ (0x82F63B78 if _c & 1 else 0)
    _CRC_TAB.append(_c)


def crc32c(data):
    c = 0xFFFFFFFF
    for b in data:
        c = _CRC_TAB[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def crc_mask(c):
    return (((c >> 15) | (c << 17)) + 0xa282ead8) & 0xFFFFFFFF


def varint64(v):
    out = bytearray()
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    return bytes(out)


def _compress_payload(block, ctype):
    """RocksDB on-disk payload for a compressed block (format_version 2
    framing: varint32 raw size + compressed bytes), via the engine's
    host compression helper."""
    lib = tikv_amd._ffi.load_lib()
    bb, oo, n, keep = as_bufs([block])
    cb = C.POINTER(C.c_uint8)()
    co = C.POINTER(C.c_uint64)()
    st = lib.copr_blocks_compress(bb, oo, 1, C.c_uint8(ctype),
                                  C.byref(cb), C.byref(co))
    if st != 0:
        raise RuntimeError("copr_blocks_compress: %d" % st)
    return C.string_at(cb, co[1])


def write_sst(data_blocks, compression=0, checksum_type=1, version=2,
              magic=0x88E241B785F4CFF7, corrupt_block_byte=None):
    """Build one BlockBasedTable file from uncompressed data-block bytes."""
    out = bytearray()
    handles = []
    for blk in data_blocks:
        payload = blk if compression == 0 else \
            _compress_payload(blk, compression)
        off = len(out)
        handles.append((off, len(payload)))
        out += payload
        out.append(compression)
        out += crc_mask(crc32c(payload + bytes([compression]))) \
            .to_bytes(4, "little")
    # metaindex: empty block (restart array only) -- never consulted
    m_off = len(out)
    meta = (0).to_bytes(4, "little") + (1).to_bytes(4, "little")
    out += meta
    out.append(0)
    out += crc_mask(crc32c(meta + b"\0")).to_bytes(4, "little")
    # index block: restart_interval 1, keys = ascending separators,
    # values = plain BlockHandles
    idx = bytearray()
    restarts = []
    for i, (off, sz) in enumerate(handles):
        key = b"idx%08d" % i + (0).to_bytes(8, "little")
        val = varint64(off) + varint64(sz)
        restarts.append(len(idx))
        idx += varint32(0) + varint32(len(key)) + varint32(len(val))
        idx += key + val
    for rr in restarts:
        idx += rr.to_bytes(4, "little")
    idx += len(restarts).to_bytes(4, "little")
    i_off = len(out)
    out += idx
    out.append(0)
    out += crc_mask(crc32c(bytes(idx) + b"\0")).to_bytes(4, "little")
    # footer (new form, 53 bytes)
    hb = (varint64(m_off) + varint64(len(meta)) +
          varint64(i_off) + varint64(len(idx)))
    footer = bytes([checksum_type]) + hb + b"\0" * (40 - len(hb))
    footer += version.to_bytes(4, "little") + magic.to_bytes(8, "little")
    out += footer
    if corrupt_block_byte is not None:
        out[corrupt_block_byte] ^= 0xFF
    return bytes(out)


def two_block_kvs():
    e1 = [(b"tabc_r0001", b"v1"), (b"tabc_r0002", b"value-two"),
          (b"tabc_r0003", b"")]
    e2 = [(b"tabd_r0001", b"x" * 40), (b"tabd_r0002", b"yy")]
    blocks = [hand_block(e1, 2), hand_block(e2, 1)]
    kvs = [(k, v) for e in (e1, e2) for (k, v) in e]
    return blocks, kvs


def test_crc32c_known_answer():
    """pins the fixture's crc32c (and, via every round-trip test below,
    the engine's and the oracle's) to the published Castagnoli KAT."""
    assert crc32c(b"123456789") == 0xE3069283


# ---- CPU: oracle walk -------------------------------------------------
def test_sst_oracle_roundtrip():
    orc = _orc()
    blocks, kvs = two_block_kvs()
    sst = write_sst(blocks)
    keys, ko, vals, vo, n = orc.sst_parse(sst)
    assert n == len(kvs)
    got = [(keys[ko[i]:ko[i + 1]], vals[vo[i]:vo[i + 1]]) for i in range(n)]
    assert got == kvs


@pytest.mark.parametrize("ctype", [4, 7])   # LZ4, ZSTD
def test_sst_oracle_compressed(ctype):
    orc = _orc()
    blocks, kvs = two_block_kvs()
    sst = write_sst(blocks, compression=ctype)
    keys, ko, vals, vo, n = orc.sst_parse(sst)
    assert n == len(kvs)
    got = [(keys[ko[i]:ko[i + 1]], vals[vo[i]:vo[i + 1]]) for i in range(n)]
    assert got == kvs


def test_sst_oracle_mixed_compression():
    """per-block compression types vary within one file (RocksDB picks
    per level); the walk honours each block's own trailer byte."""
    orc = _orc()
    blocks, kvs = two_block_kvs()
    out = bytearray()
    handles = []
    for i, blk in enumerate(blocks):
        ctype = (0, 7)[i % 2]
        payload = blk if ctype == 0 else _compress_payload(blk, ctype)
        handles.append((len(out), len(payload)))
        out += payload
        out.append(ctype)
        out += crc_mask(crc32c(payload + bytes([ctype]))).to_bytes(4, "little")
    # reuse write_sst's meta/index/footer by rebuilding around the data:
    # simplest correct path -- write a fresh file with the same blocks but
    # patch in our mixed payloads is equivalent to building it directly
    m_off = len(out)
    meta = (0).to_bytes(4, "little") + (1).to_bytes(4, "little")
    out += meta + b"\0" + crc_mask(crc32c(meta + b"\0")).to_bytes(4, "little")
    idx = bytearray()
    restarts = []
    for i, (off, sz) in enumerate(handles):
        key = b"idx%08d" % i + (0).to_bytes(8, "little")
        val = varint64(off) + varint64(sz)
        restarts.append(len(idx))
        idx += varint32(0) + varint32(len(key)) + varint32(len(val))
        idx += key + val
    for rr in restarts:
        idx += rr.to_bytes(4, "little")
    idx += len(restarts).to_bytes(4, "little")
    i_off = len(out)
    out += idx + b"\0" + crc_mask(crc32c(bytes(idx) + b"\0")) \
        .to_bytes(4, "little")
    hb = (varint64(m_off) + varint64(len(meta)) +
          varint64(i_off) + varint64(len(idx)))
    out += bytes([1]) + hb + b"\0" * (40 - len(hb))
    out += (2).to_bytes(4, "little")
    out += (0x88E241B785F4CFF7).to_bytes(8, "little")
    keys, ko, vals, vo, n = orc.sst_parse(bytes(out))
    assert n == len(kvs)
    got = [(keys[ko[i]:ko[i + 1]], vals[vo[i]:vo[i + 1]]) for i in range(n)]
    assert got == kvs


def test_sst_oracle_rejects():
    orc = _orc()
    blocks, kvs = two_block_kvs()
    # corrupt first data block byte -> checksum mismatch
    with pytest.raises(RuntimeError):
        orc.sst_parse(write_sst(blocks, corrupt_block_byte=1))
    # bad magic
    with pytest.raises(RuntimeError):
        orc.sst_parse(write_sst(blocks, magic=0x1122334455667788))
    # legacy / future footers are unsupported, loudly
    with pytest.raises(RuntimeError):
        orc.sst_parse(write_sst(blocks, version=0))
    with pytest.raises(RuntimeError):
        orc.sst_parse(write_sst(blocks, version=6))
    # truncated file
    with pytest.raises(RuntimeError):
        orc.sst_parse(write_sst(blocks)[:40])
    # checksum type 2 (xxHash): accepted, not verified -- corrupting the
    # checksum field itself must NOT fail the walk
    sst = bytearray(write_sst(blocks, checksum_type=2))
    sst[len(blocks[0]) + 1] ^= 0xFF     # first block's checksum byte
    keys, ko, vals, vo, n = orc.sst_parse(bytes(sst))
    assert n == len(kvs)


def test_sst_engine_error_paths_cpu():
    """the engine-side walk rejects the same corrupt files (host code --
    no GPU needed until the device block parse; bad files fail before)."""
    # corrupt files fail in sst_layout before any HIP call, so a real
    # engine handle is unnecessary -- exercised on GPU in the parity test
    blocks, _ = two_block_kvs()
    lib = tikv_amd._ffi.load_lib()
    bad = write_sst(blocks, magic=0xDEAD)
    buf = (C.c_uint8 * len(bad)).from_buffer_copy(bad)
    r = C.c_void_p()
    st = lib.copr_region_create_sst(None, C.cast(buf, C.POINTER(C.c_uint8)),
                                    len(bad), C.byref(r))
    assert st != 0


def test_sst_oracle_single_block():
    orc = _orc()
    e1 = [(b"tabc_r0001", b"only")]
    sst = write_sst([hand_block(e1, 1)])
    keys, ko, vals, vo, n = orc.sst_parse(sst)
    assert n == 1 and vals == b"only"


# ---- GPU: engine ingestion parity -------------------------------------
@pytest.mark.gpu
def test_sst_device_matches_direct(engine):
    """SST file (plain + compressed) -> device region == gen_blocks path
    == raw region, over a generated KV set, end to end."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=40001, table_id=5)
    try:
        blocks, offs, n, keep = tikv_amd.gen_blocks(g)
        blist = [C.string_at(
            C.cast(C.addressof(blocks.contents) + offs[i],
                   C.POINTER(C.c_uint8)), offs[i + 1] - offs[i])
            for i in range(n)]
        for comp in (0, 7):
            sst = write_sst(blist, compression=comp)
            rgn = engine.region_sst(sst)
            try:
                d = engine.dump_region(rgn)
                assert d[4] == g.n_kv
                assert d[0] == C.string_at(g.keys, g.key_offs[g.n_kv])
                assert d[2] == C.string_at(g.vals, g.val_offs[g.n_kv])
            finally:
                rgn.close()
        # corrupt file: loud storage error
        bad = write_sst(blist, corrupt_block_byte=10)
        with pytest.raises(RuntimeError):
            engine.region_sst(bad)
    finally:
        g.close()


@pytest.mark.gpu
def test_sst_query_end_to_end(engine):
    """ingest an LZ4 SST, run the cfg2-shaped scan+filter+count against
    it, compare with the oracle over the same KVs."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=30001, table_id=1)
    try:
        blocks, offs, n, keep = tikv_amd.gen_blocks(g)
        blist = [C.string_at(
            C.cast(C.addressof(blocks.contents) + offs[i],
                   C.POINTER(C.c_uint8)), offs[i + 1] - offs[i])
            for i in range(n)]
        sst = write_sst(blist, compression=4)
        cols = [tikv_amd.Col(i) for i in range(1, 17)]
        sel = tikv_amd.cmp_col_const(3, tikv_amd._ffi.SIG_LT_INT,
                                     -800_000_000)
        req = (tikv_amd.DagSelect(cols).where(sel)
               .simple_agg([tikv_amd.count_star()]).build())
        od, on = orc.dag_run(req, g.keys, g.key_offs, g.vals, g.val_offs,
                             g.n_kv)
        rgn = engine.region_sst(sst)
        try:
            gd, gn, _ = engine.dag_run(req, [rgn])
        finally:
            rgn.close()
        assert (gn, gd) == (on, od)
    finally:
        g.close()


@pytest.mark.gpu
def test_sst_mvcc_device(engine):
    """write-CF SST -> file walk + device MVCC filter == the
    region_mvcc path on the same generated data."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=20001, table_id=5,
                           row_format=3)
    try:
        blocks, offs, n, keep = tikv_amd.gen_blocks(g)
        blist = [C.string_at(
            C.cast(C.addressof(blocks.contents) + offs[i],
                   C.POINTER(C.c_uint8)), offs[i + 1] - offs[i])
            for i in range(n)]
        sst = write_sst(blist)
        rgn_s = engine.region_sst_mvcc(sst, 1000)
        rgn_d = engine.region_mvcc(g, 1000)
        try:
            s = engine.dump_region(rgn_s)
            d = engine.dump_region(rgn_d)
            assert s[4] == d[4]
            assert s[0] == d[0] and s[2] == d[2]
            assert list(s[1]) == list(d[1]) and list(s[3]) == list(d[3])
        finally:
            rgn_s.close()
            rgn_d.close()
    finally:
        g.close()
