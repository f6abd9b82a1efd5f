"""SST data-block ingestion.

RocksDB BlockBasedTable data-block format (block_builder.cc/block.cc,
public format; TiKV consumes it via engine_rocks iterators,
engine_iterator.rs:12): prefix-compressed entries with restart points,
InternalKey = user key + 8-byte (seq<<8|type) trailer. Blocks arrive
uncompressed (decompression is the feeder's concern).

CPU tests pin the oracle decoder on a hand-built block and on round trips
through the fixture writer; GPU tests compare the device parser bit-for-
bit against the oracle and check end-to-end query equality between a
block-fed region and a directly-fed one.
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


def varint32(v):
    out = bytearray()
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    return bytes(out)


def hand_block(entries, restart_interval=2):
    """entries: [(user_key, value)]; returns one block's bytes."""
    blk = bytearray()
    restarts = []
    prev = b""
    for i, (uk, val) in enumerate(entries):
        ikey = uk + ((len(entries) - i) << 8 | 1).to_bytes(8, "little")
        shared = 0
        if i % restart_interval == 0:
            restarts.append(len(blk))
        else:
            while (shared < min(len(prev), len(ikey))
                   and prev[shared] == ikey[shared]):
                shared += 1
        blk += varint32(shared) + varint32(len(ikey) - shared)
        blk += varint32(len(val)) + ikey[shared:] + val
        prev = ikey
    for rr in restarts:
        blk += rr.to_bytes(4, "little")
    blk += len(restarts).to_bytes(4, "little")
    return bytes(blk)


def as_bufs(block_bytes_list):
    allb = b"".join(block_bytes_list)
    offs = [0]
    for b in block_bytes_list:
        offs.append(offs[-1] + len(b))
    bb = (C.c_uint8 * max(len(allb), 1)).from_buffer_copy(allb or b"\0")
    oo = (C.c_uint64 * len(offs))(*offs)
    return (C.cast(bb, C.POINTER(C.c_uint8)), oo, len(block_bytes_list),
            (bb, oo))


def test_block_oracle_hand():
    orc = _orc()
    entries = [(b"tabc_r0001", b"v1"), (b"tabc_r0002", b"value-two"),
               (b"tabc_r0003", b""), (b"tabd_r0001", b"x" * 40)]
    blk = hand_block(entries)
    bb, oo, n, keep = as_bufs([blk])
    keys, ko, vals, vo, nkv = orc.block_parse(bb, oo, n)
    assert nkv == 4
    got = [(keys[ko[i]:ko[i + 1]], vals[vo[i]:vo[i + 1]]) for i in range(4)]
    assert got == entries


def test_block_oracle_roundtrip():
    """gen KVs -> fixture writer -> oracle decode == original stream."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=5000, table_id=7)
    try:
        blocks, offs, n, keep = tikv_amd.gen_blocks(g, target_block_bytes=2048)
        assert n > 1
        keys, ko, vals, vo, nkv = orc.block_parse(blocks, offs, n)
        assert nkv == g.n_kv
        orig_keys = C.string_at(g.keys, g.key_offs[g.n_kv])
        orig_vals = C.string_at(g.vals, g.val_offs[g.n_kv])
        assert keys == orig_keys and vals == orig_vals
        assert ko == [g.key_offs[i] for i in range(g.n_kv + 1)]
        assert vo == [g.val_offs[i] for i in range(g.n_kv + 1)]
    finally:
        g.close()


@pytest.mark.gpu
def test_block_device_parity(engine):
    """device block parse == oracle, and a block-fed region answers
    queries identically to a directly-fed one."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=2, n_rows=120001, table_id=5)
    try:
        blocks, offs, n, keep = tikv_amd.gen_blocks(g)
        rgn_b = engine.region_blocks(blocks, offs, n)
        try:
            d_keys, d_ko, d_vals, d_vo, d_n = engine.dump_region(rgn_b)
            o_keys, o_ko, o_vals, o_vo, o_n = orc.block_parse(blocks, offs, n)
            assert d_n == o_n == g.n_kv
            assert d_keys == o_keys and d_vals == o_vals
            assert list(d_ko) == o_ko and list(d_vo) == o_vo
            rgn_d = engine.region(g)
            try:
                cols = [tikv_amd.Col(1),
                        tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
                        tikv_amd.Col(3, tp=F.TP_VARCHAR)]
                sel = tikv_amd.cmp_col_const(0, F.SIG_LT_INT, 0)
                req = (tikv_amd.DagSelect(cols).where(sel)
                       .simple_agg([tikv_amd.count_star(),
                                    tikv_amd.sum_col(1, decimal=2)]).build())
                bd, br, _ = engine.dag_run(req, [rgn_b])
                dd, dr, _ = engine.dag_run(req, [rgn_d])
                assert (br, bd) == (dr, dd)
                cs_b = engine.checksum([rgn_b])
                cs_d = engine.checksum([rgn_d])
                assert cs_b == cs_d
            finally:
                rgn_d.close()
        finally:
            rgn_b.close()
    finally:
        g.close()


@pytest.mark.gpu
def test_block_mvcc_chain_gpu(engine):
    """write-CF blocks -> device block parse -> device MVCC filter equals
    the direct write-CF array path bit-for-bit."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=50001, table_id=1,
                           row_format=3)
    try:
        blocks, offs, n, keep = tikv_amd.gen_blocks(g)
        rgn_b = engine.region_blocks_mvcc(blocks, offs, n, 1000)
        rgn_d = engine.region_mvcc(g, 1000)
        try:
            b = engine.dump_region(rgn_b)
            d = engine.dump_region(rgn_d)
            assert b[4] == d[4]
            assert b[0] == d[0] and b[2] == d[2]
            assert list(b[1]) == list(d[1]) and list(b[3]) == list(d[3])
        finally:
            rgn_b.close()
            rgn_d.close()
    finally:
        g.close()


def _comp_bufs(gen, ctype):
    """compress the gen's blocks with the engine helper; returns ctypes."""
    lib = gen._lib
    blocks, offs, n, keep = tikv_amd.gen_blocks(gen)
    cb = C.POINTER(C.c_uint8)()
    co = C.POINTER(C.c_uint64)()
    st = lib.copr_blocks_compress(blocks, offs, n, C.c_uint8(ctype),
                                  C.byref(cb), C.byref(co))
    assert st == 0
    types = (C.c_uint8 * n)(*([ctype] * n))
    return blocks, offs, cb, co, types, n, (keep, cb, co, types)


@pytest.mark.parametrize("ctype", [4, 7])   # LZ4, ZSTD
def test_block_compress_roundtrip(ctype):
    """compress -> decompress round trip equals the original block bytes
    (RocksDB compress_format_version 2 framing)."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=3000, table_id=7)
    try:
        blocks, offs, cb, co, types, n, keep = _comp_bufs(g, ctype)
        lib = g._lib
        db = C.POINTER(C.c_uint8)()
        do = C.POINTER(C.c_uint64)()
        st = lib.copr_blocks_decompress(cb, co, types, n, C.byref(db),
                                        C.byref(do))
        assert st == 0
        total = offs[n]
        assert C.string_at(db, total) == C.string_at(blocks, total)
        assert [do[i] for i in range(n + 1)] == [offs[i] for i in range(n + 1)]
        # compression actually shrank the synthetic blocks
        assert co[n] < total
        # snappy (1) is absent in this image: loud UNSUPPORTED
        t2 = (C.c_uint8 * n)(*([1] * n))
        st = lib.copr_blocks_decompress(cb, co, t2, n, C.byref(db),
                                        C.byref(do))
        assert st != 0
    finally:
        g.close()


@pytest.mark.gpu
def test_block_compressed_device(engine):
    """compressed blocks -> host decompress -> device parse == direct."""
    orc = _orc()
    g = tikv_amd.GenRegion(config_index=1, n_rows=40001, table_id=5)
    try:
        blocks, offs, cb, co, types, n, keep = _comp_bufs(g, 7)
        r = C.c_void_p()
        st = engine._lib.copr_region_create_blocks_compressed(
            engine._h, cb, co, types, n, C.byref(r))
        assert st == 0
        from tikv_amd.runner import Region
        rgn = Region(engine, r)
        try:
            d = engine.dump_region(rgn)
            assert d[4] == g.n_kv
            assert d[0] == C.string_at(g.keys, g.key_offs[g.n_kv])
            assert d[2] == C.string_at(g.vals, g.val_offs[g.n_kv])
        finally:
            rgn.close()
    finally:
        g.close()


def test_block_oracle_malformed():
    """truncated / corrupt blocks are rejected, never misparsed."""
    orc = _orc()
    entries = [(b"tabc_r0001", b"v1"), (b"tabc_r0002", b"v2")]
    blk = bytearray(hand_block(entries))
    cases = [
        bytes(blk[:-2]),                      # truncated restart count
        bytes(blk[:3]),                       # truncated entry
        blk[:-4] + (99).to_bytes(4, "little"),  # absurd restart count
    ]
    import pytest as _pytest
    for bad in cases:
        bb, oo, n, keep = as_bufs([bytes(bad)])
        with _pytest.raises(RuntimeError):
            orc.block_parse(bb, oo, n)
    # shared > accumulated key length must be rejected
    evil = bytearray()
    evil += varint32(5) + varint32(1) + varint32(0) + b"k"   # shared=5, no prior
    evil += (0).to_bytes(4, "little")
    evil += (1).to_bytes(4, "little")
    bb, oo, n, keep = as_bufs([bytes(evil)])
    with _pytest.raises(RuntimeError):
        orc.block_parse(bb, oo, n)


def test_decompress_malformed():
    """corrupt compressed payloads error cleanly."""
    g = tikv_amd.GenRegion(config_index=1, n_rows=500, table_id=7)
    try:
        blocks, offs, cb, co, types, n, keep = _comp_bufs(g, 4)
        lib = g._lib
        # flip a byte in the first compressed block's payload
        raw = bytearray(C.string_at(cb, co[n]))
        raw[co[0] + 3] ^= 0xFF
        rb = (C.c_uint8 * len(raw)).from_buffer_copy(bytes(raw))
        db = C.POINTER(C.c_uint8)()
        do = C.POINTER(C.c_uint64)()
        st = lib.copr_blocks_decompress(C.cast(rb, C.POINTER(C.c_uint8)), co,
                                        types, n, C.byref(db), C.byref(do))
        # either an explicit error or (for lz4's tolerant cases) output that
        # differs from the original -- it must never silently equal it
        if st == 0:
            assert C.string_at(db, offs[n]) != C.string_at(blocks, offs[n])
        else:
            assert st != 0
    finally:
        g.close()
