"""RCCL merge surface (copr_comm.cpp) on hardware.

A single-rank communicator exercises the full RCCL wiring (ncclCommInitRank,
allreduce/allgather on the engine stream) on the box's one GPU; the
multi-rank fold arithmetic is covered by the gloo CPU tests
(tests/test_dist_cpu.py) and the identical host folds in copr_comm.cpp.
Reference context: the path's only collective is the final partial-aggregate
merge of the per-Region fan-out (endpoint.rs:238-248)."""
import pytest

import tikv_amd

pytestmark = pytest.mark.gpu

U64 = 2**64 - 1


def test_rccl_single_rank_merges(engine):
    cid = tikv_amd.Engine.comm_id()
    assert len(cid) == 128
    engine.comm_create(cid, 1, 0)
    try:
        assert engine.merge_count(7) == 7
        assert engine.merge_checksum(0xDEADBEEFCAFEF00D) == 0xDEADBEEFCAFEF00D
        assert engine.merge_sum_i128(5, 0) == (5, 0)
        # negative i128 partial sum survives the fold
        lo, hi = engine.merge_sum_i128((-3) & U64, U64)
        assert ((hi << 64) | lo) - (1 << 128) == -3
        assert engine.merge_sum_f64(1.5) == 1.5
    finally:
        engine.comm_destroy()


def test_comm_errors(engine):
    # merges without a communicator fail loudly, never silently no-op
    with pytest.raises(RuntimeError):
        engine.merge_count(1)
    cid = tikv_amd.Engine.comm_id()
    with pytest.raises(RuntimeError):
        engine.comm_create(cid, 2, 5)   # rank out of range
