/* copr_gpu.h — C-ABI of the MI355X-native coprocessor batch-executor engine.
 *
 * This is the drop-in boundary described in DESIGN.md §1: it sits exactly where
 * BatchDagHandler hands work to BatchExecutorsRunner
 * (reference src/coprocessor/dag/mod.rs:157-196 -> runner.rs:673,840), expressed
 * in the stable-C-ABI idiom of the reference's own plugin boundary
 * (components/coprocessor_plugin_api/src/plugin_api.rs:21-42): no exceptions
 * cross the ABI, plain pointers and sizes, engine owns what it allocates.
 *
 * The KV feed mirrors the pull side of the Storage trait
 * (components/tidb_query_common/src/storage/mod.rs:32-79): the caller hands the
 * scanned (key, value) byte stream; the engine copies it into HBM once and runs
 * DAG requests against the resident region.
 *
 * Every compute entry point REQUIRES a HIP device; there is no CPU fallback.
 */
#ifndef COPR_GPU_H
#define COPR_GPU_H

#include "copr_types.h"

#ifdef __cplusplus
extern "C" {
#endif

typedef struct copr_engine copr_engine;   /* one per GPU (per process rank) */
typedef struct copr_region copr_region;   /* one resident Region KV block   */

typedef enum copr_status {
  COPR_OK = 0,
  COPR_ERR_NO_GPU = 1,          /* no HIP device / hip runtime failure */
  COPR_ERR_INVALID_REQUEST = 2, /* descriptor validation (mirrors
                                   ErrorInner::Evaluate mapping,
                                   dag/mod.rs:219-258) */
  COPR_ERR_UNSUPPORTED = 3,     /* sig/executor outside the native subset:
                                   the Rust shim falls back to its CPU path */
  COPR_ERR_STORAGE = 4,         /* malformed KV bytes (ErrorInner::Storage) */
  COPR_ERR_OOM = 5,
  COPR_ERR_INTERNAL = 6
} copr_status;

/* Last error message for this thread (valid until the next engine call). */
const char *copr_last_error(void);

/* ---- engine lifecycle ---- */
copr_status copr_engine_create(int hip_device, copr_engine **out);
void        copr_engine_destroy(copr_engine *);

/* ---- region feed (Storage boundary) ----
 * keys/vals are concatenated byte streams with offs[n] prefix offsets
 * (offs[0] = 0, offs[n] = total bytes). Keys arrive raw — the MVCC
 * memcomparable+ts envelope already stripped, as TikvStorage does
 * (src/coprocessor/dag/storage_impl.rs:93). Copied to HBM; host buffers may be
 * freed after return. */
copr_status copr_region_create(copr_engine *,
                               const uint8_t *keys, const uint64_t *key_offs,
                               const uint8_t *vals, const uint64_t *val_offs,
                               uint64_t n_kv, copr_region **out);

/* MVCC variant: the feed is the raw write-CF stream (sorted user-key asc /
 * commit_ts desc; key = memcomparable(user_key)||BE(~commit_ts), value =
 * WriteRef bytes — txn_types/src/write.rs:296-361) and the engine's version
 * filter kernel materializes the visible rows at read_ts on device
 * (forward.rs:440-515 semantics; Put values must be short values —
 * default-CF lookup is COPR_ERR_UNSUPPORTED). */
copr_status copr_region_create_mvcc(copr_engine *,
                                    const uint8_t *keys, const uint64_t *key_offs,
                                    const uint8_t *vals, const uint64_t *val_offs,
                                    uint64_t n_kv, uint64_t read_ts,
                                    copr_region **out);
/* MVCC variant with the DEFAULT CF beside the write CF: Puts whose write
 * record carries no short value (>255 B values, write.rs:296) resolve
 * their row bytes from the default-CF stream at key =
 * memcomparable(user_key)||BE(~start_ts) (forward.rs:433-515
 * load_data_from_default_cf). dkeys/dvals must be sorted like RocksDB
 * stores them (user_key asc, start_ts desc); a Put whose default entry is
 * absent is COPR_ERR_STORAGE (the reference treats it as corruption). */
copr_status copr_region_create_mvcc_with_default(copr_engine *,
                                    const uint8_t *keys, const uint64_t *key_offs,
                                    const uint8_t *vals, const uint64_t *val_offs,
                                    uint64_t n_kv,
                                    const uint8_t *dkeys, const uint64_t *dkey_offs,
                                    const uint8_t *dvals, const uint64_t *dval_offs,
                                    uint64_t n_default,
                                    uint64_t read_ts, copr_region **out);
void        copr_region_destroy(copr_region *);
uint64_t    copr_region_num_kv(const copr_region *);

/* ---- DAG execution (REQ_TYPE_DAG = 103, src/coprocessor/mod.rs:57) ----
 * Runs the executor tree over the region; returns the SelectResponse chunk
 * payload (datum-encoded rows). Free the result with copr_result_free. */
copr_status copr_dag_run(copr_engine *, const CoprDagRequest *,
                         copr_region *const *regions, uint32_t n_regions,
                         CoprSelectResult *out);
void        copr_result_free(CoprSelectResult *);

/* ---- checksum (REQ_TYPE_CHECKSUM = 105; src/coprocessor/checksum.rs:59-114) ----
 * CRC-64/XZ per KV over key||value, XOR-folded (order-independent). */
copr_status copr_checksum(copr_engine *, copr_region *const *regions,
                          uint32_t n_regions, uint64_t *checksum,
                          uint64_t *total_kvs, uint64_t *total_bytes);

/* ---- multi-GPU partial-aggregate merge (RCCL over xGMI) ----
 * The path shards by Region across GPUs with no data-path collective
 * (TiDB's own per-Region coprocessor fan-out, endpoint.rs:238-248); the ONLY
 * exchange is the final partial-aggregate merge. These calls
 * wrap an RCCL communicator owned by the engine: one process per GPU, the
 * 128-byte id produced by rank 0's copr_comm_id and distributed out-of-band
 * (the Rust shim's gRPC / the bench's rendezvous — plumbing, like NCCL's own
 * bootstrap). Payloads are KB-scale and latency-bound. All merge calls are
 * collective: every rank in the communicator must call them in the same
 * order. */
#define COPR_COMM_ID_BYTES 128
copr_status copr_comm_id(uint8_t out[COPR_COMM_ID_BYTES]);
copr_status copr_comm_create(copr_engine *,
                             const uint8_t id[COPR_COMM_ID_BYTES],
                             int n_ranks, int rank);
void        copr_comm_destroy(copr_engine *);
/* count(*)/count(col) final merge: sum of per-rank u64 counts */
copr_status copr_merge_count(copr_engine *, uint64_t *inout);
/* CRC64 running XOR is order-independent (checksum.rs:78-87); RCCL has no
 * XOR reduce op, so this allgathers world u64s and folds */
copr_status copr_merge_checksum(copr_engine *, uint64_t *inout);
/* exact two's-complement i128 partial-sum merge (Decimal/int sums travel
 * as scaled i128 limbs): elementwise allgather + host fold
 * with carries */
copr_status copr_merge_sum_i128(copr_engine *, uint64_t *lo, uint64_t *hi);
/* f64 partial sums (Real aggregates): allreduce; parallel order keeps the
 * 1-ULP class */
copr_status copr_merge_sum_f64(copr_engine *, double *inout);

/* ---- synthetic region generator (fixture factory) ----
 * Host-side (OpenMP) generator of reference-format regions, mirroring
 * test_coprocessor's fixture store (test_coprocessor/src/store.rs:83-91:
 * table::encode_row_key + table::encode_row). Not part of the serving path —
 * it exists so tests/bench feed identical bytes to engine and oracle.
 * See DESIGN.md §6 for the RNG contract. */
typedef struct CoprGenSpec {
  int32_t  config_index;     /* BASELINE.json configs[] index (seeds the RNG) */
  int64_t  table_id;
  uint64_t n_rows;
  uint64_t first_handle;     /* shard offset: rows get handles
                                [first_handle, first_handle + n_rows) */
  uint32_t n_cols;           /* schema per config (see copr_gen.cpp) */
  int32_t  row_format;       /* 1 = row-v1 datums, 2 = row-v2 */
} CoprGenSpec;

typedef struct CoprGenOut {   /* host buffers owned by the generator */
  uint8_t  *keys;  uint64_t *key_offs;
  uint8_t  *vals;  uint64_t *val_offs;
  uint64_t  n_kv;
} CoprGenOut;

copr_status copr_gen_region(const CoprGenSpec *, CoprGenOut *out);
void        copr_gen_free(CoprGenOut *);

/* ---- SST data-block ingestion ----
 * Feed uncompressed RocksDB BlockBasedTable DATA blocks instead of
 * per-KV arrays: each block is [entries: varint32 shared | varint32
 * non_shared | varint32 value_len | key_delta | value]* followed by
 * [restart offsets u32le x n][num_restarts u32le]; keys are InternalKeys
 * (user_key + 8B (seq<<8|type) trailer, stripped on decode). The engine
 * parses blocks ON DEVICE (one wavefront lane per restart interval) and
 * materializes a resident region. Reference producer: engine_rocks
 * iterators (engine_iterator.rs:12) behind SnapshotStore::scanner
 * (txn/store.rs:431); format per RocksDB block_builder.cc/block.cc
 * (public format; compression is the feeder's concern -- blocks arrive
 * uncompressed here). */
copr_status copr_region_create_blocks(copr_engine *,
                                      const uint8_t *blocks,
                                      const uint64_t *block_offs,
                                      uint32_t n_blocks,
                                      copr_region **out);

/* blocks carrying write-CF records: device block parse, then the device
 * MVCC version filter at read_ts (the full SnapshotStore::scanner chain
 * without per-KV host round trips) */
copr_status copr_region_create_blocks_mvcc(copr_engine *,
                                           const uint8_t *blocks,
                                           const uint64_t *block_offs,
                                           uint32_t n_blocks,
                                           uint64_t read_ts,
                                           copr_region **out);

/* compressed-block variants (RocksDB CompressionType per block:
 * 0 = none, 4/5 = LZ4/LZ4HC, 7 = ZSTD; compress_format_version 2 framing
 * = varint32 decompressed size + compressed payload). Decompression runs
 * on host cores via the system liblz4/libzstd before device parse;
 * snappy (1) is not present in this image and returns UNSUPPORTED. */
copr_status copr_region_create_blocks_compressed(copr_engine *,
                                                 const uint8_t *blocks,
                                                 const uint64_t *block_offs,
                                                 const uint8_t *types,
                                                 uint32_t n_blocks,
                                                 copr_region **out);

/* host-side helpers (fixture + tests): compress / decompress a block set */
copr_status copr_blocks_compress(const uint8_t *blocks,
                                 const uint64_t *block_offs,
                                 uint32_t n_blocks, uint8_t type,
                                 uint8_t **out, uint64_t **out_offs);
copr_status copr_blocks_decompress(const uint8_t *blocks,
                                   const uint64_t *block_offs,
                                   const uint8_t *types, uint32_t n_blocks,
                                   uint8_t **out, uint64_t **out_offs);

/* ---- whole-SST ingestion (the file layer above the block layer) ----
 * One complete BlockBasedTable file: the engine parses the footer
 * (format_version 1..5, 53-byte form: [checksum_type u8][metaindex
 * BlockHandle][index BlockHandle][padding][version u32le][magic u64le]),
 * walks the index block (values = plain BlockHandles, the kBinarySearch
 * shape TiKV writes), verifies each block's RocksDB-masked crc32c
 * trailer when checksum_type == 1 (the TiKV default; xxHash kinds are
 * accepted unverified), decompresses per the per-block trailer type
 * byte (0/4/5/7), and feeds the device block parser. Legacy (v0) and
 * v6+ footers, and delta-encoded index values, are rejected loudly.
 * Reference: RocksDB format.cc / block_based_table_reader.cc (public
 * format) consumed by TiKV via rust-rocksdb (engine_iterator.rs:12). */
copr_status copr_region_create_sst(copr_engine *,
                                   const uint8_t *file, uint64_t file_len,
                                   copr_region **out);

/* SST carrying write-CF records: file walk + device block parse + the
 * device MVCC version filter at read_ts */
copr_status copr_region_create_sst_mvcc(copr_engine *,
                                        const uint8_t *file,
                                        uint64_t file_len, uint64_t read_ts,
                                        copr_region **out);

/* fixture writer: pack a KV stream into data blocks (restart-interval
 * prefix compression; ~target_block_bytes per block). Buffers owned by
 * the generator allocator; free blocks with free() and offs with free() */
copr_status copr_gen_blocks(const uint8_t *keys, const uint64_t *key_offs,
                            const uint8_t *vals, const uint64_t *val_offs,
                            uint64_t n_kv, uint32_t restart_interval,
                            uint32_t target_block_bytes,
                            uint8_t **blocks_out, uint64_t **block_offs_out,
                            uint32_t *n_blocks_out);

/* test/debug: copy a resident region back to host buffers (owned by the
 * engine allocator; free with copr_gen_free) */
copr_status copr_region_dump(copr_engine *, copr_region *, CoprGenOut *out);

#ifdef __cplusplus
}
#endif
#endif /* COPR_GPU_H */
