/* copr_internal.h — PRODUCT internal host<->kernel shared structures. */
#ifndef COPR_INTERNAL_H
#define COPR_INTERNAL_H

#include <stdint.h>
#include <stddef.h>
#include <string>
#include <vector>

#include "../../include/copr_types.h"   /* scalar sig / tp enums */

namespace copr {

/* thread-local error string (defined in copr_engine.cpp; feeds
   copr_last_error()) */
std::string &tls_err();

/* device-resident region */
struct DevRegion {
  uint8_t *d_keys = nullptr;
  uint64_t *d_key_offs = nullptr;
  uint8_t *d_vals = nullptr;
  uint64_t *d_val_offs = nullptr;
  uint64_t n_kv = 0;
  uint64_t key_bytes = 0, val_bytes = 0;
  uint32_t max_row_bytes = 0;      /* max value size over all rows */
  uint32_t max_key_bytes = 0;      /* max key size (index scans parse keys) */
  /* cell directory: 16 column-major planes of n_kv bytes, built once at
     region ingest. dir[(c)*n_kv + row] = byte offset of the row-v1 cell
     whose column id is c+1 (the id-flag byte), 0xFF = column absent,
     0xFE = row not representable (v2/empty/oversized/ids outside 1..16)
     -> sequential walk. Removes the per-row sequential cell walk from the
     filter parse, the measured wall once DMA fully overlaps (the same idea
     row-v2 bakes into its format as the in-row offsets array). */
  uint8_t *d_celldir = nullptr;
};

/* compare kinds (order matches oracle CmpKind) */
enum { CMP_LT = 0, CMP_LE, CMP_GT, CMP_GE, CMP_EQ, CMP_NE };

/* device agg kinds */
enum {
  DAGG_COUNT_ROWS = 0,   /* count(const non-null): +1 per surviving row */
  DAGG_COUNT_COL,        /* count(col): +1 per non-null */
  DAGG_SUM_INT,          /* i128 sum of int col (covers sum/avg(int) after
                            the Decimal rewrite: exact integer sum) */
  DAGG_SUM_DEC,          /* i128 sum of decimal col scaled to target frac */
  /* fold aggregates over int columns; all ride sum_lo as a u64 fold with
     an order-preserving transform so the identity is 0:
       MAX: fold max over (v ^ SIGN-bias)      (cnt gates NULL)
       MIN: fold max over ~(v ^ SIGN-bias)
       BIT_AND: fold or over ~v  (result = ~acc; empty input -> ~0,
                impl_bit_op.rs AND identity)
       BIT_OR / BIT_XOR: fold or/xor over v */
  DAGG_MAX_INT,
  DAGG_MIN_INT,
  /* real (f64) max/min fold via the order-preserving bits transform
     (sign-clear -> |SIGN, sign-set -> ~bits); values travel as f64 BITS
     in the i64 channels */
  DAGG_MAX_REAL,
  DAGG_MIN_REAL,
  DAGG_BIT_AND,
  DAGG_BIT_OR,
  DAGG_BIT_XOR,
  /* FIRST (impl_first.rs): the first row's value, NULL included. Handled
     by the segmented pipelines only (simple agg reroutes through a
     single-run stream pass): acc.cnt = 1 once set, sum_lo = value,
     sum_hi = 1 when the first row's value is NULL. */
  DAGG_FIRST,
  /* f64 sum: atomicAdd(double) on sum_lo's bits; parallel order makes it
     1-ULP-class, within north_star's float budget */
  DAGG_SUM_REAL,
  /* capture-only pseudo-agg: the cell value feeds an RPN capture channel
     (selection over >2 distinct columns); never accumulates or emits */
  DAGG_XCAP,
};

/* flattened RPN node for the device predicate evaluator (mirrors the
   oracle's eval_rpn over RpnExpression — expr_eval.rs:205,264; int-typed
   columns only; column refs resolve to capture slots 0/1 at plan time) */
#define COPR_MAX_RPN 16
struct DevRpnNode {
  int32_t kind;   /* 0 col slot, 1 const int, 2 const NULL, 3 scalar func */
  int32_t sig;    /* kind 3: CoprScalarSig */
  int32_t slot;   /* kind 0: capture slot (0 = filter chan, 1 = filter2) */
  int32_t uns;    /* value unsignedness (col ft / const kind / func out ft) */
  int64_t cval;   /* kind 1 */
};

struct DevAggSpec {
  int32_t kind;
  int64_t col_id;        /* source column id (ignored for COUNT_ROWS) */
  int32_t col_unsigned;
  int32_t target_frac;   /* SUM_DEC: scale values to this frac */
};

#define COPR_MAX_AGGS 8
#define COPR_MAX_OUT_COLS 20

struct ScanPlan {
  /* filter: cmp(col, const) over ints; has_filter 0 => keep all */
  int32_t has_filter;
  int64_t filter_col_id;
  int32_t filter_cmp;
  int64_t filter_const;          /* f64 BITS when filter_is_real */
  int32_t filter_col_unsigned;
  int32_t filter_const_unsigned;
  int32_t filter_const_null;     /* NULL const: predicate never true */
  /* Real comparers (impl_compare.rs:66-160 Real path): the filter channel
     carries f64 BITS through the i64 value slots */
  int32_t filter_is_real;
  int32_t filter2_is_real;
  /* value a row takes when the filter column is absent: the scan default
     fill (table_scan_executor.rs:456-483) decoded */
  int32_t filter_missing_null;   /* 1 => NULL */
  int64_t filter_missing_val;
  /* second ANDed selection conjunct (selection_executor.rs:86 ANDs all
     conditions); same shape as the first filter. FASTFC and the dir
     single-plane shortcut require a single conjunct, so two-filter
     requests take the generic collect path. */
  int32_t filter2_on;
  int64_t filter2_col_id;
  int32_t filter2_cmp;
  int64_t filter2_const;
  int32_t filter2_col_unsigned;
  int32_t filter2_const_unsigned;
  int32_t filter2_const_null;
  int32_t filter2_missing_null;
  int64_t filter2_missing_val;
  /* general selection predicate: when rpn_on, the keep decision comes
     from evaluating this program over the two captured columns instead of
     the fixed cmp shapes (the capture channels and missing-fill fields
     above are reused; their cmp fields are ignored) */
  int32_t rpn_on;
  int32_t rpn_n;
  DevRpnNode rpn[COPR_MAX_RPN];
  /* RPN capture channels beyond the two filter channels (selection over
     up to 4 distinct columns, selection_executor.rs:86 ANDs any number):
     channel 2+k's value rides the capture-only pseudo-agg slot
     xcap_idx[k] (kind DAGG_XCAP) */
  int32_t n_xcap;
  int32_t xcap_idx[2];
  int32_t xcap_missing_null[2];
  int64_t xcap_missing_val[2];
  /* 1 = keep every row but still export filt_vals/filt_state: used for a
     column an upstream expression decoded in place (e.g. the TopN order
     column) so the response encodes it in DECODED form
     (lazy_column.rs:165,242) */
  int32_t filter_decode_only;
  /* second expression-decoded column for the TopN sub-region project (the
     original request's FILTER column when it also appears in the output):
     captured like the filter channel, output in decoded form. 0 = none. */
  int64_t dec2_col_id;
  int32_t dec2_col_unsigned;
  int32_t dec2_missing_null;
  int64_t dec2_missing_val;

  int32_t mode;                  /* 0 project, 1 simple agg, 2 hash agg */
  /* index scan (BatchIndexScanExecutor): the parsed stream is the KEY
     stream; columns are POSITIONAL comparable datums after the 19-byte
     prefix, with the int handle as the trailing datum (position
     index_n_cols). col ids in this plan are positions. */
  int32_t index_mode;
  int32_t index_n_cols;
  /* index mode: the ORIGINAL value stream (unique-index value handles,
     index_scan_executor.rs:416-422, and the new TailLen|Options layouts
     :322-371). Loaded lazily — only for rows whose key has no trailing
     handle datum or whose value is new-format. */
  const uint8_t *aux_vals;
  const uint64_t *aux_val_offs;
  /* max value length in the region: <= 9 means every value is old-format
     (no restore data possible), so the value is only read for rows whose
     key carries no trailing handle */
  uint32_t aux_max_vlen;
  /* index position -> the reference column_id (restore-data row lookups,
     RestoreData::V4: extract_columns_from_row_format :483-501) */
  int64_t index_real_ids[COPR_MAX_OUT_COLS];
  int32_t n_aggs;
  DevAggSpec aggs[COPR_MAX_AGGS];

  /* hash agg (mode 2) */
  int64_t group_col_id;
  int32_t group_col_unsigned;
  uint32_t table_size;           /* power of two */

  /* project (mode 0): capture raw cell spans for these column ids */
  int32_t n_out;
  int64_t out_col_ids[COPR_MAX_OUT_COLS];
  int32_t out_is_handle[COPR_MAX_OUT_COLS]; /* 1 => decoded int handle */

  /* filter column's cell-directory plane (d_celldir + (col-1)*n_kv) or
     null; consumed by the FASTFC pipe kernel only */
  const uint8_t *dir_plane;
  /* full directory base + rows for the generic collect path (hash agg /
     multi-agg): host sets it only when every needed col id is in 1..16 */
  const uint8_t *celldir;
  uint64_t celldir_n;
  /* diagnostics: 1 = stage tiles but skip the parse (bandwidth ceiling probe;
     COPR_DIAG_STAGE_ONLY=1; results are garbage, never used in tests) */
  int32_t diag_stage_only;
  /* tiling */
  uint32_t rows_per_tile;
  uint32_t lds_bytes;            /* dynamic LDS per block */
  /* glds pipeline (k_scan_agg_pipe): per-buffer slab sizes, both 1 KiB
     multiples; lds_bytes = 2 * (offs_slab + vals_slab). use_pipe 0 falls
     back to the single-buffer kernel. */
  int32_t use_pipe;
  int32_t glds_nt;               /* nt (aux=2) on the values stream */
  uint32_t offs_slab;
  uint32_t vals_slab;
  /* dir_slab = TOTAL bytes per buffer staging tile slices of needed
     directory planes (n_dir_slabs x 1 KiB): the per-row dir bytes head the
     parse dependency chain, so they ride the DMA instead of being per-row
     random global loads. Slab 0 is the filter plane (FASTFC reads it).
     dirslab_* = slab index per consumer, -1 = fall back to a global
     celldir load. */
  uint32_t dir_slab;
#define COPR_MAX_DIR_SLABS 6
  int32_t n_dir_slabs;
  const uint8_t *dir_planes_staged[COPR_MAX_DIR_SLABS];
  int32_t dirslab_f, dirslab_f2, dirslab_g;
  int32_t dirslab_a[COPR_MAX_AGGS];
  /* hash agg: per-block LDS pre-aggregation table (0 = disabled).
     Low-cardinality GROUP BY otherwise serializes on a handful of global
     atomic addresses. */
  uint32_t lds_agg_slots;        /* power of two */
  uint32_t lds_agg_off;          /* byte offset of the table in LDS */
  /* simple agg (mode 1): the two HIGH sum limbs per agg ([n_aggs*2]),
     extending decimal/int sums to 256 bits (wide Decimal sums) */
  unsigned long long *simple_ext;
};

/* simple-agg accumulators (device buffer, one per agg) */
struct SimpleAggAcc {
  unsigned long long cnt;
  unsigned long long sum_lo;
  unsigned long long sum_hi;     /* two's-complement high word */
};

/* hash-agg table (device): parallel arrays.
 * keys[]: EMPTY sentinel = INT64_MIN bias — the real INT64_MIN key and the
 * NULL key get dedicated accumulator blocks (reserved[0]=int64_min,
 * reserved[1]=null).
 * ext/rsvd_ext: the two HIGH limbs extending each acc's sum to 256 bits
 * (wide Decimal sums, decimal.rs:927-942 word_buf range; values up to 38
 * digits, sums up to ~77). Narrow adds ripple carries in lazily. */
struct HashAggTable {
  long long *keys;               /* [table_size] */
  SimpleAggAcc *accs;            /* [table_size * n_aggs] */
  SimpleAggAcc *reserved;        /* [2 * n_aggs] */
  unsigned long long *ext;       /* [table_size * n_aggs * 2] or null */
  unsigned long long *rsvd_ext;  /* [2 * n_aggs * 2] or null */
  unsigned long long *rsvd_seen; /* [2]: row counts for the 2 reserved keys */
  unsigned int *error;           /* [0]=table full, [1]=parse error */
  unsigned long long *n_groups;  /* occupied slot count */
};

/* project-mode outputs */
struct ProjectOut {
  unsigned long long *cells;     /* [n_rows * n_out]: off(44) | len(20);
                                    len 0xFFFFF => missing column */
  long long *handles;            /* [n_rows] if any handle col */
  uint8_t *keep;                 /* [n_rows] */
  /* the filter column is decoded in place by predicate eval, so its output
     form is the DECODED datum (lazy_column.rs:165,242 + expr eval
     ensure_columns_decoded): per-row value + state for the host encoder */
  long long *filt_vals;          /* [n_rows] */
  uint8_t *filt_state;           /* [n_rows]: 0 value, 1 NULL, 2 missing */
  long long *dec2_vals;          /* [n_rows] (dec2_col_id != 0) */
  uint8_t *dec2_state;
  unsigned int *error;
};

int dev_scan_launch(const ScanPlan &plan, const DevRegion &rgn,
                    SimpleAggAcc *d_simple, const HashAggTable *ht,
                    const ProjectOut *po, void *stream);
int dev_crc64_launch(const DevRegion &rgn, const uint64_t *d_tables /*8*256*/,
                     unsigned long long *d_xor, void *stream);
/* MVCC write-CF filter: builds a visible-row DevRegion from raw write-CF
 * arrays already on device. 0 ok, -1 malformed, -2 oom, -3 unsupported. */
int dev_celldir_build(DevRegion &rgn, hipStream_t s);
int dev_stream_agg(const ScanPlan &plan, const DevRegion &rgn, void *stream,
                   std::vector<SimpleAggAcc> *h_accs,
                   std::vector<long long> *h_gk, std::vector<uint8_t> *h_gs);
int dev_topn_select(const ScanPlan &plan, const DevRegion &rgn,
                    uint64_t topn_n, int desc, void *stream,
                    std::vector<uint32_t> *winners);
/* fill keys[] with the EMPTY sentinel (INT64_MIN) */
void dev_fill_keys(long long *keys, uint64_t n, void *stream);
int dev_ht_compact(const HashAggTable &ht, uint32_t tsize, int n_aggs,
                   void *stream, std::vector<long long> *h_keys,
                   std::vector<SimpleAggAcc> *h_accs,
                   std::vector<unsigned long long> *h_ext = nullptr);
int dev_int_sorted_agg(const ScanPlan &plan, const DevRegion &rgn,
                       void *stream, std::vector<SimpleAggAcc> *h_accs,
                       std::vector<long long> *h_gk,
                       std::vector<uint8_t> *h_gs);
int dev_bytes_agg(const ScanPlan &plan, const DevRegion &rgn, void *stream,
                  std::vector<SimpleAggAcc> *h_accs,
                  std::vector<uint64_t> *h_kofs, std::vector<uint32_t> *h_klen,
                  std::vector<uint8_t> *h_kst);
int dev_blocks_build(const uint8_t *h_blocks, const uint64_t *h_block_offs,
                     uint32_t n_blocks, DevRegion *out, void *stream);
int dev_subregion_build(const DevRegion &src, const uint32_t *h_rows,
                        uint64_t m, DevRegion *out, void *stream);
/* dd_* = the DEFAULT-CF stream (sorted user_key asc / start_ts desc), or
 * null: a Put without a short value then fails loudly (unsupported). */
/* device TypeChunk encode (project mode, fixed-8 int/real columns only):
 * per output column, how the chunk value is sourced. Mirrors the host
 * datum-row encode loop + e_chunk_append_datum round trip (chunk data is
 * the 8-byte LE of the decoded value either way). */
struct ChunkColSpec {
  int32_t kind;         /* 0 cell[j], 1 handle, 2 filter chan, 3 dec2 chan */
  int32_t j;            /* kind 0: index into po.cells row */
  int32_t uns;          /* kind 0: column UNSIGNED (v2 raw-payload decode) */
  int32_t is_real;      /* kind 0: DOUBLE column (v2 rows unsupported->err) */
  int32_t missing_null; /* kind 0/2/3: missing column -> NULL */
  int64_t missing_val;  /* else the decoded default (int or f64 bits) */
};
/* 0 ok; -1 hip error; -2 oom; -3 a row needs the host path (fall back).
 * idxp: project over an index scan — spans reference the key stream. */
int dev_chunk_encode(const ProjectOut &po, const DevRegion &rgn, int idxp,
                     uint64_t scan_end, const ChunkColSpec *h_specs,
                     int n_cols, int n_out,
                     const std::vector<uint64_t> &chunk_rows, void *stream,
                     std::vector<uint8_t> *out_resp);
int dev_mvcc_build(const uint8_t *d_keys, const uint64_t *d_ko,
                   const uint8_t *d_vals, const uint64_t *d_vo, uint64_t n,
                   const uint8_t *dd_keys, const uint64_t *dd_ko,
                   const uint8_t *dd_vals, const uint64_t *dd_vo, uint64_t dn,
                   uint64_t read_ts, DevRegion *out, int *unsupported,
                   void *stream);

}  // namespace copr

/* ---- engine / region handles (C-ABI opaque types) ---- */
struct copr_engine {
  int device = 0;
  hipStream_t stream = nullptr;
  uint64_t *d_crc_tables = nullptr;   /* 8*256 u64, built lazily */
  /* internal channel for the TopN sub-region project: column OFFSET the
     order expression decoded in place (response encodes it decoded,
     lazy_column.rs:165,242); -1 = none. dec2 = the original request's
     filter column when distinct from the order column. */
  int dec_col_off = -1;
  int dec2_col_off = -1;
  /* RCCL communicator state (copr_comm.cpp); null until copr_comm_create */
  void *comm_state = nullptr;
  /* hash-agg table cache: the per-request hipMalloc/hipFree set measured
     ~3 ms/step at cfg3 scale; buffers persist across dag_run calls and
     re-init with async fills */
  copr::HashAggTable ht_cache{};
  uint32_t ht_cache_tsize = 0;
  int ht_cache_naggs = 0;
  bool ht_cache_ext = false;
};

struct copr_region {
  copr_engine *eng = nullptr;
  copr::DevRegion dev;
  std::vector<uint64_t> h_key_offs, h_val_offs;   /* host copies for encode */
};

namespace copr {
/* frees the engine's RCCL state (no-op when absent); copr_comm.cpp */
void comm_free(copr_engine *);
}
#endif
