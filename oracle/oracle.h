/* oracle.h — ORACLE public C API (test infrastructure ONLY).
 *
 * CPU restatement of the reference coprocessor batch-executor pipeline
 * (components/tidb_query_executors/src/runner.rs:840 handle_request and the
 * executors it drives) over the same CoprDagRequest descriptor the engine
 * consumes. Used ONLY by tests/, __graft_entry__.smoke() and bench.py's
 * cpu_baseline leg — the product path never links or calls this library.
 */
#ifndef ORACLE_H
#define ORACLE_H

#include <stdint.h>
#include <stddef.h>
#include "../include/copr_types.h"

#ifdef __cplusplus
extern "C" {
#endif

typedef struct OrcResult {
  uint8_t  *data;       /* datum-encoded response rows (runner.rs:1188) */
  uint64_t  data_len;
  uint64_t  n_rows;
  uint64_t  resume_row; /* paging resume point (row index the scan stopped
                           at; UINT64_MAX = drained) — runner.rs:915-943 */
} OrcResult;

/* 0 = ok; nonzero = error (message via orc_last_error) */
int  orc_dag_run(const CoprDagRequest *req,
                 const uint8_t *keys, const uint64_t *key_offs,
                 const uint8_t *vals, const uint64_t *val_offs,
                 uint64_t n_kv, OrcResult *out);
void orc_result_free(OrcResult *);
const char *orc_last_error(void);

/* checksum restatement (src/coprocessor/checksum.rs:59-114) */
int orc_checksum(const uint8_t *keys, const uint64_t *key_offs,
                 const uint8_t *vals, const uint64_t *val_offs,
                 uint64_t n_kv,
                 uint64_t *checksum, uint64_t *total_kvs, uint64_t *total_bytes);

/* ---- codec primitives exposed for golden-vector tests (ctypes) ---- */
uint64_t orc_crc64_xz(const uint8_t *p, uint64_t len);
uint64_t orc_test_memcmp_encode(const uint8_t *src, uint64_t len, int desc,
                                uint8_t *out /* >= (len/8+1)*9 */);
uint64_t orc_test_memcmp_decode(const uint8_t *src, uint64_t len,
                                uint8_t *out /* >= len */, uint64_t *out_len);
uint64_t orc_test_var_i64_encode(int64_t v, uint8_t out[10]);
int      orc_test_var_i64_decode(const uint8_t *p, uint64_t len, int64_t *v,
                                 uint64_t *consumed);
void     orc_test_row_key(int64_t table_id, int64_t handle, uint8_t out[19]);
int      orc_test_int_handle(const uint8_t *key, uint64_t len, int64_t *handle);
/* decode a row-v2 value: writes the v1 datum re-encode of column `col_id`
 * into out (cap 64); returns datum length, 0 if column absent/null-marked
 * (is_null=1 when in the null-id list), -1 on parse error */
int orc_test_row_v2_col(const uint8_t *val, uint64_t len, int64_t col_id,
                        int32_t tp, uint32_t ft_flag,
                        uint8_t *out, int *is_null);
/* decimal: decode datum-payload -> add -> encode with prec_and_frac;
 * used to cross-check GPU decimal sums. Returns encoded length or -1. */
int orc_test_dec_add_encode(const uint8_t *a, uint64_t alen,
                            const uint8_t *b, uint64_t blen, uint8_t *out);
int orc_test_dec_from_i64_encode(int64_t v, uint8_t *out);

/* ---- MVCC write-CF version filter (test-side restatement of
 * forward.rs:433-515 LatestKvPolicy + write.rs:296-361 WriteRef::parse,
 * for records without gc_fence/last_change tags) ---- */
typedef struct OrcRegion {
  uint8_t *keys; uint64_t *key_offs;
  uint8_t *vals; uint64_t *val_offs;
  uint64_t n_kv;
} OrcRegion;
/* 0 ok; 1 malformed; 2 unsupported (gc_fence / default-CF value) */
int  orc_mvcc_filter(const uint8_t *keys, const uint64_t *key_offs,
                     const uint8_t *vals, const uint64_t *val_offs,
                     uint64_t n_kv, uint64_t read_ts, OrcRegion *out);
/* with the DEFAULT CF beside the write CF (forward.rs:433-515
 * load_data_from_default_cf): Puts without a short value resolve from
 * dkeys at memcomparable(user_key)||BE(~start_ts). */
int  orc_mvcc_filter2(const uint8_t *keys, const uint64_t *key_offs,
                      const uint8_t *vals, const uint64_t *val_offs,
                      uint64_t n_kv,
                      const uint8_t *dkeys, const uint64_t *dkey_offs,
                      const uint8_t *dvals, const uint64_t *dval_offs,
                      uint64_t n_default, uint64_t read_ts, OrcRegion *out);
void orc_region_free(OrcRegion *);

/* ---- whole-SST walk (BlockBasedTable footer + index + per-block
 * trailer restatement; RocksDB format.cc public format). 0 ok,
 * -1 malformed/checksum mismatch, -2 unsupported footer/index shape. */
int orc_sst_parse(const uint8_t *file, uint64_t file_len, OrcRegion *out);

#ifdef __cplusplus
}
#endif
#endif
