/* orc_codec.h — ORACLE (test infrastructure ONLY).
 *
 * CPU restatement of the TiKV byte codecs the coprocessor hot path parses.
 * Each function cites the reference file:line it follows. This code is the
 * parity checker: only tests/, __graft_entry__.smoke() and bench.py's
 * cpu_baseline leg may link or execute it. The product path never does.
 */
#ifndef ORC_CODEC_H
#define ORC_CODEC_H

#include <stdint.h>
#include <stddef.h>
#include <vector>
#include <string>

namespace orc {

/* ---------- varint (components/codec/src/number.rs) ---------- */
/* encode_var_u64: number.rs:417-433 (LEB128, 7 bits/byte, cont bit 0x80) */
size_t encode_var_u64(uint8_t *buf, uint64_t v);
/* try_decode_var_u64: number.rs:445-484 (<=10 bytes; 10th byte contributes 1 bit) */
bool decode_var_u64(const uint8_t *p, size_t len, uint64_t *v, size_t *n);
/* encode_var_i64: number.rs:496-501 (zigzag: uv=(v<<1); if v<0 uv=!uv) */
size_t encode_var_i64(uint8_t *buf, int64_t v);
/* try_decode_var_i64: number.rs:513-520 */
bool decode_var_i64(const uint8_t *p, size_t len, int64_t *v, size_t *n);

/* ---------- memcomparable numbers (codec/src/number.rs:142-157, convert.rs) ---------- */
static const uint64_t SIGN_MARK = 0x8000000000000000ull;
void encode_comparable_u64(uint8_t *buf, uint64_t v);          /* big-endian */
uint64_t decode_comparable_u64(const uint8_t *buf);
void encode_comparable_i64(uint8_t *buf, int64_t v);           /* BE(v ^ SIGN) */
int64_t decode_comparable_i64(const uint8_t *buf);
void encode_comparable_u64_desc(uint8_t *buf, uint64_t v);     /* number.rs:121-133 */
uint64_t decode_comparable_u64_desc(const uint8_t *buf);
void encode_comparable_f64(uint8_t *buf, double v);            /* convert.rs:16-22 */
double decode_comparable_f64(const uint8_t *buf);

/* ---------- memcomparable bytes (codec/src/byte.rs:67-101,1517-1520) ----------
 * 8-byte groups + marker byte; asc: pad 0x00, marker = 0xFF - pad. */
size_t memcmp_encoded_len(size_t src_len);                      /* (len/8+1)*9 */
size_t memcmp_encode_all(const uint8_t *src, size_t len, uint8_t *dest);
size_t memcmp_encode_all_desc(const uint8_t *src, size_t len, uint8_t *dest);
/* decode: byte.rs try_decode_first (asc); returns consumed encoded bytes, or 0 on error */
size_t memcmp_decode(const uint8_t *src, size_t len, std::vector<uint8_t> *out);

/* ---------- compact bytes (byte.rs:506-530) ---------- */
size_t compact_bytes_encode(const uint8_t *src, size_t len, uint8_t *dest);
bool compact_bytes_decode(const uint8_t *p, size_t len,
                          const uint8_t **data, size_t *data_len, size_t *consumed);

/* ---------- datum flags (tidb_query_datatype/src/codec/datum.rs:35-47) ---------- */
enum DatumFlag : uint8_t {
  NIL_FLAG = 0, BYTES_FLAG = 1, COMPACT_BYTES_FLAG = 2, INT_FLAG = 3,
  UINT_FLAG = 4, FLOAT_FLAG = 5, DECIMAL_FLAG = 6, DURATION_FLAG = 7,
  VAR_INT_FLAG = 8, VAR_UINT_FLAG = 9, JSON_FLAG = 10,
  VECTOR_FLOAT32_FLAG = 20, MAX_FLAG = 250
};

/* split_datum (datum.rs:1117-1155): length of the first datum (flag+payload).
 * Returns false on malformed input. */
bool split_datum(const uint8_t *p, size_t len, size_t *datum_len);

/* ---------- table keys (tidb_query_datatype/src/codec/table.rs) ---------- */
/* encode_row_key: table.rs:187-193  't' || BE(tid^sign) || "_r" || BE(h^sign) */
void encode_row_key(int64_t table_id, int64_t handle, uint8_t out[19]);
/* decode_int_handle: table.rs:214-218 (key[11..19] comparable i64) */
bool decode_int_handle(const uint8_t *key, size_t len, int64_t *handle);

/* ---------- Decimal (tidb_query_datatype/src/codec/mysql/decimal.rs) ---------- */
struct Decimal {               /* decimal.rs:927-942 */
  uint8_t int_cnt = 1, frac_cnt = 0, result_frac_cnt = 0;
  bool negative = false;
  uint32_t word_buf[9] = {0};
};
Decimal dec_zero();                               /* decimal.rs:996 */
Decimal dec_from_i64(int64_t v);                  /* decimal.rs:1787-1798 */
Decimal dec_from_u64(uint64_t v);                 /* decimal.rs:1799-1815 */
/* &Decimal + &Decimal (decimal.rs:2340-2353): sign dispatch over
 * do_add (decimal.rs:492-590) / do_sub (decimal.rs:346-439).
 * Returns 0 = Ok, 1 = truncated, 2 = overflow (Res semantics). */
int dec_add(const Decimal &a, const Decimal &b, Decimal *out);
/* prec_and_frac: decimal.rs:1043-1051 */
void dec_prec_and_frac(const Decimal &d, uint8_t *prec, uint8_t *frac);
/* write_decimal (decimal.rs:2022-2133): returns bytes written into out
 * (caller reserves >= 42); includes the [prec][frac] header. */
size_t dec_encode(const Decimal &d, uint8_t prec, uint8_t frac, uint8_t *out);
/* read_decimal (decimal.rs:2204-2289). consumed = header+payload bytes. */
bool dec_decode(const uint8_t *p, size_t len, Decimal *d, size_t *consumed);
/* dec_encoded_len given [prec][frac] header (decimal.rs:169-192) */
bool dec_encoded_len(const uint8_t *p, size_t len, size_t *elen);
int dec_cmp(const Decimal &a, const Decimal &b);  /* decimal.rs PartialOrd */
std::string dec_to_string(const Decimal &d);      /* debugging */

/* ---------- CRC-64/XZ (crc64fast 0.1.0; third-party, not in the reference
 * tree — pinned by algorithm identity: poly 0x42F0E1EBA9EA3693 reflected,
 * init/xorout all-ones; KAT "123456789" -> 0x995DC9BBDF1939FA.
 * Call sites: src/coprocessor/checksum.rs:75,105) ---------- */
uint64_t crc64_xz(const uint8_t *p, size_t len);
uint64_t crc64_xz_update(uint64_t state, const uint8_t *p, size_t len); /* state = internal (pre-xorout) */
uint64_t crc64_xz_init();
uint64_t crc64_xz_finish(uint64_t state);

/* ---------- row v2 (codec/row/v2/row_slice.rs:76-168) ---------- */
struct RowSliceV2 {
  bool big = false;
  const uint8_t *non_null_ids = nullptr; uint16_t non_null_cnt = 0;
  const uint8_t *null_ids = nullptr;     uint16_t null_cnt = 0;
  const uint8_t *offsets = nullptr;      /* u16le or u32le each */
  const uint8_t *values = nullptr;       uint32_t values_len = 0;
};
bool row_v2_parse(const uint8_t *p, size_t len, RowSliceV2 *rs);
/* search_in_non_null_ids (row_slice.rs:125-151): binary search; returns value
 * byte range [start,end) into rs->values. */
bool row_v2_find(const RowSliceV2 &rs, int64_t col_id, uint32_t *start, uint32_t *end);
bool row_v2_is_null(const RowSliceV2 &rs, int64_t col_id);

/* v2 cell -> v1 datum re-encode (row/v2/compat_v1.rs:28-126), appends to out.
 * tp = CoprFieldTp, flag/decimal from the column's FieldType. */
bool row_v2_cell_to_v1_datum(const uint8_t *cell, size_t cell_len,
                             int32_t tp, uint32_t ft_flag,
                             std::vector<uint8_t> *out);

} // namespace orc
#endif
