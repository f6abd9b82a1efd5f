/* copr_gen.cpp — synthetic Region generator (fixture factory; PRODUCT lib).
 *
 * Writes reference-format KV regions, mirroring the reference's own fixture
 * writer (test_coprocessor/src/store.rs:83-91):
 *   key  = table::encode_row_key (codec/table.rs:187-193)
 *        = 't' || BE(table_id ^ SIGN) || "_r" || BE(handle ^ SIGN)
 *   value= table::encode_row (codec/table.rs:166-184): repeated
 *          [VAR_INT datum(col_id), datum(value)] in non-comparable form
 *          (datum.rs write_datum, comparable=false: I64 -> VAR_INT zigzag
 *          varint; Decimal -> DECIMAL flag + write_decimal; Bytes ->
 *          COMPACT_BYTES).
 *
 * RNG contract: xoshiro256++ whose 256-bit state is seeded
 * per row by a splitmix64 chain over (0xC0FFEE + config_index, handle), so
 * generation is deterministic AND parallel. Schemas per BASELINE.json
 * configs.
 */
#include "../../include/copr_gpu.h"
#include "prod_decimal.h"

#include <cstring>
#include <cstdlib>
#include <vector>
#include <string>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace {

/* ---- encode primitives (cites as in header comment) ---- */
static inline size_t enc_var_u64(uint8_t *buf, uint64_t v) {
  size_t i = 0;
  while (v >= 0x80) { buf[i++] = 0x80 | (v & 0x7f); v >>= 7; }
  buf[i++] = (uint8_t)v;
  return i;
}
static inline size_t enc_var_i64(uint8_t *buf, int64_t v) {
  uint64_t uv = (uint64_t)v << 1;
  if (v < 0) uv = ~uv;
  return enc_var_u64(buf, uv);
}
static inline void enc_cmp_i64(uint8_t *buf, int64_t v) {
  uint64_t u = (uint64_t)v ^ 0x8000000000000000ull;
  for (int i = 7; i >= 0; i--) { buf[i] = (uint8_t)u; u >>= 8; }
}
static inline void enc_row_key(int64_t table_id, int64_t handle, uint8_t out[19]) {
  out[0] = 't';
  enc_cmp_i64(out + 1, table_id);
  out[9] = '_'; out[10] = 'r';
  enc_cmp_i64(out + 11, handle);
}

/* ---- RNG: splitmix64 (Vigna) + xoshiro256++ (Blackman/Vigna), public
 * domain algorithms ---- */
static inline uint64_t splitmix64(uint64_t &x) {
  uint64_t z = (x += 0x9E3779B97F4A7C15ull);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}
struct Xo256 {
  uint64_t s[4];
  void seed(uint64_t config_seed, uint64_t row) {
    uint64_t x = config_seed * 0x9E3779B97F4A7C15ull ^ (row + 0x243F6A8885A308D3ull);
    for (int i = 0; i < 4; i++) s[i] = splitmix64(x);
  }
  static inline uint64_t rotl(uint64_t x, int k) { return (x << k) | (x >> (64 - k)); }
  uint64_t next() {
    uint64_t result = rotl(s[0] + s[3], 23) + s[0];
    uint64_t t = s[1] << 17;
    s[2] ^= s[0]; s[3] ^= s[1]; s[1] ^= s[2]; s[0] ^= s[3];
    s[2] ^= t; s[3] = rotl(s[3], 45);
    return result;
  }
};

static inline int64_t uniform_pm(Xo256 &rng, int64_t bound) {
  /* uniform in [-bound, bound] */
  return (int64_t)(rng.next() % (uint64_t)(2 * bound + 1)) - bound;
}

/* ---- per-config row writers. Append value bytes to buf. ---- */
struct GenCfg {
  int32_t config_index;
  uint64_t param;           /* cfg3: group count K */
};

/* cell helper: [VAR_INT col_id][datum] */
static inline void put_cell_i64(std::string &buf, int64_t col_id, int64_t v) {
  uint8_t tmp[24];
  size_t n = 0;
  tmp[n++] = 8;                       /* VAR_INT flag (col id datum) */
  n += enc_var_i64(tmp + n, col_id);
  tmp[n++] = 8;                       /* VAR_INT flag (value) */
  n += enc_var_i64(tmp + n, v);
  buf.append((const char *)tmp, n);
}
static inline void put_cell_decimal(std::string &buf, int64_t col_id,
                                    int64_t scaled, uint8_t frac) {
  uint8_t tmp[64];
  size_t n = 0;
  tmp[n++] = 8;
  n += enc_var_i64(tmp + n, col_id);
  tmp[n++] = 6;                       /* DECIMAL flag */
  prod::PDec d = prod::pdec_from_scaled_i128(scaled, frac);
  uint8_t prec, fr;
  prod::pdec_prec_and_frac(d, &prec, &fr);
  n += prod::pdec_encode(d, prec, fr, tmp + n);
  buf.append((const char *)tmp, n);
}
static inline void put_cell_bytes(std::string &buf, int64_t col_id,
                                  const uint8_t *data, size_t len) {
  uint8_t tmp[16];
  size_t n = 0;
  tmp[n++] = 8;
  n += enc_var_i64(tmp + n, col_id);
  tmp[n++] = 2;                       /* COMPACT_BYTES flag */
  n += enc_var_i64(tmp + n, (int64_t)len);
  buf.append((const char *)tmp, n);
  buf.append((const char *)data, len);
}

/* memcomparable bytes (codec/src/byte.rs:67-101): 8-byte groups + marker */
static inline void enc_memcmp(std::string &out, const uint8_t *src, size_t len) {
  size_t full = len / 8;
  for (size_t g = 0; g < full; g++) {
    out.append((const char *)src + g * 8, 8);
    out.push_back((char)0xFF);
  }
  size_t rem = len - full * 8;
  out.append((const char *)src + full * 8, rem);
  out.append(8 - rem, '\0');
  out.push_back((char)(0xFF - (8 - rem)));
}

static bool write_row(const GenCfg &cfg, Xo256 &rng, std::string &buf) {
  switch (cfg.config_index) {
    case 0: {                          /* cfg1: 4 x i64, ids 1..4, ±1e9 */
      for (int64_t c = 1; c <= 4; c++) put_cell_i64(buf, c, uniform_pm(rng, 1000000000));
      return true;
    }
    case 1: {                          /* cfg2: 16 x i64, ids 1..16, ±1e9 */
      for (int64_t c = 1; c <= 16; c++) put_cell_i64(buf, c, uniform_pm(rng, 1000000000));
      return true;
    }
    case 2: {                          /* cfg3: i64 group, Decimal(12,2), VarBytes */
      uint64_t k = cfg.param ? cfg.param : 64;
      put_cell_i64(buf, 1, (int64_t)(rng.next() % k));
      put_cell_decimal(buf, 2, uniform_pm(rng, 999999999999ll), 2);
      uint8_t bytes[24];
      size_t blen = 8 + rng.next() % 17;
      for (size_t i = 0; i < blen; i++) {
        double u = (double)(rng.next() >> 11) * (1.0 / 9007199254740992.0);
        bytes[i] = (uint8_t)('a' + (int)(26.0 * u * u * u));  /* low-biased */
      }
      put_cell_bytes(buf, 3, bytes, blen);
      return true;
    }
    case 4: {                          /* cfg5: handled by the key writer */
      buf.push_back('0');              /* old-encoding non-unique value */
      return true;
    }
    case 3: {                          /* cfg4: opaque value 100..200 B */
      size_t blen = 100 + rng.next() % 101;
      uint8_t chunk[8];
      for (size_t i = 0; i < blen; i += 8) {
        uint64_t x = rng.next();
        memcpy(chunk, &x, 8);
        buf.append((const char *)chunk, (blen - i) < 8 ? (blen - i) : 8);
      }
      return true;
    }
    default:
      return false;
  }
}

/* row v2 writer (small layout): header [128][flags][nn u16le][null u16le],
 * sorted u8 ids (non-null then null), u16le end-offsets, then cell payloads
 * (codec/row/v2/row_slice.rs:76-168; cell encodings compat_v1.rs:12-38:
 * ints little-endian minimal width — 1/2/4/8, negatives 8). */
static void v2_int_cell(std::string &cells, int64_t v) {
  if (v >= 0 && v <= 0xFF) cells.push_back((char)(uint8_t)v);
  else if (v >= 0 && v <= 0xFFFF) {
    cells.push_back((char)(uint8_t)v); cells.push_back((char)(uint8_t)(v >> 8));
  } else if (v >= 0 && v <= 0xFFFFFFFFll) {
    for (int b = 0; b < 4; b++) cells.push_back((char)(uint8_t)(v >> (8 * b)));
  } else {
    for (int b = 0; b < 8; b++) cells.push_back((char)(uint8_t)(v >> (8 * b)));
  }
}

static bool write_row_v2(const GenCfg &cfg, Xo256 &rng, std::string &buf) {
  /* cells in id order; occasional NULL column for coverage */
  uint8_t nn_ids[20], null_ids[20];
  uint32_t nn = 0, nl = 0;
  std::string cells;
  uint16_t ends[20];
  auto add_int = [&](uint8_t id, int64_t v) {
    if ((rng.next() & 63) == 0) { null_ids[nl++] = id; return; }
    nn_ids[nn] = id;
    v2_int_cell(cells, v);
    ends[nn] = (uint16_t)cells.size();
    nn++;
  };
  switch (cfg.config_index) {
    case 0:
      for (uint8_t c = 1; c <= 4; c++) add_int(c, uniform_pm(rng, 1000000000));
      break;
    case 1:
      for (uint8_t c = 1; c <= 16; c++) add_int(c, uniform_pm(rng, 1000000000));
      break;
    case 2: {
      uint64_t k = cfg.param ? cfg.param : 64;
      add_int(1, (int64_t)(rng.next() % k));
      if ((rng.next() & 63) == 0) { null_ids[nl++] = 2; }
      else {
        nn_ids[nn] = 2;
        prod::PDec d = prod::pdec_from_scaled_i128(uniform_pm(rng, 999999999999ll), 2);
        uint8_t prec, fr, tmp[48];
        prod::pdec_prec_and_frac(d, &prec, &fr);
        size_t m = prod::pdec_encode(d, prec, fr, tmp);
        cells.append((const char *)tmp, m);
        ends[nn] = (uint16_t)cells.size();
        nn++;
      }
      nn_ids[nn] = 3;
      uint8_t bytes[24];
      size_t blen = 8 + rng.next() % 17;
      for (size_t i = 0; i < blen; i++) {
        double u = (double)(rng.next() >> 11) * (1.0 / 9007199254740992.0);
        bytes[i] = (uint8_t)('a' + (int)(26.0 * u * u * u));
      }
      cells.append((const char *)bytes, blen);
      ends[nn] = (uint16_t)cells.size();
      nn++;
      break;
    }
    default:
      return false;
  }
  buf.push_back((char)128);
  buf.push_back(0);                                 /* flags: small, no cksum */
  buf.push_back((char)(uint8_t)nn); buf.push_back(0);
  buf.push_back((char)(uint8_t)nl); buf.push_back(0);
  for (uint32_t i = 0; i < nn; i++) buf.push_back((char)nn_ids[i]);
  for (uint32_t i = 0; i < nl; i++) buf.push_back((char)null_ids[i]);
  for (uint32_t i = 0; i < nn; i++) {
    buf.push_back((char)(uint8_t)ends[i]);
    buf.push_back((char)(uint8_t)(ends[i] >> 8));
  }
  buf += cells;
  return true;
}

}  // namespace

static copr_status copr_gen_region_mvcc(const CoprGenSpec *spec, CoprGenOut *out);


/* MVCC write-CF wrap (row_format=3): each logical row becomes 1..3 versions
 * of a write-CF entry, sorted user-key asc / commit_ts desc:
 *   key   = memcomparable(record_key) || BE(~commit_ts)
 *           (txn_types/src/types.rs:152-161, Key::from_raw + append_ts)
 *   value = [type 'P'|'D'|'L'|'R'][varint start_ts]
 *           [optional 'v' len short_value]   (txn_types/src/write.rs:296-361)
 * Reader semantics target read_ts = COPR_MVCC_READ_TS (default 1000):
 * newest version with commit_ts <= read_ts; Put -> visible (short value),
 * Delete -> key invisible, Lock/Rollback -> look older
 * (forward.rs:433-515). */
static copr_status copr_gen_region_mvcc(const CoprGenSpec *spec, CoprGenOut *out) {
  GenCfg cfg{spec->config_index, 0};
  if (spec->config_index == 2) cfg.param = spec->n_cols ? spec->n_cols : 64;
  uint64_t n = spec->n_rows;
  uint64_t seed = 0xC0FFEEull + (uint64_t)spec->config_index + 0x77ull;

  int T = 1;
#ifdef _OPENMP
  T = omp_get_max_threads();
#endif
  if ((uint64_t)T > n / 1024 + 1) T = (int)(n / 1024 + 1);
  std::vector<std::string> kchunk(T), vchunk(T);
  std::vector<std::vector<uint32_t>> ksz(T), vsz(T);
  uint64_t per = (n + T - 1) / T;

#ifdef _OPENMP
#pragma omp parallel for schedule(static, 1)
#endif
  for (int t = 0; t < T; t++) {
    uint64_t lo = (uint64_t)t * per, hi = lo + per;
    if (hi > n) hi = n;
    if (lo >= hi) continue;
    Xo256 rng;
    std::string row;
    uint8_t rk[19];
    for (uint64_t i = lo; i < hi; i++) {
      int64_t handle = (int64_t)(spec->first_handle + i);
      enc_row_key(spec->table_id, handle, rk);
      rng.seed(seed, (uint64_t)handle);
      uint32_t nver = 1 + (uint32_t)(rng.next() % 3);
      uint64_t ts = 900 + rng.next() % 200;           /* newest; read_ts=1000 */
      for (uint32_t v = 0; v < nver; v++) {
        /* key */
        size_t k0 = kchunk[t].size();
        enc_memcmp(kchunk[t], rk, 19);
        uint8_t tsb[8];
        uint64_t desc = ~ts;
        for (int b = 7; b >= 0; b--) { tsb[b] = (uint8_t)desc; desc >>= 8; }
        kchunk[t].append((const char *)tsb, 8);
        ksz[t].push_back((uint32_t)(kchunk[t].size() - k0));
        /* value */
        size_t v0 = vchunk[t].size();
        uint32_t pick = (uint32_t)(rng.next() % 10);
        char type = pick < 7 ? 'P' : pick == 7 ? 'D' : pick == 8 ? 'R' : 'L';
        vchunk[t].push_back(type);
        uint8_t tmp[10];
        size_t nn = enc_var_u64(tmp, ts ? ts - 1 : 0);  /* start_ts */
        vchunk[t].append((const char *)tmp, nn);
        if (type == 'P') {
          row.clear();
          if (!write_row(cfg, rng, row)) row.clear();
          if (row.size() > 255) row.resize(255);       /* short-value cap */
          vchunk[t].push_back('v');
          vchunk[t].push_back((char)(uint8_t)row.size());
          vchunk[t] += row;
          if ((rng.next() & 15) == 0) {
            /* gc fence tag: 0 (valid) or a ts below read_ts (invalidates) */
            uint64_t fence = (rng.next() & 1) ? 0 : 500;
            vchunk[t].push_back('F');
            for (int b = 7; b >= 0; b--) vchunk[t].push_back((char)(uint8_t)(fence >> (8 * b)));
          }
        } else if (type == 'R' && (rng.next() & 3) == 0) {
          /* protected rollback short value (write.rs:35) */
          vchunk[t].push_back('v');
          vchunk[t].push_back((char)1);
          vchunk[t].push_back('p');
        } else if ((type == 'L' || type == 'R') && (rng.next() & 7) == 0) {
          /* last_change NotExist: BE(0) + varint(1) (types.rs:703-718) */
          vchunk[t].push_back('l');
          for (int b = 0; b < 8; b++) vchunk[t].push_back('\0');
          vchunk[t].push_back((char)1);
        }
        vsz[t].push_back((uint32_t)(vchunk[t].size() - v0));
        ts -= 1 + rng.next() % 100;
      }
    }
  }

  uint64_t total_e = 0, ktot = 0, vtot = 0;
  std::vector<uint64_t> ebase(T), kbase(T), vbase(T);
  for (int t = 0; t < T; t++) {
    ebase[t] = total_e; total_e += ksz[t].size();
    kbase[t] = ktot; ktot += kchunk[t].size();
    vbase[t] = vtot; vtot += vchunk[t].size();
  }
  uint8_t *keys = (uint8_t *)malloc(ktot ? ktot : 1);
  uint8_t *vals = (uint8_t *)malloc(vtot ? vtot : 1);
  uint64_t *key_offs = (uint64_t *)malloc((total_e + 1) * 8);
  uint64_t *val_offs = (uint64_t *)malloc((total_e + 1) * 8);
  if (!keys || !vals || !key_offs || !val_offs) return COPR_ERR_OOM;

#ifdef _OPENMP
#pragma omp parallel for schedule(static, 1)
#endif
  for (int t = 0; t < T; t++) {
    if (!kchunk[t].empty()) memcpy(keys + kbase[t], kchunk[t].data(), kchunk[t].size());
    if (!vchunk[t].empty()) memcpy(vals + vbase[t], vchunk[t].data(), vchunk[t].size());
    uint64_t ko = kbase[t], vo = vbase[t], e = ebase[t];
    for (size_t j = 0; j < ksz[t].size(); j++) {
      key_offs[e] = ko; val_offs[e] = vo;
      ko += ksz[t][j]; vo += vsz[t][j];
      e++;
    }
  }
  key_offs[total_e] = ktot;
  val_offs[total_e] = vtot;
  out->keys = keys; out->key_offs = key_offs;
  out->vals = vals; out->val_offs = val_offs;
  out->n_kv = total_e;
  return COPR_OK;
}

extern "C" {

copr_status copr_gen_region(const CoprGenSpec *spec, CoprGenOut *out) {
  memset(out, 0, sizeof(*out));
  GenCfg cfg{spec->config_index, 0};
  /* for config_index==2 (cfg3), n_cols carries the group count K */
  if (spec->config_index == 2) cfg.param = spec->n_cols ? spec->n_cols : 64;
  uint64_t n = spec->n_rows;
  uint64_t seed = 0xC0FFEEull + (uint64_t)spec->config_index;

  if (spec->row_format == 3) return copr_gen_region_mvcc(spec, out);

  uint64_t *key_offs = (uint64_t *)malloc((n + 1) * sizeof(uint64_t));
  uint64_t *val_offs = (uint64_t *)malloc((n + 1) * sizeof(uint64_t));
  /* cfg5 index layout variants (spec->n_cols): 0 = non-unique old (handle
     datum in key, value '0'); 1 = unique old (no key handle, value = 8B BE
     handle, index_scan_executor.rs:336-345); 2 = unique new-format version
     0 with V4 restore data (value = [TailLen=8]||row-v2(ids 1,2)||BE
     handle — the shape of the reference's own
     test_new_collation_unique_int_handle_index :1826-1870); 3 = non-unique
     new-format version 1 with a partition-id option (value =
     [0][125][1][126][pid 8B], handle from key). */
  uint32_t il = spec->config_index == 4 ? (uint32_t)spec->n_cols : 0;
  uint64_t klen = spec->config_index == 4 ? ((il == 1 || il == 2) ? 37 : 46)
                                          : 19;
  uint8_t *keys = (uint8_t *)malloc(n * klen + 1);
  if (!key_offs || !val_offs || !keys) return COPR_ERR_OOM;
  for (uint64_t i = 0; i <= n; i++) key_offs[i] = i * klen;

  int T = 1;
#ifdef _OPENMP
  T = omp_get_max_threads();
#endif
  if ((uint64_t)T > n / 1024 + 1) T = (int)(n / 1024 + 1);
  std::vector<std::string> chunk_buf(T);
  std::vector<std::vector<uint32_t>> chunk_sizes(T);
  uint64_t per = (n + T - 1) / T;
  bool ok = true;

#ifdef _OPENMP
#pragma omp parallel for schedule(static, 1)
#endif
  for (int t = 0; t < T; t++) {
    uint64_t lo = (uint64_t)t * per, hi = lo + per;
    if (hi > n) hi = n;
    if (lo >= hi) continue;
    std::string &buf = chunk_buf[t];
    buf.reserve((hi - lo) * 160);
    chunk_sizes[t].resize(hi - lo);
    Xo256 rng;
    std::string row;
    for (uint64_t i = lo; i < hi; i++) {
      int64_t handle = (int64_t)(spec->first_handle + i);
      rng.seed(seed, (uint64_t)handle);
      if (spec->config_index == 4) {
        /* cfg5: TPCC-order-line-shaped secondary index entry (index_id 1):
           't'||tid||'_i'||BE(1)|| INT datums (ol_w_d i64 in [0,3000),
           amount i64 +-1e6) [|| INT handle datum for non-unique layouts]
           (index_scan_executor.rs key format; encode_index_seek_key,
           table.rs:229-235) */
        int64_t c1 = (int64_t)(rng.next() % 3000);
        int64_t c2 = uniform_pm(rng, 1000000);
        uint8_t *k = keys + i * klen;
        k[0] = 't';
        enc_cmp_i64(k + 1, spec->table_id);
        k[9] = '_'; k[10] = 'i';
        enc_cmp_i64(k + 11, 1);
        k[19] = 3;  /* INT_FLAG */
        enc_cmp_i64(k + 20, c1);
        k[28] = 3;
        enc_cmp_i64(k + 29, c2);
        if (il == 0 || il == 3) {
          k[37] = 3;
          enc_cmp_i64(k + 38, handle);
        }
        row.clear();
        switch (il) {
          case 0:
            row.push_back('0');          /* old-encoding non-unique value */
            break;
          case 1:                        /* unique old: 8B BE handle */
            for (int b = 7; b >= 0; b--)
              row.push_back((char)(uint8_t)((uint64_t)handle >> (8 * b)));
            break;
          case 2: {                      /* unique new v0 + V4 restore row */
            std::string v2;
            uint8_t nn = 2;
            std::string cells;
            uint16_t ends[2];
            v2_int_cell(cells, c1); ends[0] = (uint16_t)cells.size();
            v2_int_cell(cells, c2); ends[1] = (uint16_t)cells.size();
            v2.push_back((char)128); v2.push_back(0);
            v2.push_back((char)nn); v2.push_back(0);
            v2.push_back(0); v2.push_back(0);
            v2.push_back(1); v2.push_back(2);          /* ids */
            for (int j = 0; j < 2; j++) {
              v2.push_back((char)(uint8_t)ends[j]);
              v2.push_back((char)(uint8_t)(ends[j] >> 8));
            }
            v2 += cells;
            row.push_back(8);            /* TailLen = 8 (handle only) */
            row += v2;                   /* restore data (flag byte 128) */
            for (int b = 7; b >= 0; b--)
              row.push_back((char)(uint8_t)((uint64_t)handle >> (8 * b)));
            break;
          }
          case 3:                        /* non-unique new v1 + pid option */
            row.push_back(0);            /* TailLen = 0 */
            row.push_back(125);          /* INDEX_VALUE_VERSION_FLAG */
            row.push_back(1);            /* version 1 */
            row.push_back(126);          /* INDEX_VALUE_PARTITION_ID_FLAG */
            for (int b = 7; b >= 0; b--)
              row.push_back((char)(uint8_t)((uint64_t)spec->table_id >> (8 * b)));
            break;
          default:
            ok = false;
        }
        if (!ok) break;
        chunk_sizes[t][i - lo] = (uint32_t)row.size();
        buf += row;
        continue;
      }
      enc_row_key(spec->table_id, handle, keys + i * 19);
      row.clear();
      bool wok = spec->row_format == 2 ? write_row_v2(cfg, rng, row)
                                       : write_row(cfg, rng, row);
      if (!wok) { ok = false; break; }
      chunk_sizes[t][i - lo] = (uint32_t)row.size();
      buf += row;
    }
  }
  if (!ok) { free(key_offs); free(val_offs); free(keys); return COPR_ERR_INVALID_REQUEST; }

  uint64_t total = 0;
  std::vector<uint64_t> chunk_base(T);
  for (int t = 0; t < T; t++) { chunk_base[t] = total; total += chunk_buf[t].size(); }
  uint8_t *vals = (uint8_t *)malloc(total ? total : 1);
  if (!vals) { free(key_offs); free(val_offs); free(keys); return COPR_ERR_OOM; }

#ifdef _OPENMP
#pragma omp parallel for schedule(static, 1)
#endif
  for (int t = 0; t < T; t++) {
    if (!chunk_buf[t].empty())
      memcpy(vals + chunk_base[t], chunk_buf[t].data(), chunk_buf[t].size());
    uint64_t lo = (uint64_t)t * per, hi = lo + per;
    if (hi > n) hi = n;
    uint64_t off = chunk_base[t];
    for (uint64_t i = lo; i < hi; i++) {
      val_offs[i] = off;
      off += chunk_sizes[t][i - lo];
    }
  }
  val_offs[n] = total;

  out->keys = keys; out->key_offs = key_offs;
  out->vals = vals; out->val_offs = val_offs;
  out->n_kv = n;
  return COPR_OK;
}

void copr_gen_free(CoprGenOut *out) {
  if (!out) return;
  free(out->keys); free(out->key_offs); free(out->vals); free(out->val_offs);
  memset(out, 0, sizeof(*out));
}


/* ---- SST data-block fixture writer (RocksDB BlockBasedTable data block:
 * prefix-compressed entries + restart array; block_builder.cc) ---- */
static void put_varint32(std::vector<uint8_t> *out, uint32_t v) {
  while (v >= 0x80) { out->push_back((uint8_t)(v | 0x80)); v >>= 7; }
  out->push_back((uint8_t)v);
}

extern "C" copr_status copr_gen_blocks(
    const uint8_t *keys, const uint64_t *key_offs, const uint8_t *vals,
    const uint64_t *val_offs, uint64_t n_kv, uint32_t restart_interval,
    uint32_t target_block_bytes, uint8_t **blocks_out,
    uint64_t **block_offs_out, uint32_t *n_blocks_out) {
  if (!restart_interval) restart_interval = 16;
  if (!target_block_bytes) target_block_bytes = 4096;
  std::vector<uint8_t> all;
  std::vector<uint64_t> boffs{0};
  std::vector<uint8_t> blk;
  std::vector<uint32_t> restarts;
  std::string prev_ikey;
  uint32_t counter = 0;
  auto flush = [&]() {
    if (blk.empty()) return;
    for (uint32_t r : restarts) {
      blk.push_back((uint8_t)r); blk.push_back((uint8_t)(r >> 8));
      blk.push_back((uint8_t)(r >> 16)); blk.push_back((uint8_t)(r >> 24));
    }
    uint32_t nr = (uint32_t)restarts.size();
    blk.push_back((uint8_t)nr); blk.push_back((uint8_t)(nr >> 8));
    blk.push_back((uint8_t)(nr >> 16)); blk.push_back((uint8_t)(nr >> 24));
    all.insert(all.end(), blk.begin(), blk.end());
    boffs.push_back(all.size());
    blk.clear(); restarts.clear(); prev_ikey.clear(); counter = 0;
  };
  for (uint64_t i = 0; i < n_kv; i++) {
    /* InternalKey = user key + 8B (seq<<8 | kTypeValue) little-endian */
    std::string ikey((const char *)(keys + key_offs[i]),
                     (size_t)(key_offs[i + 1] - key_offs[i]));
    uint64_t trailer = ((n_kv - i) << 8) | 0x1;
    for (int b = 0; b < 8; b++) ikey.push_back((char)(trailer >> (8 * b)));
    uint32_t shared = 0;
    if (counter % restart_interval == 0) {
      restarts.push_back((uint32_t)blk.size());
    } else {
      size_t m = std::min(prev_ikey.size(), ikey.size());
      while (shared < m && prev_ikey[shared] == ikey[shared]) shared++;
    }
    uint32_t vlen = (uint32_t)(val_offs[i + 1] - val_offs[i]);
    put_varint32(&blk, shared);
    put_varint32(&blk, (uint32_t)ikey.size() - shared);
    put_varint32(&blk, vlen);
    blk.insert(blk.end(), ikey.begin() + shared, ikey.end());
    blk.insert(blk.end(), vals + val_offs[i], vals + val_offs[i + 1]);
    prev_ikey = ikey;
    counter++;
    if (blk.size() >= target_block_bytes && i + 1 < n_kv) flush();
  }
  flush();
  uint8_t *ab = (uint8_t *)malloc(all.size() ? all.size() : 1);
  memcpy(ab, all.data(), all.size());
  uint64_t *ob = (uint64_t *)malloc(boffs.size() * 8);
  memcpy(ob, boffs.data(), boffs.size() * 8);
  *blocks_out = ab;
  *block_offs_out = ob;
  *n_blocks_out = (uint32_t)(boffs.size() - 1);
  return COPR_OK;
}

}  // extern "C"
