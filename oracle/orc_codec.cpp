/* orc_codec.cpp — ORACLE (test infrastructure ONLY). See orc_codec.h. */
#include "orc_codec.h"
#include <cstring>
#include <algorithm>

namespace orc {

/* ============================ varint ============================ */
/* encode_var_u64: components/codec/src/number.rs:417-433 */
size_t encode_var_u64(uint8_t *buf, uint64_t v) {
  size_t i = 0;
  while (v >= 0x80) { buf[i++] = 0x80 | (v & 0x7f); v >>= 7; }
  buf[i++] = (uint8_t)v;
  return i;
}

/* try_decode_var_u64: number.rs:445-484 */
bool decode_var_u64(const uint8_t *p, size_t len, uint64_t *v, size_t *n) {
  uint64_t val = 0;
  if (len >= 10) {
    uint64_t b; int shift = 0;
    for (size_t i = 1; i <= 9; i++) {
      b = *p;
      val |= (b & 0x7f) << shift;
      if (b < 0x80) { *v = val; *n = i; return true; }
      p++; shift += 7;
    }
    b = *p;
    val |= (b & 0x01) << shift;
    *v = val; *n = 10; return true;
  }
  const uint8_t *end = p + len;
  const uint8_t *start = p;
  int shift = 0;
  while (p != end && *p >= 0x80) {
    val |= (uint64_t)(*p & 0x7f) << shift;
    shift += 7;
    p++;
  }
  if (p == end) return false;
  val |= (uint64_t)(*p) << shift;
  *v = val; *n = (size_t)(p - start) + 1;
  return true;
}

/* encode_var_i64: number.rs:496-501 */
size_t encode_var_i64(uint8_t *buf, int64_t v) {
  uint64_t uv = (uint64_t)v << 1;
  if (v < 0) uv = ~uv;
  return encode_var_u64(buf, uv);
}

/* try_decode_var_i64: number.rs:513-520 */
bool decode_var_i64(const uint8_t *p, size_t len, int64_t *v, size_t *n) {
  uint64_t uv;
  if (!decode_var_u64(p, len, &uv, n)) return false;
  uint64_t half = uv >> 1;
  *v = (uv & 1) ? (int64_t)~half : (int64_t)half;
  return true;
}

/* ================== memcomparable numbers ================== */
/* number.rs:142-157 (write_i64 = BE(v as u64 ^ SIGN_MARK)); convert.rs:6-22 */
void encode_comparable_u64(uint8_t *buf, uint64_t v) {
  for (int i = 7; i >= 0; i--) { buf[i] = (uint8_t)v; v >>= 8; }
}
uint64_t decode_comparable_u64(const uint8_t *buf) {
  uint64_t v = 0;
  for (int i = 0; i < 8; i++) v = (v << 8) | buf[i];
  return v;
}
void encode_comparable_i64(uint8_t *buf, int64_t v) {
  encode_comparable_u64(buf, (uint64_t)v ^ SIGN_MARK);
}
int64_t decode_comparable_i64(const uint8_t *buf) {
  return (int64_t)(decode_comparable_u64(buf) ^ SIGN_MARK);
}
/* desc order: number.rs:121-133 (additionally !) */
void encode_comparable_u64_desc(uint8_t *buf, uint64_t v) {
  encode_comparable_u64(buf, ~v);
}
uint64_t decode_comparable_u64_desc(const uint8_t *buf) {
  return ~decode_comparable_u64(buf);
}
/* f64: convert.rs:16-22 — if >=0: bits|SIGN_MARK else !bits */
void encode_comparable_f64(uint8_t *buf, double v) {
  uint64_t bits;
  memcpy(&bits, &v, 8);
  uint64_t u = (bits & SIGN_MARK) ? ~bits : (bits | SIGN_MARK);
  encode_comparable_u64(buf, u);
}
double decode_comparable_f64(const uint8_t *buf) {
  uint64_t u = decode_comparable_u64(buf);
  uint64_t bits = (u & SIGN_MARK) ? (u & ~SIGN_MARK) : ~u;
  double v;
  memcpy(&v, &bits, 8);
  return v;
}

/* ================== memcomparable bytes ================== */
/* byte.rs:1517-1520: GROUP=8, MARKER=0xFF, PAD=0x00; encode byte.rs:67-101 */
static const size_t MEMCMP_GROUP = 8;
size_t memcmp_encoded_len(size_t src_len) { return (src_len / MEMCMP_GROUP + 1) * (MEMCMP_GROUP + 1); }

size_t memcmp_encode_all(const uint8_t *src, size_t len, uint8_t *dest) {
  uint8_t *d0 = dest;
  size_t full = len / MEMCMP_GROUP;
  for (size_t g = 0; g < full; g++) {
    memcpy(dest, src, MEMCMP_GROUP);
    src += MEMCMP_GROUP; dest += MEMCMP_GROUP;
    *dest++ = 0xFF;
  }
  size_t rem = len - full * MEMCMP_GROUP;
  size_t pad = MEMCMP_GROUP - rem;
  memcpy(dest, src, rem);
  memset(dest + rem, 0x00, pad);
  dest += MEMCMP_GROUP;
  *dest++ = (uint8_t)~pad;           /* marker = !padding_size = 0xFF - pad */
  return (size_t)(dest - d0);
}

size_t memcmp_encode_all_desc(const uint8_t *src, size_t len, uint8_t *dest) {
  size_t n = memcmp_encode_all(src, len, dest);
  for (size_t i = 0; i < n; i++) dest[i] = ~dest[i];
  return n;
}

/* asc decode (byte.rs try_decode_first semantics) */
size_t memcmp_decode(const uint8_t *src, size_t len, std::vector<uint8_t> *out) {
  out->clear();
  size_t pos = 0;
  for (;;) {
    if (pos + 9 > len) return 0;
    const uint8_t *g = src + pos;
    uint8_t marker = g[8];
    pos += 9;
    if (marker == 0xFF) {
      out->insert(out->end(), g, g + 8);
    } else {
      size_t pad = 0xFF - marker;
      if (pad > 8) return 0;
      size_t datalen = 8 - pad;
      for (size_t i = datalen; i < 8; i++)
        if (g[i] != 0x00) return 0;
      out->insert(out->end(), g, g + datalen);
      return pos;
    }
  }
}

/* ================== compact bytes ================== */
/* byte.rs:518-530: varint_i64(len) || raw bytes */
size_t compact_bytes_encode(const uint8_t *src, size_t len, uint8_t *dest) {
  size_t n = encode_var_i64(dest, (int64_t)len);
  memcpy(dest + n, src, len);
  return n + len;
}
bool compact_bytes_decode(const uint8_t *p, size_t len,
                          const uint8_t **data, size_t *data_len, size_t *consumed) {
  int64_t l; size_t n;
  if (!decode_var_i64(p, len, &l, &n)) return false;
  if (l < 0 || n + (uint64_t)l > len) return false;
  *data = p + n; *data_len = (size_t)l; *consumed = n + (size_t)l;
  return true;
}

/* ================== split_datum ================== */
/* datum.rs:1117-1155 — length of first datum; non-comparable (desc=false) */
bool split_datum(const uint8_t *p, size_t len, size_t *datum_len) {
  if (len == 0) return false;
  uint8_t flag = p[0];
  const uint8_t *pl = p + 1;
  size_t rem = len - 1;
  size_t payload;
  switch (flag) {
    case NIL_FLAG: payload = 0; break;
    case INT_FLAG: case UINT_FLAG: case FLOAT_FLAG: case DURATION_FLAG:
      if (rem < 8) return false;
      payload = 8; break;
    case VAR_INT_FLAG: case VAR_UINT_FLAG: {
      uint64_t v; size_t n;
      if (!decode_var_u64(pl, rem, &v, &n)) return false;
      payload = n; break;
    }
    case BYTES_FLAG: {
      std::vector<uint8_t> tmp;
      size_t n = memcmp_decode(pl, rem, &tmp);
      if (n == 0) return false;
      payload = n; break;
    }
    case COMPACT_BYTES_FLAG: {
      const uint8_t *d; size_t dl, c;
      if (!compact_bytes_decode(pl, rem, &d, &dl, &c)) return false;
      payload = c; break;
    }
    case DECIMAL_FLAG: {
      size_t elen;
      if (!dec_encoded_len(pl, rem, &elen)) return false;
      payload = elen; break;
    }
    default:
      return false;  /* JSON / VECTOR_FLOAT32 unsupported in this build */
  }
  if (payload > rem) return false;
  *datum_len = 1 + payload;
  return true;
}

/* ================== table keys ================== */
/* table.rs:187-193 + TableEncoder (table.rs:61-67): 't' BE(tid^S) "_r" BE(h^S) */
void encode_row_key(int64_t table_id, int64_t handle, uint8_t out[19]) {
  out[0] = 't';
  encode_comparable_i64(out + 1, table_id);
  out[9] = '_'; out[10] = 'r';
  encode_comparable_i64(out + 11, handle);
}
/* table.rs:214-218 */
bool decode_int_handle(const uint8_t *key, size_t len, int64_t *handle) {
  if (len < 19 || key[0] != 't' || key[9] != '_' || key[10] != 'r') return false;
  *handle = decode_comparable_i64(key + 11);
  return true;
}

/* ================== Decimal ================== */
/* decimal.rs:132-134,140-141 */
static const int WORD_BUF_LEN = 9;
static const int DIGITS_PER_WORD = 9;
static const uint32_t WORD_BASE = 1000000000u;
static const uint32_t WORD_MAX = WORD_BASE - 1;
static const uint32_t TEN_POW[10] = {1, 10, 100, 1000, 10000, 100000,
                                     1000000, 10000000, 100000000, 1000000000};
static const uint8_t DIG_2_BYTES[10] = {0, 1, 1, 2, 2, 3, 3, 4, 4, 4};

static inline int word_cnt(int len) {        /* word_cnt! macro decimal.rs:144-166 */
  if (len > 0 && len > DIGITS_PER_WORD * WORD_BUF_LEN) return WORD_BUF_LEN + 1;
  if (len <= 0) return 0;
  return (len + DIGITS_PER_WORD - 1) / DIGITS_PER_WORD;
}

Decimal dec_zero() { Decimal d; d.int_cnt = 1; d.frac_cnt = 0; d.result_frac_cnt = 0; return d; }

static Decimal dec_new(uint8_t int_cnt, uint8_t frac_cnt, bool neg) { /* decimal.rs:981 */
  Decimal d;
  d.int_cnt = int_cnt; d.frac_cnt = frac_cnt; d.result_frac_cnt = frac_cnt;
  d.negative = neg;
  memset(d.word_buf, 0, sizeof(d.word_buf));
  return d;
}

Decimal dec_from_u64(uint64_t u) {           /* decimal.rs:1799-1815 */
  uint64_t x = u; int wi = 1;
  while (x >= WORD_BASE) { wi++; x /= WORD_BASE; }
  Decimal d = dec_new((uint8_t)(wi * DIGITS_PER_WORD), 0, false);
  x = u;
  while (wi > 0) { wi--; d.word_buf[wi] = (uint32_t)(x % WORD_BASE); x /= WORD_BASE; }
  return d;
}
Decimal dec_from_i64(int64_t i) {            /* decimal.rs:1787-1798 */
  bool neg = i < 0;
  Decimal d = dec_from_u64(neg ? (uint64_t)(-(uint64_t)i) : (uint64_t)i);
  d.negative = neg;
  return d;
}

/* count_leading_zeroes: decimal.rs:196-206 */
static uint8_t count_leading_zeroes(uint8_t i, uint32_t word) {
  uint8_t c = 0;
  while (TEN_POW[i] > word) { i--; c++; }
  return c;
}

/* remove_leading_zeroes: decimal.rs:1002-1018 */
static void remove_leading_zeroes(const Decimal &d, uint8_t prec,
                                  size_t *word_idx_out, uint8_t *cnt_out) {
  int cnt = prec;
  int i = ((cnt + DIGITS_PER_WORD - 1) % DIGITS_PER_WORD) + 1;
  size_t word_idx = 0;
  while (cnt > 0 && d.word_buf[word_idx] == 0) {
    cnt -= i; i = DIGITS_PER_WORD; word_idx++;
  }
  if (cnt > 0)
    cnt -= count_leading_zeroes((uint8_t)((cnt - 1) % DIGITS_PER_WORD), d.word_buf[word_idx]);
  *word_idx_out = word_idx;
  *cnt_out = (uint8_t)(cnt < 0 ? 0 : cnt);
}

void dec_prec_and_frac(const Decimal &d, uint8_t *prec, uint8_t *frac) { /* decimal.rs:1043-1051 */
  size_t wi; uint8_t int_cnt;
  remove_leading_zeroes(d, d.int_cnt, &wi, &int_cnt);
  uint8_t p = int_cnt + d.frac_cnt;
  if (p == 0) { *prec = 1; *frac = d.frac_cnt; }
  else { *prec = p; *frac = d.frac_cnt; }
}

/* add/sub word helpers: decimal.rs:216-265 */
static inline void word_add(uint32_t a, uint32_t b, uint32_t *carry, uint32_t *res) {
  uint32_t sum = a + b + *carry;
  if (sum >= WORD_BASE) { *res = sum - WORD_BASE; *carry = 1; }
  else { *res = sum; *carry = 0; }
}
static inline void word_sub(uint32_t l, uint32_t r, int32_t *carry, uint32_t *res) {
  int32_t diff = (int32_t)l - (int32_t)r - *carry;
  if (diff < 0) { *carry = 1; *res = (uint32_t)(diff + (int32_t)WORD_BASE); }
  else { *carry = 0; *res = (uint32_t)diff; }
}

/* fix_word_cnt_err: decimal.rs:228-236. Returns 0 ok / 1 truncated / 2 overflow */
static int fix_word_cnt_err(int int_word_cnt, int frac_word_cnt, int *iw, int *fw) {
  if (int_word_cnt + frac_word_cnt > WORD_BUF_LEN) {
    if (int_word_cnt > WORD_BUF_LEN) { *iw = WORD_BUF_LEN; *fw = 0; return 2; }
    *iw = int_word_cnt; *fw = WORD_BUF_LEN - int_word_cnt; return 1;
  }
  *iw = int_word_cnt; *fw = frac_word_cnt; return 0;
}

/* max_decimal: decimal.rs:448-479 */
static Decimal max_decimal(uint8_t prec, uint8_t frac_cnt) {
  uint8_t int_cnt = prec - frac_cnt;
  Decimal res = dec_new(int_cnt, frac_cnt, false);
  int idx = 0;
  if (int_cnt > 0) {
    uint8_t first = int_cnt % DIGITS_PER_WORD;
    if (first > 0) res.word_buf[idx++] = TEN_POW[first] - 1;
    for (int i = 0; i < int_cnt / DIGITS_PER_WORD; i++) res.word_buf[idx++] = WORD_MAX;
  }
  if (frac_cnt > 0) {
    static const uint32_t FRAC_MAX[8] = {900000000, 990000000, 999000000, 999900000,
                                         999990000, 999999000, 999999900, 999999990};
    uint8_t last = frac_cnt % DIGITS_PER_WORD;
    for (int i = 0; i < frac_cnt / DIGITS_PER_WORD; i++) res.word_buf[idx++] = WORD_MAX;
    if (last > 0) res.word_buf[idx] = FRAC_MAX[last - 1];
  }
  return res;
}

/* calc_sub_carry: decimal.rs:278-345.
 * carry: -1 = equal (None), 0 = |l|>|r|, 1 = |l|<|r| */
struct SubTmp { size_t start; size_t int_word_cnt; int frac_word_cnt; };
static void calc_sub_carry(const Decimal &lhs, const Decimal &rhs, int *carry,
                           int *frac_word_to, SubTmp *l_res, SubTmp *r_res) {
  int l_int_word_cnt = word_cnt(lhs.int_cnt), l_frac_word_cnt = word_cnt(lhs.frac_cnt);
  int r_int_word_cnt = word_cnt(rhs.int_cnt), r_frac_word_cnt = word_cnt(rhs.frac_cnt);
  *frac_word_to = std::max(l_frac_word_cnt, r_frac_word_cnt);

  size_t l_stop = (size_t)l_int_word_cnt, l_idx = 0;
  while (l_idx < l_stop && lhs.word_buf[l_idx] == 0) l_idx++;
  size_t l_start = l_idx;
  size_t l_iwc = l_stop - l_idx;

  size_t r_stop = (size_t)r_int_word_cnt, r_idx = 0;
  while (r_idx < r_stop && rhs.word_buf[r_idx] == 0) r_idx++;
  size_t r_start = r_idx;
  size_t r_iwc = r_stop - r_idx;

  int c;
  if (r_iwc > l_iwc) c = 1;
  else if (r_iwc < l_iwc) c = 0;
  else {
    intptr_t l_end = (intptr_t)(l_stop + l_frac_word_cnt) - 1;
    intptr_t r_end = (intptr_t)(r_stop + r_frac_word_cnt) - 1;
    while ((intptr_t)l_idx <= l_end && lhs.word_buf[l_end] == 0) l_end--;
    while ((intptr_t)r_idx <= r_end && rhs.word_buf[r_end] == 0) r_end--;
    l_frac_word_cnt = (int)std::max<intptr_t>(0, l_end + 1 - (intptr_t)l_stop);
    r_frac_word_cnt = (int)std::max<intptr_t>(0, r_end + 1 - (intptr_t)r_stop);
    while ((intptr_t)l_idx <= l_end && (intptr_t)r_idx <= r_end &&
           lhs.word_buf[l_idx] == rhs.word_buf[r_idx]) { l_idx++; r_idx++; }
    if ((intptr_t)l_idx <= l_end) {
      if ((intptr_t)r_idx <= r_end && rhs.word_buf[r_idx] > lhs.word_buf[l_idx]) c = 1;
      else c = 0;
    } else if ((intptr_t)r_idx <= r_end) c = 1;
    else c = -1;
  }
  *carry = c;
  *l_res = {l_start, l_iwc, l_frac_word_cnt};
  *r_res = {r_start, r_iwc, r_frac_word_cnt};
}

/* do_sub: decimal.rs:346-439. Returns Res code. */
static int do_sub(const Decimal *lhs, const Decimal *rhs, Decimal *out) {
  int carry_cls, frac_word_to;
  SubTmp l_res, r_res;
  calc_sub_carry(*lhs, *rhs, &carry_cls, &frac_word_to, &l_res, &r_res);
  if (carry_cls < 0) { *out = dec_zero(); return 0; }
  size_t l_start = l_res.start, r_start = r_res.start;
  size_t l_int_word_cnt = l_res.int_word_cnt, r_int_word_cnt = r_res.int_word_cnt;
  int l_frac_word_cnt = l_res.frac_word_cnt, r_frac_word_cnt = r_res.frac_word_cnt;

  bool negative;
  if (carry_cls > 0) {
    std::swap(lhs, rhs);
    std::swap(l_start, r_start);
    std::swap(l_int_word_cnt, r_int_word_cnt);
    std::swap(l_frac_word_cnt, r_frac_word_cnt);
    negative = !rhs->negative;
  } else {
    negative = lhs->negative;
  }

  int iw, fw;
  int rescode = fix_word_cnt_err((int)l_int_word_cnt, frac_word_to, &iw, &fw);
  l_int_word_cnt = (size_t)iw; frac_word_to = fw;
  size_t idx_to = l_int_word_cnt + (size_t)frac_word_to;
  uint8_t frac_cnt = std::max(lhs->frac_cnt, rhs->frac_cnt);
  uint8_t int_cnt = (uint8_t)(l_int_word_cnt * DIGITS_PER_WORD);
  if (rescode != 0) {
    frac_cnt = std::min<int>(frac_cnt, frac_word_to * DIGITS_PER_WORD);
    l_frac_word_cnt = std::min(l_frac_word_cnt, frac_word_to);
    r_frac_word_cnt = std::min(r_frac_word_cnt, frac_word_to);
    r_int_word_cnt = std::min(r_int_word_cnt, l_int_word_cnt);
  }
  int32_t carry = 0;
  *out = dec_new(int_cnt, frac_cnt, negative);

  size_t l_idx = l_start + l_int_word_cnt + (size_t)l_frac_word_cnt;
  size_t r_idx = r_start + r_int_word_cnt + (size_t)r_frac_word_cnt;
  if (l_frac_word_cnt > r_frac_word_cnt) {
    size_t l_stop = l_start + l_int_word_cnt + (size_t)r_frac_word_cnt;
    if (l_frac_word_cnt < frac_word_to)
      idx_to -= (size_t)(frac_word_to - l_frac_word_cnt);
    while (l_idx > l_stop) {
      idx_to--; l_idx--;
      out->word_buf[idx_to] = lhs->word_buf[l_idx];
    }
  } else {
    size_t r_stop = r_start + r_int_word_cnt + (size_t)l_frac_word_cnt;
    if (frac_word_to > r_frac_word_cnt)
      idx_to -= (size_t)(frac_word_to - r_frac_word_cnt);
    while (r_idx > r_stop) {
      idx_to--; r_idx--;
      word_sub(0, rhs->word_buf[r_idx], &carry, &out->word_buf[idx_to]);
    }
  }
  while (r_idx > r_start) {
    idx_to--; l_idx--; r_idx--;
    word_sub(lhs->word_buf[l_idx], rhs->word_buf[r_idx], &carry, &out->word_buf[idx_to]);
  }
  while (carry > 0 && l_idx > l_start) {
    idx_to--; l_idx--;
    word_sub(lhs->word_buf[l_idx], 0, &carry, &out->word_buf[idx_to]);
  }
  while (l_idx > l_start) {
    idx_to--; l_idx--;
    out->word_buf[idx_to] = lhs->word_buf[l_idx];
  }
  return rescode;
}

/* do_add: decimal.rs:492-590 */
static int do_add(const Decimal *lhs, const Decimal *rhs, Decimal *out) {
  int l_int_word_cnt = word_cnt(lhs->int_cnt), l_frac_word_cnt = word_cnt(lhs->frac_cnt);
  int r_int_word_cnt = word_cnt(rhs->int_cnt), r_frac_word_cnt = word_cnt(rhs->frac_cnt);
  int int_word_to = std::max(l_int_word_cnt, r_int_word_cnt);
  int frac_word_to = std::max(l_frac_word_cnt, r_frac_word_cnt);
  uint32_t x;
  if (l_int_word_cnt > r_int_word_cnt) x = lhs->word_buf[0];
  else if (l_int_word_cnt < r_int_word_cnt) x = rhs->word_buf[0];
  else x = lhs->word_buf[0] + rhs->word_buf[0];
  if (x > WORD_MAX - 1) int_word_to++;

  int iw, fw;
  int rescode = fix_word_cnt_err(int_word_to, frac_word_to, &iw, &fw);
  if (rescode == 2) {
    *out = max_decimal(WORD_BUF_LEN * DIGITS_PER_WORD, 0);
    return 2;
  }
  int_word_to = iw; frac_word_to = fw;
  size_t idx_to = (size_t)(int_word_to + frac_word_to);
  *out = dec_new((uint8_t)(int_word_to * DIGITS_PER_WORD),
                 std::max(lhs->frac_cnt, rhs->frac_cnt), lhs->negative);
  out->word_buf[0] = 0;
  if (rescode != 0) {
    out->frac_cnt = std::min<int>(frac_word_to * DIGITS_PER_WORD, out->frac_cnt);
    l_frac_word_cnt = std::min(frac_word_to, l_frac_word_cnt);
    r_frac_word_cnt = std::min(r_frac_word_cnt, frac_word_to);
    l_int_word_cnt = std::min(l_int_word_cnt, int_word_to);
    r_int_word_cnt = std::min(r_int_word_cnt, int_word_to);
  }
  size_t l_idx, r_idx, l_stop, r_stop;
  bool exchanged;
  if (l_frac_word_cnt > r_frac_word_cnt) {
    l_idx = (size_t)(l_int_word_cnt + l_frac_word_cnt);
    l_stop = (size_t)(l_int_word_cnt + r_frac_word_cnt);
    r_idx = (size_t)(r_int_word_cnt + r_frac_word_cnt);
    r_stop = (size_t)(l_int_word_cnt > r_int_word_cnt ? l_int_word_cnt - r_int_word_cnt : 0);
    exchanged = false;
  } else {
    l_idx = (size_t)(r_int_word_cnt + r_frac_word_cnt);
    l_stop = (size_t)(r_int_word_cnt + l_frac_word_cnt);
    r_idx = (size_t)(l_int_word_cnt + l_frac_word_cnt);
    r_stop = (size_t)(r_int_word_cnt > l_int_word_cnt ? r_int_word_cnt - l_int_word_cnt : 0);
    std::swap(lhs, rhs);
    exchanged = true;
  }
  while (l_idx > l_stop) {
    idx_to--; l_idx--;
    out->word_buf[idx_to] = lhs->word_buf[l_idx];
  }
  uint32_t carry = 0;
  while (l_idx > r_stop) {
    l_idx--; r_idx--; idx_to--;
    word_add(lhs->word_buf[l_idx], rhs->word_buf[r_idx], &carry, &out->word_buf[idx_to]);
  }
  if (l_int_word_cnt > r_int_word_cnt) {
    l_idx = (size_t)(l_int_word_cnt - r_int_word_cnt);
    if (exchanged) std::swap(lhs, rhs);
  } else {
    l_idx = (size_t)(r_int_word_cnt - l_int_word_cnt);
    if (!exchanged) std::swap(lhs, rhs);
  }
  while (l_idx > 0) {
    idx_to--; l_idx--;
    word_add(lhs->word_buf[l_idx], 0, &carry, &out->word_buf[idx_to]);
  }
  if (carry > 0) {
    idx_to--;
    out->word_buf[idx_to] = 1;
  }
  return rescode;
}

/* &a + &b: decimal.rs:2340-2353 */
int dec_add(const Decimal &a, const Decimal &b, Decimal *out) {
  uint8_t result_frac_cnt = std::max(a.result_frac_cnt, b.result_frac_cnt);
  int res;
  if (a.negative == b.negative) res = do_add(&a, &b, out);
  else res = do_sub(&a, &b, out);
  out->result_frac_cnt = result_frac_cnt;
  return res;
}

/* dec_cmp — via subtraction sign (decimal.rs PartialOrd impl uses do_sub) */
int dec_cmp(const Decimal &a, const Decimal &b) {
  if (a.negative == b.negative) {
    Decimal d;
    int carry_cls, frac_word_to; SubTmp l, r;
    calc_sub_carry(a, b, &carry_cls, &frac_word_to, &l, &r);
    (void)d;
    if (carry_cls < 0) return 0;
    /* carry 0 => |a|>|b|; carry 1 => |a|<|b| */
    int mag = carry_cls == 0 ? 1 : -1;
    return a.negative ? -mag : mag;
  }
  return a.negative ? -1 : 1;
}

/* write_decimal: decimal.rs:2022-2133 */
size_t dec_encode(const Decimal &d, uint8_t prec, uint8_t frac, uint8_t *out) {
  size_t written = 0;
  out[written++] = prec;
  out[written++] = frac;
  size_t payload_start = written;
  uint32_t mask = d.negative ? 0xFFFFFFFFu : 0;
  int int_cnt = prec - frac;
  int int_word_cnt = int_cnt / DIGITS_PER_WORD;
  int leading_digits = int_cnt - int_word_cnt * DIGITS_PER_WORD;
  int frac_word_cnt = frac / DIGITS_PER_WORD;
  int trailing_digits = frac - frac_word_cnt * DIGITS_PER_WORD;
  int src_frac_word_cnt = d.frac_cnt / DIGITS_PER_WORD;
  int src_trailing_digits = d.frac_cnt - src_frac_word_cnt * DIGITS_PER_WORD;
  int int_size = int_word_cnt * 4 + DIG_2_BYTES[leading_digits];
  int frac_size = frac_word_cnt * 4 + DIG_2_BYTES[trailing_digits];
  int src_frac_size = src_frac_word_cnt * 4 + DIG_2_BYTES[src_trailing_digits];

  size_t src_word_start_idx; uint8_t src_int_cnt;
  remove_leading_zeroes(d, d.int_cnt, &src_word_start_idx, &src_int_cnt);
  if (src_int_cnt + src_frac_size == 0) { mask = 0; int_cnt = 1; }

  int src_int_word_cnt = src_int_cnt / DIGITS_PER_WORD;
  int src_leading_digits = src_int_cnt - src_int_word_cnt * DIGITS_PER_WORD;
  int src_int_size = src_int_word_cnt * 4 + DIG_2_BYTES[src_leading_digits];

  size_t pw = 0;  /* payload bytes written (the "written" the macros track) */
  auto write_u8_m = [&](uint8_t b) {
    if (pw == 0) b ^= 0x80;
    out[payload_start + pw] = b; pw++;
  };
  auto write_word_m = [&](uint32_t word, int size) {
    uint8_t data[4];
    switch (size) {
      case 1: data[0] = (uint8_t)word; break;
      case 2: data[0] = (uint8_t)(word >> 8); data[1] = (uint8_t)word; break;
      case 3: data[0] = (uint8_t)(word >> 16); data[1] = (uint8_t)(word >> 8);
              data[2] = (uint8_t)word; break;
      default: data[0] = (uint8_t)(word >> 24); data[1] = (uint8_t)(word >> 16);
               data[2] = (uint8_t)(word >> 8); data[3] = (uint8_t)word; break;
    }
    if (pw == 0) data[0] ^= 0x80;
    memcpy(out + payload_start + pw, data, (size_t)size);
    pw += (size_t)size;
  };

  if (int_cnt < (int)src_int_cnt) {
    /* overflow arm: decimal.rs:2054-2070 */
    src_word_start_idx += (size_t)(src_int_word_cnt - int_word_cnt);
    if (src_leading_digits > 0) src_word_start_idx += 1;
    if (leading_digits > 0) src_word_start_idx -= 1;
    src_int_word_cnt = int_word_cnt;
    src_leading_digits = leading_digits;
  } else if (int_size > src_int_size) {
    for (int i = src_int_size; i < int_size; i++) write_u8_m((uint8_t)mask);
  }

  if (frac_size < src_frac_size) {
    src_frac_word_cnt = frac_word_cnt;
    src_trailing_digits = trailing_digits;
  } else if (frac_size > src_frac_size && src_trailing_digits > 0) {
    if (frac_word_cnt == src_frac_word_cnt) {
      src_trailing_digits = trailing_digits;
      frac_size = src_frac_size;
    } else {
      src_frac_word_cnt += 1;
      src_trailing_digits = 0;
    }
  }

  if (src_leading_digits > 0) {
    int i = DIG_2_BYTES[src_leading_digits];
    uint32_t x = (d.word_buf[src_word_start_idx] % TEN_POW[src_leading_digits]) ^ mask;
    src_word_start_idx += 1;
    write_word_m(x, i);
  }
  size_t stop = src_word_start_idx + (size_t)src_int_word_cnt + (size_t)src_frac_word_cnt;
  while (src_word_start_idx < stop) {
    write_word_m(d.word_buf[src_word_start_idx] ^ mask, 4);
    src_word_start_idx++;
  }
  if (src_trailing_digits > 0) {
    int i = DIG_2_BYTES[src_trailing_digits];
    int lim = (src_frac_word_cnt < frac_word_cnt) ? DIGITS_PER_WORD : trailing_digits;
    while (src_trailing_digits < lim && DIG_2_BYTES[src_trailing_digits] == i)
      src_trailing_digits++;
    uint32_t x = (d.word_buf[src_word_start_idx] /
                  TEN_POW[DIGITS_PER_WORD - src_trailing_digits]) ^ mask;
    write_word_m(x, i);
  }
  if (frac_size > src_frac_size) {
    size_t target = (size_t)(int_size + frac_size);
    for (int i = src_frac_size; i < frac_size && pw < target; i++)
      write_u8_m((uint8_t)mask);
  }
  return payload_start + pw;
}

/* dec_encoded_len: decimal.rs:169-192 */
bool dec_encoded_len(const uint8_t *p, size_t len, size_t *elen) {
  if (len < 2) return false;
  uint8_t precision = p[0], frac_cnt = p[1];
  if (precision < frac_cnt) return false;
  int int_cnt = precision - frac_cnt;
  int int_word_cnt = int_cnt / DIGITS_PER_WORD;
  int frac_word_cnt = frac_cnt / DIGITS_PER_WORD;
  int int_left = int_cnt - int_word_cnt * DIGITS_PER_WORD;
  int frac_left = frac_cnt - frac_word_cnt * DIGITS_PER_WORD;
  *elen = (size_t)(int_word_cnt * 4 + DIG_2_BYTES[int_left] +
                   frac_word_cnt * 4 + DIG_2_BYTES[frac_left] + 2);
  return true;
}

/* read_word: decimal.rs:2160-2200 */
static bool read_word(const uint8_t *&p, size_t &rem, int size, bool &is_first, uint32_t *out) {
  if ((size_t)size > rem) return false;
  uint8_t first = p[0];
  if (is_first) { first ^= 0x80; is_first = false; }
  uint32_t res;
  switch (size) {
    case 1: res = (uint32_t)(int32_t)(int8_t)first; break;
    case 2: res = (uint32_t)(((int32_t)(int8_t)first << 8) + (int32_t)p[1]); break;
    case 3:
      if (first & 128)
        res = (255u << 24) | ((uint32_t)first << 16) | ((uint32_t)p[1] << 8) | (uint32_t)p[2];
      else
        res = ((uint32_t)first << 16) | ((uint32_t)p[1] << 8) | (uint32_t)p[2];
      break;
    default:
      res = (uint32_t)(((int32_t)(int8_t)first << 24) + ((int32_t)p[1] << 16) +
                       ((int32_t)p[2] << 8) + (int32_t)p[3]);
      break;
  }
  p += size; rem -= (size_t)size;
  *out = res;
  return true;
}

/* read_decimal: decimal.rs:2204-2289 */
bool dec_decode(const uint8_t *p0, size_t len, Decimal *d, size_t *consumed) {
  const uint8_t *p = p0;
  size_t rem = len;
  if (rem < 3) return false;
  uint8_t prec = p[0], frac_cnt = p[1];
  p += 2; rem -= 2;
  if (prec < frac_cnt) return false;
  int int_cnt = prec - frac_cnt;
  int int_word_cnt = int_cnt / DIGITS_PER_WORD;
  int leading_digits = int_cnt - int_word_cnt * DIGITS_PER_WORD;
  int frac_word_cnt = frac_cnt / DIGITS_PER_WORD;
  int trailing_digits = frac_cnt - frac_word_cnt * DIGITS_PER_WORD;
  int int_word_to = int_word_cnt + (leading_digits > 0 ? 1 : 0);
  int frac_word_to = frac_word_cnt + (trailing_digits > 0 ? 1 : 0);
  uint32_t mask = (p[0] & 0x80) ? 0 : 0xFFFFFFFFu;
  int iw, fw;
  if (fix_word_cnt_err(int_word_to, frac_word_to, &iw, &fw) != 0) return false;
  *d = dec_new((uint8_t)int_cnt, frac_cnt, mask != 0);
  d->result_frac_cnt = frac_cnt;
  size_t word_idx = 0;
  bool is_first = true;
  if (leading_digits > 0) {
    int i = DIG_2_BYTES[leading_digits];
    if (!read_word(p, rem, i, is_first, &d->word_buf[word_idx])) return false;
    d->word_buf[word_idx] ^= mask;
    if (d->word_buf[word_idx] >= TEN_POW[leading_digits + 1]) return false;
    if (d->word_buf[word_idx] != 0) word_idx++;
    else d->int_cnt -= (uint8_t)leading_digits;
  }
  for (int k = 0; k < int_word_cnt; k++) {
    if (!read_word(p, rem, 4, is_first, &d->word_buf[word_idx])) return false;
    d->word_buf[word_idx] ^= mask;
    if (d->word_buf[word_idx] > WORD_MAX) return false;
    if (word_idx > 0 || d->word_buf[word_idx] != 0) word_idx++;
    else d->int_cnt -= DIGITS_PER_WORD;
  }
  for (int k = 0; k < frac_word_cnt; k++) {
    if (!read_word(p, rem, 4, is_first, &d->word_buf[word_idx])) return false;
    d->word_buf[word_idx] ^= mask;
    if (d->word_buf[word_idx] > WORD_MAX) return false;
    word_idx++;
  }
  if (trailing_digits > 0) {
    uint32_t x;
    if (!read_word(p, rem, DIG_2_BYTES[trailing_digits], is_first, &x)) return false;
    x ^= mask;
    uint64_t v = (uint64_t)x * TEN_POW[DIGITS_PER_WORD - trailing_digits];
    if (v > WORD_MAX) return false;
    d->word_buf[word_idx] = (uint32_t)v;
  }
  if (d->int_cnt == 0 && d->frac_cnt == 0) {
    uint8_t rfc = d->result_frac_cnt;
    *d = dec_zero();
    d->result_frac_cnt = rfc;
  }
  d->result_frac_cnt = frac_cnt;
  *consumed = (size_t)(p - p0);
  return true;
}

std::string dec_to_string(const Decimal &d) {
  std::string s;
  if (d.negative) s += '-';
  int iwc = word_cnt(d.int_cnt), fwc = word_cnt(d.frac_cnt);
  bool started = false;
  char buf[16];
  for (int i = 0; i < iwc; i++) {
    if (!started && d.word_buf[i] == 0 && i + 1 < iwc) continue;
    snprintf(buf, sizeof buf, started ? "%09u" : "%u", d.word_buf[i]);
    s += buf; started = true;
  }
  if (!started) s += '0';
  if (d.frac_cnt > 0) {
    s += '.';
    std::string fs;
    for (int i = 0; i < fwc; i++) {
      snprintf(buf, sizeof buf, "%09u", d.word_buf[iwc + i]);
      fs += buf;
    }
    fs.resize(d.frac_cnt, '0');
    s += fs;
  }
  return s;
}

/* ================== CRC-64/XZ ================== */
/* Published algorithm (crc64fast 0.1.0 implements CRC-64/XZ):
 * reflected, poly 0x42F0E1EBA9EA3693, init 0xFFFF..., xorout 0xFFFF... */
static uint64_t crc64_table[8][256];
static bool crc64_init_done = false;
static void crc64_build_tables() {
  const uint64_t POLY = 0x42F0E1EBA9EA3693ull;
  /* reflected poly */
  uint64_t rpoly = 0;
  for (int i = 0; i < 64; i++)
    if (POLY & (1ull << i)) rpoly |= 1ull << (63 - i);
  for (int i = 0; i < 256; i++) {
    uint64_t crc = (uint64_t)i;
    for (int j = 0; j < 8; j++)
      crc = (crc >> 1) ^ ((crc & 1) ? rpoly : 0);
    crc64_table[0][i] = crc;
  }
  for (int t = 1; t < 8; t++)
    for (int i = 0; i < 256; i++)
      crc64_table[t][i] = crc64_table[0][crc64_table[t-1][i] & 0xFF] ^ (crc64_table[t-1][i] >> 8);
  crc64_init_done = true;
}
uint64_t crc64_xz_init() {
  if (!crc64_init_done) crc64_build_tables();
  return ~0ull;
}
uint64_t crc64_xz_update(uint64_t crc, const uint8_t *p, size_t len) {
  if (!crc64_init_done) crc64_build_tables();
  /* slice-by-8 */
  while (len >= 8) {
    uint64_t x;
    memcpy(&x, p, 8);
    crc ^= x;     /* little-endian hosts only (x86/amdgpu) */
    crc = crc64_table[7][crc & 0xFF] ^ crc64_table[6][(crc >> 8) & 0xFF] ^
          crc64_table[5][(crc >> 16) & 0xFF] ^ crc64_table[4][(crc >> 24) & 0xFF] ^
          crc64_table[3][(crc >> 32) & 0xFF] ^ crc64_table[2][(crc >> 40) & 0xFF] ^
          crc64_table[1][(crc >> 48) & 0xFF] ^ crc64_table[0][(crc >> 56) & 0xFF];
    p += 8; len -= 8;
  }
  while (len--) crc = crc64_table[0][(crc ^ *p++) & 0xFF] ^ (crc >> 8);
  return crc;
}
uint64_t crc64_xz_finish(uint64_t state) { return ~state; }
uint64_t crc64_xz(const uint8_t *p, size_t len) {
  return crc64_xz_finish(crc64_xz_update(crc64_xz_init(), p, len));
}

/* ================== row v2 ================== */
static inline uint16_t rd_u16le(const uint8_t *p) { return (uint16_t)(p[0] | (p[1] << 8)); }
static inline uint32_t rd_u32le(const uint8_t *p) {
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) | ((uint32_t)p[3] << 24);
}

/* row_slice.rs:76-117 + mod.rs:8-15 (CODEC_VERSION=128, BIG=1, WITH_CHECKSUM=2) */
bool row_v2_parse(const uint8_t *p, size_t len, RowSliceV2 *rs) {
  if (len < 6 || p[0] != 128) return false;
  uint8_t flags = p[1];
  rs->big = (flags & 1) != 0;
  bool with_checksum = (flags & 2) != 0;
  rs->non_null_cnt = rd_u16le(p + 2);
  rs->null_cnt = rd_u16le(p + 4);
  size_t pos = 6;
  size_t idw = rs->big ? 4 : 1;
  size_t offw = rs->big ? 4 : 2;
  if (pos + idw * rs->non_null_cnt + idw * rs->null_cnt + offw * rs->non_null_cnt > len)
    return false;
  rs->non_null_ids = p + pos; pos += idw * rs->non_null_cnt;
  rs->null_ids = p + pos;     pos += idw * rs->null_cnt;
  rs->offsets = p + pos;      pos += offw * rs->non_null_cnt;
  rs->values = p + pos;
  rs->values_len = (uint32_t)(len - pos);
  if (with_checksum) {
    /* checksum trailer (5 or 9 B) sits after the last value byte
       (row_slice.rs:105-117): values end at offsets[last]. */
    uint32_t vend = 0;
    if (rs->non_null_cnt > 0) {
      const uint8_t *o = rs->offsets + offw * (rs->non_null_cnt - 1);
      vend = rs->big ? rd_u32le(o) : rd_u16le(o);
    }
    if (vend > rs->values_len) return false;
    rs->values_len = vend;
  }
  return true;
}

static bool rv2_id_at(const RowSliceV2 &rs, const uint8_t *ids, uint16_t i, uint32_t *id) {
  if (rs.big) *id = rd_u32le(ids + 4 * i);
  else *id = ids[i];
  return true;
}

/* search_in_non_null_ids: row_slice.rs:125-168 (ids sorted; binary search).
 * id_valid: 0 < id <= max of id width (row_slice.rs:188-196). */
bool row_v2_find(const RowSliceV2 &rs, int64_t col_id, uint32_t *start, uint32_t *end) {
  if (col_id <= 0) return false;
  if (!rs.big && col_id > 255) return false;
  if (rs.big && col_id > 0xFFFFFFFFll) return false;
  uint32_t target = (uint32_t)col_id;
  int lo = 0, hi = (int)rs.non_null_cnt - 1;
  while (lo <= hi) {
    int mid = (lo + hi) / 2;
    uint32_t v; rv2_id_at(rs, rs.non_null_ids, (uint16_t)mid, &v);
    if (v == target) {
      size_t offw = rs.big ? 4 : 2;
      const uint8_t *o = rs.offsets + offw * mid;
      uint32_t off = rs.big ? rd_u32le(o) : rd_u16le(o);
      uint32_t st = 0;
      if (mid > 0) {
        const uint8_t *po = rs.offsets + offw * (mid - 1);
        st = rs.big ? rd_u32le(po) : rd_u16le(po);
      }
      *start = st; *end = off;
      return off <= rs.values_len && st <= off;
    }
    if (v < target) lo = mid + 1; else hi = mid - 1;
  }
  return false;
}

bool row_v2_is_null(const RowSliceV2 &rs, int64_t col_id) {
  if (col_id <= 0) return false;
  if (!rs.big && col_id > 255) return false;
  uint32_t target = (uint32_t)col_id;
  int lo = 0, hi = (int)rs.null_cnt - 1;
  while (lo <= hi) {
    int mid = (lo + hi) / 2;
    uint32_t v; rv2_id_at(rs, rs.null_ids, (uint16_t)mid, &v);
    if (v == target) return true;
    if (v < target) lo = mid + 1; else hi = mid - 1;
  }
  return false;
}

/* compat_v1.rs:12-38 + write_v2_as_datum (compat_v1.rs:54-126) */
static bool rv2_decode_i64(const uint8_t *v, size_t n, int64_t *out) {
  switch (n) {
    case 1: *out = (int64_t)(int8_t)v[0]; return true;
    case 2: *out = (int64_t)(int16_t)rd_u16le(v); return true;
    case 4: *out = (int64_t)(int32_t)rd_u32le(v); return true;
    case 8: { uint64_t x = (uint64_t)rd_u32le(v) | ((uint64_t)rd_u32le(v + 4) << 32);
              *out = (int64_t)x; return true; }
    default: return false;
  }
}
static bool rv2_decode_u64(const uint8_t *v, size_t n, uint64_t *out) {
  switch (n) {
    case 1: *out = v[0]; return true;
    case 2: *out = rd_u16le(v); return true;
    case 4: *out = rd_u32le(v); return true;
    case 8: *out = (uint64_t)rd_u32le(v) | ((uint64_t)rd_u32le(v + 4) << 32); return true;
    default: return false;
  }
}

bool row_v2_cell_to_v1_datum(const uint8_t *cell, size_t cell_len,
                             int32_t tp, uint32_t ft_flag,
                             std::vector<uint8_t> *out) {
  uint8_t tmp[16];
  switch (tp) {
    case 1: case 2: case 9: case 3: case 8: {  /* Tiny/Short/Int24/Long/LongLong */
      if (ft_flag & (1u << 5)) {               /* unsigned -> UINT datum */
        uint64_t u;
        if (!rv2_decode_u64(cell, cell_len, &u)) return false;
        out->push_back(UINT_FLAG);
        encode_comparable_u64(tmp, u);
        out->insert(out->end(), tmp, tmp + 8);
      } else {
        int64_t i;
        if (!rv2_decode_i64(cell, cell_len, &i)) return false;
        out->push_back(INT_FLAG);
        encode_comparable_i64(tmp, i);
        out->insert(out->end(), tmp, tmp + 8);
      }
      return true;
    }
    case 4: case 5:                            /* Float/Double: payload as-is */
      out->push_back(FLOAT_FLAG);
      out->insert(out->end(), cell, cell + cell_len);
      return true;
    case 15: case 0xfd: case 0xfe: case 0xfc: { /* VarChar/VarString/String/Blob */
      out->push_back(COMPACT_BYTES_FLAG);
      uint8_t hdr[10];
      size_t n = encode_var_i64(hdr, (int64_t)cell_len);
      out->insert(out->end(), hdr, hdr + n);
      out->insert(out->end(), cell, cell + cell_len);
      return true;
    }
    case 0xf6:                                 /* NewDecimal: payload as-is */
      out->push_back(DECIMAL_FLAG);
      out->insert(out->end(), cell, cell + cell_len);
      return true;
    case 11: {                                 /* Duration -> DURATION datum */
      int64_t i;
      if (!rv2_decode_i64(cell, cell_len, &i)) return false;
      out->push_back(DURATION_FLAG);
      encode_comparable_i64(tmp, i);
      out->insert(out->end(), tmp, tmp + 8);
      return true;
    }
    case 10: case 12: case 7: {                /* Date/DateTime/Timestamp -> UINT */
      uint64_t u;
      if (!rv2_decode_u64(cell, cell_len, &u)) return false;
      out->push_back(UINT_FLAG);
      encode_comparable_u64(tmp, u);
      out->insert(out->end(), tmp, tmp + 8);
      return true;
    }
    case 6:                                    /* Null */
      out->push_back(NIL_FLAG);
      return true;
    default:
      return false;
  }
}

} // namespace orc
