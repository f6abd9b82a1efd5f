/* prod_decimal.cpp — PRODUCT host-side Decimal. See prod_decimal.h. */
#include "prod_decimal.h"
#include <cstring>

namespace prod {

static const int DIGITS_PER_WORD = 9;
static const uint32_t WORD_BASE = 1000000000u;
static const uint32_t TEN_POW[10] = {1, 10, 100, 1000, 10000, 100000,
                                     1000000, 10000000, 100000000, 1000000000};
static const uint8_t DIG_2_BYTES[10] = {0, 1, 1, 2, 2, 3, 3, 4, 4, 4};

PDec pdec_from_scaled_i128(__int128 scaled, uint8_t frac) {
  PDec d;
  if (scaled == 0) {
    /* a running decimal sum that cancels to exactly zero passes through
       do_sub's equal branch and becomes Decimal::zero() (int_cnt 1, frac 0,
       positive) — decimal.rs:348-350,996. (A sum of literal +0.00 inputs
       would keep frac — not distinguished here; see DESIGN.md.) */
    d.int_cnt = 1; d.frac_cnt = 0; d.negative = false;
    return d;
  }
  d.negative = scaled < 0;
  unsigned __int128 mag = d.negative ? (unsigned __int128)(-scaled) : (unsigned __int128)scaled;
  /* split into frac words (low `frac` digits, padded to whole words at the
     fraction's word grid) and int words */
  int frac_words = (frac + DIGITS_PER_WORD - 1) / DIGITS_PER_WORD;
  uint32_t fw[9] = {0};
  /* low digit block sits in the LAST frac word's high digits:
     a decimal 12.3 with frac=2 has word layout int:[12] frac:[300000000] */
  unsigned __int128 v = mag;
  for (int w = frac_words - 1; w >= 0; w--) {
    int dig = frac - w * DIGITS_PER_WORD;         /* digits in this word */
    if (dig > DIGITS_PER_WORD) dig = DIGITS_PER_WORD;
    uint32_t chunk = (uint32_t)(v % TEN_POW[dig]);
    v /= TEN_POW[dig];
    fw[w] = chunk * TEN_POW[DIGITS_PER_WORD - dig];
  }
  /* remaining v = integer part */
  uint32_t iw[9] = {0};
  int int_words = 0;
  while (v > 0) { iw[int_words++] = (uint32_t)(v % WORD_BASE); v /= WORD_BASE; }
  if (int_words == 0) int_words = 1;  /* at least one zero int word */
  d.int_cnt = (uint8_t)(int_words * DIGITS_PER_WORD);
  d.frac_cnt = frac;
  for (int i = 0; i < int_words; i++) d.word_buf[i] = iw[int_words - 1 - i];
  for (int w = 0; w < frac_words; w++) d.word_buf[int_words + w] = fw[w];
  return d;
}

namespace {
/* 256-bit unsigned helper for the wide-sum path: long division by a small
 * base extracts the decimal digits the word_buf wants */
struct U256 {
  uint64_t l[4];
  bool is_zero() const { return !(l[0] | l[1] | l[2] | l[3]); }
  uint32_t divmod(uint32_t div) {
    unsigned __int128 rem = 0;
    for (int i = 3; i >= 0; i--) {
      unsigned __int128 cur = (rem << 64) | l[i];
      l[i] = (uint64_t)(cur / div);
      rem = cur % div;
    }
    return (uint32_t)rem;
  }
};
}  // namespace

PDec pdec_from_scaled_i256(const uint64_t limbs[4], uint8_t frac, bool *ovf) {
  if (ovf) *ovf = false;
  uint64_t sgn = (limbs[3] >> 63) ? ~0ull : 0ull;
  if (limbs[2] == sgn && limbs[3] == sgn) {   /* fits i128 */
    __int128 v =
        (__int128)(((unsigned __int128)limbs[1] << 64) | limbs[0]);
    return pdec_from_scaled_i128(v, frac);
  }
  PDec d;
  d.negative = sgn != 0;
  U256 mag{{limbs[0], limbs[1], limbs[2], limbs[3]}};
  if (d.negative) {                            /* two's-complement negate */
    unsigned __int128 c = 1;
    for (int i = 0; i < 4; i++) {
      unsigned __int128 t = (unsigned __int128)(uint64_t)~mag.l[i] + c;
      mag.l[i] = (uint64_t)t;
      c = t >> 64;
    }
  }
  int frac_words = (frac + DIGITS_PER_WORD - 1) / DIGITS_PER_WORD;
  uint32_t fw[9] = {0};
  for (int w = frac_words - 1; w >= 0; w--) {
    int dig = frac - w * DIGITS_PER_WORD;
    if (dig > DIGITS_PER_WORD) dig = DIGITS_PER_WORD;
    uint32_t chunk = mag.divmod(TEN_POW[dig]);
    fw[w] = chunk * TEN_POW[DIGITS_PER_WORD - dig];
  }
  uint32_t iw[9] = {0};
  int int_words = 0;
  while (!mag.is_zero()) {
    if (int_words + frac_words >= 9) {
      /* beyond word_buf capacity (81 digits): the reference's Res::Overflow
         — callers surface it as an error */
      if (ovf) *ovf = true;
      return d;
    }
    iw[int_words++] = mag.divmod(WORD_BASE);
  }
  if (int_words == 0) int_words = 1;
  d.int_cnt = (uint8_t)(int_words * DIGITS_PER_WORD);
  d.frac_cnt = frac;
  for (int i = 0; i < int_words; i++) d.word_buf[i] = iw[int_words - 1 - i];
  for (int w = 0; w < frac_words; w++) d.word_buf[int_words + w] = fw[w];
  return d;
}

static uint8_t count_leading_zeroes(uint8_t i, uint32_t word) {
  uint8_t c = 0;
  while (TEN_POW[i] > word) { i--; c++; }
  return c;
}
static void remove_leading_zeroes(const PDec &d, uint8_t prec,
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

void pdec_prec_and_frac(const PDec &d, uint8_t *prec, uint8_t *frac) {
  size_t wi; uint8_t int_cnt;
  remove_leading_zeroes(d, d.int_cnt, &wi, &int_cnt);
  uint8_t p = int_cnt + d.frac_cnt;
  *prec = p == 0 ? 1 : p;
  *frac = d.frac_cnt;
}

size_t pdec_encode(const PDec &d, uint8_t prec, uint8_t frac, uint8_t *out) {
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

  size_t pw = 0;
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

bool pdec_encoded_len(const uint8_t *p, size_t len, size_t *elen) {
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

bool pdec_decode_scaled(const uint8_t *p0, size_t len, int64_t *scaled,
                        uint8_t *frac_out, size_t *consumed) {
  const uint8_t *p = p0;
  size_t rem = len;
  if (rem < 3) return false;
  uint8_t prec = p[0], frac_cnt = p[1];
  p += 2; rem -= 2;
  if (prec < frac_cnt || prec > 18) return false;   /* scaled-i64 fast path */
  int int_cnt = prec - frac_cnt;
  int int_word_cnt = int_cnt / DIGITS_PER_WORD;
  int leading_digits = int_cnt - int_word_cnt * DIGITS_PER_WORD;
  int frac_word_cnt = frac_cnt / DIGITS_PER_WORD;
  int trailing_digits = frac_cnt - frac_word_cnt * DIGITS_PER_WORD;
  uint32_t mask = (p[0] & 0x80) ? 0 : 0xFFFFFFFFu;
  bool negative = mask != 0;
  bool is_first = true;
  /* accumulate digits most-significant first */
  unsigned __int128 acc = 0;
  if (leading_digits > 0) {
    uint32_t w;
    if (!read_word(p, rem, DIG_2_BYTES[leading_digits], is_first, &w)) return false;
    w ^= mask;
    if (w >= TEN_POW[leading_digits + 1]) return false;
    acc = w;
  }
  for (int k = 0; k < int_word_cnt; k++) {
    uint32_t w;
    if (!read_word(p, rem, 4, is_first, &w)) return false;
    w ^= mask;
    if (w > WORD_BASE - 1) return false;
    acc = acc * WORD_BASE + w;
  }
  for (int k = 0; k < frac_word_cnt; k++) {
    uint32_t w;
    if (!read_word(p, rem, 4, is_first, &w)) return false;
    w ^= mask;
    if (w > WORD_BASE - 1) return false;
    acc = acc * WORD_BASE + w;
  }
  if (trailing_digits > 0) {
    uint32_t w;
    if (!read_word(p, rem, DIG_2_BYTES[trailing_digits], is_first, &w)) return false;
    w ^= mask;
    if (w >= TEN_POW[trailing_digits]) return false;
    acc = acc * TEN_POW[trailing_digits] + w;
  }
  if (acc > (unsigned __int128)INT64_MAX) return false;
  *scaled = negative ? -(int64_t)acc : (int64_t)acc;
  *frac_out = frac_cnt;
  *consumed = (size_t)(p - p0);
  return true;
}


/* full read_decimal (decimal.rs:2204-2289): parse a [prec][frac] encoded
 * decimal into the in-memory word representation, with the reference's
 * normalization (leading zero words reduce int_cnt; trailing digits scale
 * up to a full word; all-zero collapses to zero). Used by the TypeChunk
 * response encoder, which dumps this struct (decimal.rs:2135-2142). */
bool pdec_decode(const uint8_t *p0, size_t len, PDec *d, uint8_t *result_frac,
                 size_t *consumed) {
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
  int tot_words = int_word_cnt + (leading_digits ? 1 : 0) + frac_word_cnt +
                  (trailing_digits ? 1 : 0);
  if (tot_words > 9) return false;
  uint32_t mask = (p[0] & 0x80) ? 0 : 0xFFFFFFFFu;
  *d = PDec{};
  d->int_cnt = (uint8_t)int_cnt;
  d->frac_cnt = frac_cnt;
  d->negative = mask != 0;
  bool is_first = true;
  int word_idx = 0;
  if (leading_digits > 0) {
    uint32_t w;
    if (!read_word(p, rem, DIG_2_BYTES[leading_digits], is_first, &w))
      return false;
    w ^= mask;
    if (w >= TEN_POW[leading_digits + 1]) return false;
    d->word_buf[word_idx] = w;
    if (w != 0) word_idx++;
    else d->int_cnt -= (uint8_t)leading_digits;
  }
  for (int k = 0; k < int_word_cnt; k++) {
    uint32_t w;
    if (!read_word(p, rem, 4, is_first, &w)) return false;
    w ^= mask;
    if (w > WORD_BASE - 1) return false;
    d->word_buf[word_idx] = w;
    if (word_idx > 0 || w != 0) word_idx++;
    else d->int_cnt -= (uint8_t)DIGITS_PER_WORD;
  }
  for (int k = 0; k < frac_word_cnt; k++) {
    uint32_t w;
    if (!read_word(p, rem, 4, is_first, &w)) return false;
    w ^= mask;
    if (w > WORD_BASE - 1) return false;
    d->word_buf[word_idx++] = w;
  }
  if (trailing_digits > 0) {
    uint32_t w;
    if (!read_word(p, rem, DIG_2_BYTES[trailing_digits], is_first, &w))
      return false;
    w ^= mask;
    unsigned long long v =
        (unsigned long long)w * TEN_POW[DIGITS_PER_WORD - trailing_digits];
    if (v > WORD_BASE - 1) return false;
    d->word_buf[word_idx] = (uint32_t)v;
  }
  if (d->int_cnt == 0 && d->frac_cnt == 0) {
    /* Decimal::zero() (decimal.rs:996) */
    *d = PDec{};
    d->int_cnt = 1;
    d->negative = false;
  }
  *result_frac = frac_cnt;
  *consumed = (size_t)(p - p0);
  return true;
}

}  // namespace prod
