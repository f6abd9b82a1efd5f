/* prod_decimal.h — PRODUCT host-side MySQL Decimal support.
 *
 * The engine's own (oracle-independent) implementation of the TiKV Decimal
 * representation and binary codec, used by the response encoder and the
 * fixture generator. Follows tidb_query_datatype/src/codec/mysql/decimal.rs:
 * struct :927-942, word constants :132-134, write_decimal :2022-2133,
 * read_decimal :2204-2289, prec_and_frac :1043-1051.
 *
 * The GPU kernels do NOT use this type: decimals whose digit count fits 18
 * digits travel through kernels as scaled integers (see kernels.hip); this
 * type materializes final aggregate values for the datum response.
 */
#ifndef PROD_DECIMAL_H
#define PROD_DECIMAL_H

#include <stdint.h>
#include <stddef.h>

namespace prod {

struct PDec {
  uint8_t int_cnt = 1, frac_cnt = 0;
  bool negative = false;
  uint32_t word_buf[9] = {0};
};

/* Build from a scaled two's-complement 128-bit integer: value = scaled / 10^frac.
 * Exact for |scaled| < 10^81 (word_buf capacity); callers stay far below. */
PDec pdec_from_scaled_i128(__int128 scaled, uint8_t frac);

/* 256-bit variant (wide Decimal sums): limbs = little-endian two's
 * complement. *ovf set when the magnitude exceeds the word_buf capacity
 * (the reference's Res::Overflow). */
PDec pdec_from_scaled_i256(const uint64_t limbs[4], uint8_t frac,
                           bool *ovf = nullptr);

/* least (prec, frac) encoding this value completely (prec_and_frac) */
void pdec_prec_and_frac(const PDec &d, uint8_t *prec, uint8_t *frac);

/* write_decimal with [prec][frac] header; out must hold >= 42 bytes;
 * returns bytes written */
size_t pdec_encode(const PDec &d, uint8_t prec, uint8_t frac, uint8_t *out);

/* parse an encoded decimal ([prec][frac] header form) into a scaled integer
 * (value * 10^frac_out). Returns false if it does not fit 18 digits.
 * consumed = total encoded length. */
bool pdec_decode_scaled(const uint8_t *p, size_t len, int64_t *scaled,
                        uint8_t *frac_out, size_t *consumed);
bool pdec_encoded_len(const uint8_t *p, size_t len, size_t *elen);

/* full read_decimal (decimal.rs:2204-2289) into the word representation;
 * result_frac = the encoded frac count (result_frac_cnt in the struct dump) */
bool pdec_decode(const uint8_t *p, size_t len, PDec *d, uint8_t *result_frac,
                 size_t *consumed);

}  // namespace prod
#endif
