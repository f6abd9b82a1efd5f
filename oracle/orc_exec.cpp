/* orc_exec.cpp — ORACLE (test infrastructure ONLY).
 *
 * CPU restatement of the reference batch-executor pipeline over the
 * CoprDagRequest descriptor:
 *   - runner loop:      runner.rs:840,986-1242 (batch 32 ->x2-> 1024)
 *   - table scan:       util/scan_executor.rs:226,114 +
 *                       table_scan_executor.rs:210-291,375-485
 *   - selection:        selection_executor.rs:81-195
 *   - simple agg:       simple_aggr_executor.rs
 *   - fast hash agg:    fast_hash_aggr_executor.rs:226-440
 *   - aggregates:       impl_count.rs / impl_sum.rs / impl_avg.rs /
 *                       impl_max_min.rs / impl_first.rs / impl_bit_op.rs
 *   - limit:            limit_executor.rs
 *   - response encode:  runner.rs:1188 -> lazy_column_vec.rs:172 ->
 *                       lazy_column.rs:242 / vector.rs:362 / datum_codec.rs:248-294
 *   - checksum:         src/coprocessor/checksum.rs:59-114
 */
#include "oracle.h"
#include "orc_codec.h"

#include <cstring>
#include <cstdio>
#include <dlfcn.h>
#include <unordered_map>
#include <map>
#include <vector>
#include <string>
#include <memory>
#include <algorithm>

using namespace orc;

static thread_local std::string g_err;
extern "C" const char *orc_last_error(void) { return g_err.c_str(); }
#define FAIL(msg) do { g_err = (msg); return false; } while (0)

namespace {

/* ---------------- eval types ---------------- */
enum class ET { Int, Real, Decimal, Bytes };

/* EvalType::try_from(FieldTypeTp) — tidb_query_datatype eval type mapping */
static bool et_of_tp(int32_t tp, ET *et) {
  switch (tp) {
    case COPR_TP_TINY: case COPR_TP_SHORT: case COPR_TP_INT24:
    case COPR_TP_LONG: case COPR_TP_LONGLONG: case COPR_TP_YEAR:
      *et = ET::Int; return true;
    case COPR_TP_FLOAT: case COPR_TP_DOUBLE:
      *et = ET::Real; return true;
    case COPR_TP_NEWDECIMAL:
      *et = ET::Decimal; return true;
    case COPR_TP_VARCHAR: case COPR_TP_VARSTRING: case COPR_TP_STRING:
    case COPR_TP_BLOB:
      *et = ET::Bytes; return true;
    default:
      return false;
  }
}

static bool is_unsigned(const CoprFieldType &ft) { return (ft.flag & COPR_FLAG_UNSIGNED) != 0; }

/* A typed vector with null mask (VectorValue, data_type/vector.rs:15) */
struct TypedVec {
  ET et = ET::Int;
  std::vector<uint8_t> nulls;      /* 1 = NULL */
  std::vector<int64_t> i;
  std::vector<double> r;
  std::vector<Decimal> d;
  std::vector<std::string> b;
  size_t size() const { return nulls.size(); }
  void push_null() {
    nulls.push_back(1);
    switch (et) {
      case ET::Int: i.push_back(0); break;
      case ET::Real: r.push_back(0); break;
      case ET::Decimal: d.push_back(dec_zero()); break;
      case ET::Bytes: b.emplace_back(); break;
    }
  }
  void push_int(int64_t v) { nulls.push_back(0); i.push_back(v); }
  void push_real(double v) { nulls.push_back(0); r.push_back(v); }
  void push_dec(const Decimal &v) { nulls.push_back(0); d.push_back(v); }
  void push_bytes(std::string v) { nulls.push_back(0); b.push_back(std::move(v)); }
};

/* LazyBatchColumn (codec/batch/lazy_column.rs:28): Raw or Decoded */
struct Column {
  bool raw = true;
  std::vector<std::string> raw_vals;  /* raw datum bytes (flag+payload) */
  TypedVec dec;
  size_t size() const { return raw ? raw_vals.size() : dec.size(); }
};

/* decode_int_datum (datum_codec.rs:401-420) */
static bool decode_int_datum(const uint8_t *p, size_t len, bool *is_null, int64_t *v) {
  if (len == 0) FAIL("empty datum");
  uint8_t flag = p[0]; p++; len--;
  *is_null = false;
  switch (flag) {
    case NIL_FLAG: *is_null = true; return true;
    case INT_FLAG: if (len < 8) FAIL("short INT"); *v = decode_comparable_i64(p); return true;
    case UINT_FLAG: if (len < 8) FAIL("short UINT"); *v = (int64_t)decode_comparable_u64(p); return true;
    case VAR_INT_FLAG: { size_t n; if (!decode_var_i64(p, len, v, &n)) FAIL("bad varint"); return true; }
    case VAR_UINT_FLAG: { uint64_t u; size_t n; if (!decode_var_u64(p, len, &u, &n)) FAIL("bad varuint");
                          *v = (int64_t)u; return true; }
    default: FAIL("unsupported datum flag for Int vector");
  }
}
/* decode_real_datum (datum_codec.rs:423-446) */
static bool decode_real_datum(const uint8_t *p, size_t len, int32_t tp, bool *is_null, double *v) {
  if (len == 0) FAIL("empty datum");
  uint8_t flag = p[0]; p++; len--;
  *is_null = false;
  switch (flag) {
    case NIL_FLAG: *is_null = true; return true;
    case FLOAT_FLAG: {
      if (len < 8) FAIL("short FLOAT");
      double x = decode_comparable_f64(p);
      if (tp == COPR_TP_FLOAT) x = (double)(float)x;
      *v = x; return true;
    }
    default: FAIL("unsupported datum flag for Real vector");
  }
}
/* decode_decimal_datum (datum_codec.rs:448-465) */
static bool decode_decimal_datum(const uint8_t *p, size_t len, bool *is_null, Decimal *v) {
  if (len == 0) FAIL("empty datum");
  uint8_t flag = p[0]; p++; len--;
  *is_null = false;
  switch (flag) {
    case NIL_FLAG: *is_null = true; return true;
    case DECIMAL_FLAG: { size_t c; if (!dec_decode(p, len, v, &c)) FAIL("bad decimal"); return true; }
    default: FAIL("unsupported datum flag for Decimal vector");
  }
}
/* decode_bytes_datum (datum_codec.rs:467-486) */
static bool decode_bytes_datum(const uint8_t *p, size_t len, bool *is_null, std::string *v) {
  if (len == 0) FAIL("empty datum");
  uint8_t flag = p[0]; p++; len--;
  *is_null = false;
  switch (flag) {
    case NIL_FLAG: *is_null = true; return true;
    case BYTES_FLAG: { std::vector<uint8_t> out; size_t n = memcmp_decode(p, len, &out);
                       if (!n) FAIL("bad bytes"); v->assign(out.begin(), out.end()); return true; }
    case COMPACT_BYTES_FLAG: { const uint8_t *d; size_t dl, c;
                               if (!compact_bytes_decode(p, len, &d, &dl, &c)) FAIL("bad cbytes");
                               v->assign((const char*)d, dl); return true; }
    default: FAIL("unsupported datum flag for Bytes vector");
  }
}

/* ensure_decoded (lazy_column.rs:165): raw column -> typed vector */
static bool ensure_decoded(Column *col, const CoprFieldType &ft) {
  if (!col->raw) return true;
  ET et;
  if (!et_of_tp(ft.tp, &et)) FAIL("unsupported field type");
  TypedVec tv; tv.et = et;
  for (auto &rv : col->raw_vals) {
    const uint8_t *p = (const uint8_t *)rv.data();
    size_t len = rv.size();
    bool nul; int64_t iv; double dv; Decimal dd; std::string bv;
    switch (et) {
      case ET::Int:
        if (!decode_int_datum(p, len, &nul, &iv)) return false;
        if (nul) tv.push_null(); else tv.push_int(iv);
        break;
      case ET::Real:
        if (!decode_real_datum(p, len, ft.tp, &nul, &dv)) return false;
        if (nul) tv.push_null(); else tv.push_real(dv);
        break;
      case ET::Decimal:
        if (!decode_decimal_datum(p, len, &nul, &dd)) return false;
        if (nul) tv.push_null(); else tv.push_dec(dd);
        break;
      case ET::Bytes:
        if (!decode_bytes_datum(p, len, &nul, &bv)) return false;
        if (nul) tv.push_null(); else tv.push_bytes(std::move(bv));
        break;
    }
  }
  col->raw = false;
  col->dec = std::move(tv);
  return true;
}

/* ---------------- batch ---------------- */
struct Batch {
  std::vector<Column> cols;              /* physical columns */
  std::vector<uint32_t> logical_rows;
};

/* ---------------- RPN eval ---------------- */
struct StackEntry {
  bool is_scalar = false;
  bool ft_unsigned = false;   /* producing node's ft UNSIGNED bit */
  /* scalar */
  bool s_null = true;
  int64_t s_i = 0; double s_r = 0; Decimal s_d; std::string s_b;
  ET s_et = ET::Int;
  /* vector: one entry per LOGICAL row */
  TypedVec vec;
};

enum class CmpKind { LT, LE, GT, GE, EQ, NE };
static int64_t cmp_result(CmpKind k, int ord) {
  switch (k) {
    case CmpKind::LT: return ord < 0;
    case CmpKind::LE: return ord <= 0;
    case CmpKind::GT: return ord > 0;
    case CmpKind::GE: return ord >= 0;
    case CmpKind::EQ: return ord == 0;
    case CmpKind::NE: return ord != 0;
  }
  return 0;
}
/* Basic/UintUint/UintInt/IntUint comparers: impl_compare.rs:66-160 */
static int cmp_int(int64_t l, int64_t r, bool lu, bool ru) {
  if (lu && ru) { uint64_t a = (uint64_t)l, b = (uint64_t)r; return a < b ? -1 : a > b ? 1 : 0; }
  if (!lu && !ru) return l < r ? -1 : l > r ? 1 : 0;
  if (lu && !ru) {
    if (r < 0 || (uint64_t)l > (uint64_t)INT64_MAX) return 1;
    return l < r ? -1 : l > r ? 1 : 0;
  }
  if (l < 0 || (uint64_t)r > (uint64_t)INT64_MAX) return -1;
  return l < r ? -1 : l > r ? 1 : 0;
}

static bool eval_rpn(const CoprExpr &expr, Batch *batch,
                     const std::vector<CoprFieldType> &schema,
                     StackEntry *out) {
  std::vector<StackEntry> stack;
  size_t n_logical = batch->logical_rows.size();
  for (uint32_t ni = 0; ni < expr.n_nodes; ni++) {
    const CoprExprNode &node = expr.nodes[ni];
    switch (node.kind) {
      case COPR_EXPR_COLUMN_REF: {
        size_t off = (size_t)node.i64_val;
        if (off >= batch->cols.size()) FAIL("column offset out of range");
        if (!ensure_decoded(&batch->cols[off], schema[off])) return false;
        StackEntry e; e.is_scalar = false;
        e.ft_unsigned = is_unsigned(schema[off]);
        const TypedVec &src = batch->cols[off].dec;
        e.vec.et = src.et;
        for (uint32_t li = 0; li < n_logical; li++) {
          uint32_t pi = batch->logical_rows[li];
          if (src.nulls[pi]) { e.vec.push_null(); continue; }
          switch (src.et) {
            case ET::Int: e.vec.push_int(src.i[pi]); break;
            case ET::Real: e.vec.push_real(src.r[pi]); break;
            case ET::Decimal: e.vec.push_dec(src.d[pi]); break;
            case ET::Bytes: e.vec.push_bytes(src.b[pi]); break;
          }
        }
        stack.push_back(std::move(e));
        break;
      }
      case COPR_EXPR_CONST_NULL: case COPR_EXPR_CONST_INT: case COPR_EXPR_CONST_UINT:
      case COPR_EXPR_CONST_REAL: case COPR_EXPR_CONST_BYTES: case COPR_EXPR_CONST_DECIMAL: {
        StackEntry e; e.is_scalar = true;
        e.ft_unsigned = is_unsigned(node.ft) || node.kind == COPR_EXPR_CONST_UINT;
        ET et = ET::Int;
        et_of_tp(node.ft.tp, &et);
        e.s_et = et;
        switch (node.kind) {
          case COPR_EXPR_CONST_NULL: e.s_null = true; break;
          case COPR_EXPR_CONST_INT: case COPR_EXPR_CONST_UINT:
            e.s_null = false; e.s_i = node.i64_val; break;
          case COPR_EXPR_CONST_REAL: e.s_null = false; e.s_r = node.f64_val; break;
          case COPR_EXPR_CONST_BYTES:
            e.s_null = false; e.s_b.assign((const char *)node.bytes_val, node.bytes_len); break;
          case COPR_EXPR_CONST_DECIMAL: {
            size_t c;
            if (!node.bytes_val || !dec_decode(node.bytes_val, node.bytes_len, &e.s_d, &c))
              FAIL("bad const decimal");
            e.s_null = false; break;
          }
        }
        stack.push_back(std::move(e));
        break;
      }
      case COPR_EXPR_SCALAR_FUNC: {
        if ((int)stack.size() < node.n_args) FAIL("rpn underflow");
        size_t base = stack.size() - node.n_args;
        StackEntry out_e; out_e.is_scalar = false; out_e.vec.et = ET::Int;
        out_e.ft_unsigned = is_unsigned(node.ft);
        auto arg_null = [&](int a, uint32_t li) -> bool {
          const StackEntry &s = stack[base + a];
          return s.is_scalar ? s.s_null : s.vec.nulls[li] != 0;
        };
        auto arg_i = [&](int a, uint32_t li) -> int64_t {
          const StackEntry &s = stack[base + a];
          return s.is_scalar ? s.s_i : s.vec.i[li];
        };
        auto arg_r = [&](int a, uint32_t li) -> double {
          const StackEntry &s = stack[base + a];
          return s.is_scalar ? s.s_r : s.vec.r[li];
        };
        switch (node.sig) {
          case COPR_SIG_LT_INT: case COPR_SIG_LE_INT: case COPR_SIG_GT_INT:
          case COPR_SIG_GE_INT: case COPR_SIG_EQ_INT: case COPR_SIG_NE_INT: {
            CmpKind k = node.sig == COPR_SIG_LT_INT ? CmpKind::LT :
                        node.sig == COPR_SIG_LE_INT ? CmpKind::LE :
                        node.sig == COPR_SIG_GT_INT ? CmpKind::GT :
                        node.sig == COPR_SIG_GE_INT ? CmpKind::GE :
                        node.sig == COPR_SIG_EQ_INT ? CmpKind::EQ : CmpKind::NE;
            bool lu = stack[base].ft_unsigned, ru = stack[base + 1].ft_unsigned;
            for (uint32_t li = 0; li < n_logical; li++) {
              if (arg_null(0, li) || arg_null(1, li)) { out_e.vec.push_null(); continue; }
              out_e.vec.push_int(cmp_result(k, cmp_int(arg_i(0, li), arg_i(1, li), lu, ru)));
            }
            break;
          }
          case COPR_SIG_LT_REAL: case COPR_SIG_LE_REAL: case COPR_SIG_GT_REAL:
          case COPR_SIG_GE_REAL: case COPR_SIG_EQ_REAL: case COPR_SIG_NE_REAL: {
            CmpKind k = node.sig == COPR_SIG_LT_REAL ? CmpKind::LT :
                        node.sig == COPR_SIG_LE_REAL ? CmpKind::LE :
                        node.sig == COPR_SIG_GT_REAL ? CmpKind::GT :
                        node.sig == COPR_SIG_GE_REAL ? CmpKind::GE :
                        node.sig == COPR_SIG_EQ_REAL ? CmpKind::EQ : CmpKind::NE;
            for (uint32_t li = 0; li < n_logical; li++) {
              if (arg_null(0, li) || arg_null(1, li)) { out_e.vec.push_null(); continue; }
              double a = arg_r(0, li), b = arg_r(1, li);
              out_e.vec.push_int(cmp_result(k, a < b ? -1 : a > b ? 1 : 0));
            }
            break;
          }
          case COPR_SIG_LOGICAL_AND:   /* impl_op.rs logical_and (NULL-aware) */
            for (uint32_t li = 0; li < n_logical; li++) {
              bool n0 = arg_null(0, li), n1 = arg_null(1, li);
              bool f0 = !n0 && arg_i(0, li) == 0, f1 = !n1 && arg_i(1, li) == 0;
              if (f0 || f1) out_e.vec.push_int(0);
              else if (n0 || n1) out_e.vec.push_null();
              else out_e.vec.push_int(1);
            }
            break;
          case COPR_SIG_LOGICAL_OR:
            for (uint32_t li = 0; li < n_logical; li++) {
              bool n0 = arg_null(0, li), n1 = arg_null(1, li);
              bool t0 = !n0 && arg_i(0, li) != 0, t1 = !n1 && arg_i(1, li) != 0;
              if (t0 || t1) out_e.vec.push_int(1);
              else if (n0 || n1) out_e.vec.push_null();
              else out_e.vec.push_int(0);
            }
            break;
          case COPR_SIG_UNARY_NOT:
            for (uint32_t li = 0; li < n_logical; li++) {
              if (arg_null(0, li)) out_e.vec.push_null();
              else out_e.vec.push_int(arg_i(0, li) == 0 ? 1 : 0);
            }
            break;
          case COPR_SIG_PLUS_INT: case COPR_SIG_MINUS_INT: case COPR_SIG_MULTIPLY_INT:
            for (uint32_t li = 0; li < n_logical; li++) {
              if (arg_null(0, li) || arg_null(1, li)) { out_e.vec.push_null(); continue; }
              int64_t a = arg_i(0, li), b = arg_i(1, li), res;
              bool ovf;
              if (node.sig == COPR_SIG_PLUS_INT) ovf = __builtin_add_overflow(a, b, &res);
              else if (node.sig == COPR_SIG_MINUS_INT) ovf = __builtin_sub_overflow(a, b, &res);
              else ovf = __builtin_mul_overflow(a, b, &res);
              if (ovf) FAIL("BIGINT value is out of range");
              out_e.vec.push_int(res);
            }
            break;
          case COPR_SIG_INT_IS_NULL:
            for (uint32_t li = 0; li < n_logical; li++)
              out_e.vec.push_int(arg_null(0, li) ? 1 : 0);
            break;
          case COPR_SIG_INT_IS_TRUE:
            for (uint32_t li = 0; li < n_logical; li++)
              out_e.vec.push_int(!arg_null(0, li) && arg_i(0, li) != 0 ? 1 : 0);
            break;
          case COPR_SIG_INT_IS_FALSE:
            for (uint32_t li = 0; li < n_logical; li++)
              out_e.vec.push_int(!arg_null(0, li) && arg_i(0, li) == 0 ? 1 : 0);
            break;
          default:
            FAIL("unsupported ScalarFuncSig");
        }
        stack.resize(base);
        stack.push_back(std::move(out_e));
        break;
      }
      default:
        FAIL("unknown expr node kind");
    }
  }
  if (stack.size() != 1) FAIL("rpn did not reduce to one value");
  *out = std::move(stack[0]);
  return true;
}

/* ---------------- table scan ---------------- */
struct ScanState {
  std::vector<CoprColumnInfo> cols;
  std::vector<int> handle_indices;              /* pk_handle columns */
  std::unordered_map<int64_t, int> col_id_index;
  bool index_scan = false;                      /* BatchIndexScanExecutor */
  uint64_t next_kv = 0;                         /* scan cursor */
};

/* Index VALUE layout split (index_scan_executor.rs:322-371,700-885):
 *   old (len <= 9): [8B BE handle][flag] for unique / '0' or empty for
 *     non-unique (:336-345);
 *   new (len > 9): TailLen | [VersionFlag(125) Version] | Options | tail;
 *     options = [127 len u16le CHandle] (common handle — outside the
 *     int-handle subset), [126 pid 8B], [128.. restore-data row-v2 to the
 *     segment end]; handle = first 8 BE bytes of the last tail_len bytes
 *     when tail_len >= 8 (decode_int_handle_from_value :416-422 applied to
 *     build_operations' tail :794-835). V4 restore data (version 0) makes
 *     the restore row the source of ALL index columns (:903-907); V5 is
 *     skipped for int columns (need_restored_data false, :639-698). */
static bool split_index_value(const uint8_t *v, size_t n, uint64_t *handle,
                              bool *has_handle, const uint8_t **restore,
                              size_t *rlen) {
  *has_handle = false;
  *restore = nullptr;
  *rlen = 0;
  if (n <= 9) {
    if (n >= 8) {
      uint64_t u = 0;
      for (int b = 0; b < 8; b++) u = (u << 8) | v[b];   /* plain BE u64 */
      *handle = u;
      *has_handle = true;
    }
    return true;
  }
  size_t tail_len = v[0];
  if (tail_len >= n) return false;
  int version = 0;
  size_t opt = 1;
  if ((tail_len == 0 || tail_len == 1) && v[1] == 125) {
    version = v[2];
    opt = 3;
  }
  if (n < opt + tail_len) return false;
  size_t opt_end = n - tail_len;
  while (opt < opt_end) {
    uint8_t f = v[opt];
    if (f == 127) return false;          /* common handle: not int-handle */
    if (f == 126) {
      if (opt + 9 > opt_end) return false;
      opt += 9;
      continue;
    }
    if (f == 128) {
      if (version == 0) {
        *restore = v + opt;
        *rlen = opt_end - opt;
      }
      opt = opt_end;
      break;
    }
    return false;
  }
  if (tail_len >= 8) {
    uint64_t u = 0;
    for (int b = 0; b < 8; b++) u = (u << 8) | v[n - tail_len + b];
    *handle = u;
    *has_handle = true;
  }
  return true;
}

/* IndexScan process_kv_pair (index_scan_executor.rs:373-560): key =
 * 't'||tid||'_i'||index_id|| comparable datums || [int-handle datum];
 * columns are POSITIONAL raw comparable datum slices
 * (extract_columns_from_datum_format :504-517) unless a V4 restore row
 * overrides them; the int handle comes from the key tail for non-unique
 * indexes (decode_int_handle_from_key :460-481) or the value for unique
 * ones (old 8-byte form :417-422, new TailLen form :833-835). */
static bool scan_process_kv_index(ScanState &st, const uint8_t *key, size_t klen,
                                  const uint8_t *val, size_t vlen, Batch *batch) {
  if (klen < 19 || key[0] != 't' || key[9] != '_' || key[10] != 'i')
    FAIL("not an index key");
  uint64_t vhandle = 0;
  bool v_has_handle = false;
  const uint8_t *restore = nullptr;
  size_t rlen = 0;
  if (!split_index_value(val, vlen, &vhandle, &v_has_handle, &restore, &rlen))
    FAIL("bad index value layout");
  const uint8_t *p = key + 19;
  size_t rem = klen - 19;
  size_t ncols = st.cols.size();
  size_t n_idx_cols = 0;
  for (auto &c : st.cols) if (!c.pk_handle) n_idx_cols++;

  if (restore) {
    /* V4 restore data: every index column from the restore row-v2, keyed
       by the reference column id (:903-907, :483-501); key datums are
       skipped positionally to reach a key-form handle */
    RowSliceV2 rs;
    if (!row_v2_parse(restore, rlen, &rs)) FAIL("bad restore-data row");
    for (size_t i = 0, ci = 0; i < ncols && ci < n_idx_cols; i++) {
      if (st.cols[i].pk_handle) continue;
      uint32_t s, e;
      if (row_v2_find(rs, st.cols[i].column_id, &s, &e)) {
        std::vector<uint8_t> d;
        if (!row_v2_cell_to_v1_datum(rs.values + s, e - s, st.cols[i].ft.tp,
                                     st.cols[i].ft.flag, &d))
          FAIL("bad restore-data cell");
        batch->cols[i].raw_vals.emplace_back((const char *)d.data(), d.size());
      } else if (row_v2_is_null(rs, st.cols[i].column_id)) {
        batch->cols[i].raw_vals.emplace_back(1, '\0');   /* NULL datum */
      } else {
        FAIL("restore-data row missing column");
      }
      ci++;
    }
    for (size_t ci = 0; ci < n_idx_cols && rem > 0; ci++) {
      size_t dlen;
      if (!split_datum(p, rem, &dlen)) FAIL("bad index key datum");
      p += dlen; rem -= dlen;
    }
  } else {
    size_t ci = 0;
    for (size_t i = 0; i < ncols && ci < n_idx_cols; i++) {
      if (st.cols[i].pk_handle) continue;
      size_t dlen;
      if (!split_datum(p, rem, &dlen)) FAIL("bad index key datum");
      batch->cols[i].raw_vals.emplace_back((const char *)p, dlen);
      p += dlen; rem -= dlen;
      ci++;
    }
  }
  for (size_t i = 0; i < ncols; i++) {
    if (!st.cols[i].pk_handle) continue;
    int64_t handle;
    if (rem > 0) {
      if (rem < 9 || (p[0] != 3 && p[0] != 4)) FAIL("bad index handle datum");
      handle = p[0] == 3 ? decode_comparable_i64(p + 1)
                         : (int64_t)decode_comparable_u64(p + 1);
    } else if (v_has_handle) {
      handle = (int64_t)vhandle;
    } else {
      FAIL("index entry carries no int handle");
    }
    batch->cols[i].dec.push_int(handle);
  }
  return true;
}

/* process_kv_pair (+_v1/_v2): table_scan_executor.rs:375-485,209-291 */
static bool scan_process_kv(ScanState &st, const uint8_t *key, size_t klen,
                            const uint8_t *val, size_t vlen, Batch *batch) {
  size_t ncols = st.cols.size();
  std::vector<uint8_t> filled(ncols, 0);
  size_t decoded = 0;

  if (vlen == 0 || (vlen == 1 && val[0] == NIL_FLAG)) {
    /* row with no columns (table_scan_executor.rs:387-388) */
  } else if (val[0] == 128) {
    /* row v2 (process_v2, table_scan_executor.rs:259-291) */
    RowSliceV2 rs;
    if (!row_v2_parse(val, vlen, &rs)) FAIL("bad row v2");
    for (auto &kv : st.col_id_index) {
      int idx = kv.second;
      if (filled[idx]) continue;
      uint32_t s, e;
      if (row_v2_find(rs, kv.first, &s, &e)) {
        std::vector<uint8_t> d;
        if (!row_v2_cell_to_v1_datum(rs.values + s, e - s, st.cols[idx].ft.tp,
                                     st.cols[idx].ft.flag, &d))
          FAIL("bad v2 cell");
        batch->cols[idx].raw_vals.emplace_back((const char *)d.data(), d.size());
        decoded++; filled[idx] = 1;
      } else if (row_v2_is_null(rs, kv.first)) {
        batch->cols[idx].raw_vals.emplace_back(1, (char)NIL_FLAG);
        decoded++; filled[idx] = 1;
      }
    }
  } else {
    /* row v1 (process_v1, table_scan_executor.rs:209-256) */
    const uint8_t *p = val; size_t rem = vlen;
    while (rem > 0 && decoded < ncols) {
      if (p[0] != VAR_INT_FLAG) FAIL("column id must be VAR_INT");
      p++; rem--;
      int64_t col_id; size_t n;
      if (!decode_var_i64(p, rem, &col_id, &n)) FAIL("bad col id");
      p += n; rem -= n;
      size_t dlen;
      if (!split_datum(p, rem, &dlen)) FAIL("bad datum in row");
      auto it = st.col_id_index.find(col_id);
      if (it != st.col_id_index.end()) {
        int idx = it->second;
        if (!filled[idx]) {
          batch->cols[idx].raw_vals.emplace_back((const char *)p, dlen);
          decoded++; filled[idx] = 1;
        }
      }
      p += dlen; rem -= dlen;
    }
  }

  if (!st.handle_indices.empty()) {
    int64_t handle;
    if (!decode_int_handle(key, klen, &handle)) FAIL("bad record key");
    for (int hi : st.handle_indices) {
      if (!filled[hi]) {
        batch->cols[hi].dec.push_int(handle);
        decoded++; filled[hi] = 1;
      }
    }
  } else {
    if (klen < 19 || key[0] != 't' || key[9] != '_' || key[10] != 'r')
      FAIL("not a record key");
  }

  /* default / NULL fill: table_scan_executor.rs:456-483 */
  for (size_t i = 0; i < ncols; i++) {
    if (filled[i]) continue;
    if (!st.handle_indices.empty() &&
        std::find(st.handle_indices.begin(), st.handle_indices.end(), (int)i) !=
            st.handle_indices.end())
      continue;  /* decoded handle col, already pushed */
    const CoprColumnInfo &ci = st.cols[i];
    if (ci.default_val && ci.default_val_len > 0) {
      batch->cols[i].raw_vals.emplace_back((const char *)ci.default_val, ci.default_val_len);
    } else if (!(ci.ft.flag & COPR_FLAG_NOT_NULL)) {
      batch->cols[i].raw_vals.emplace_back(1, (char)NIL_FLAG);
    } else {
      FAIL("missing data for NOT NULL column");
    }
  }
  return true;
}

/* ---------------- aggregates ---------------- */
struct AggState {             /* one state per (group, agg) */
  /* count */
  uint64_t count = 0;
  /* sum/avg (impl_sum.rs / impl_avg.rs): sum + presence */
  bool has_value = false;
  Decimal dsum;               /* Summable Decimal */
  double rsum = 0;            /* Summable Real */
  /* max/min/first */
  bool mm_set = false;
  int64_t mm_i = 0; double mm_r = 0; Decimal mm_d; std::string mm_b;
  bool mm_null = false;       /* FIRST: first value may be NULL */
  /* bit ops (impl_bit_op.rs:43-45: u64 fold; NULL ignored) */
  uint64_t bits = 0;
};

struct AggDefRt {
  int32_t func;
  CoprExpr arg;
  CoprFieldType out_ft;
  ET in_et;        /* eval type after the sum/avg rewrite */
  bool arg_unsigned;
};

static void agg_state_init(AggState *s, const AggDefRt &def) {
  *s = AggState();
  if (def.func == COPR_AGG_SUM || def.func == COPR_AGG_AVG) {
    if (def.in_et == ET::Decimal) s->dsum = dec_zero();
  } else if (def.func == COPR_AGG_BIT_AND) {
    s->bits = ~0ull;          /* impl_bit_op.rs: AND starts at all-ones */
  }
}

/* update one row (impl_count.rs:65 / impl_sum.rs AggrFnStateSum::update /
 * impl_avg.rs:121-133 / impl_max_min.rs / impl_first.rs / impl_bit_op.rs) */
static bool agg_update(AggState *s, const AggDefRt &def, const StackEntry &v, uint32_t li) {
  bool nul = v.is_scalar ? v.s_null : v.vec.nulls[li] != 0;
  switch (def.func) {
    case COPR_AGG_COUNT:
      if (!nul) s->count++;
      return true;
    case COPR_AGG_SUM: case COPR_AGG_AVG:
      if (nul) return true;
      s->has_value = true;
      s->count++;
      if (def.in_et == ET::Decimal) {
        const Decimal &x = v.is_scalar ? v.s_d : v.vec.d[li];
        Decimal out;
        int res = dec_add(s->dsum, x, &out);
        if (res == 2) FAIL("DECIMAL value is out of range");
        s->dsum = out;
      } else if (def.in_et == ET::Real) {
        s->rsum += v.is_scalar ? v.s_r : v.vec.r[li];
      } else {
        FAIL("sum/avg over unsupported type");
      }
      return true;
    case COPR_AGG_MAX: case COPR_AGG_MIN: {
      if (nul) return true;
      bool mx = def.func == COPR_AGG_MAX;
      switch (def.in_et) {
        case ET::Int: {
          int64_t x = v.is_scalar ? v.s_i : v.vec.i[li];
          if (!s->mm_set) { s->mm_i = x; s->mm_set = true; }
          else {
            int c = cmp_int(x, s->mm_i, def.arg_unsigned, def.arg_unsigned);
            if (mx ? c > 0 : c < 0) s->mm_i = x;
          }
          break;
        }
        case ET::Real: {
          double x = v.is_scalar ? v.s_r : v.vec.r[li];
          if (!s->mm_set) { s->mm_r = x; s->mm_set = true; }
          else if (mx ? x > s->mm_r : x < s->mm_r) s->mm_r = x;
          break;
        }
        case ET::Decimal: {
          const Decimal &x = v.is_scalar ? v.s_d : v.vec.d[li];
          if (!s->mm_set) { s->mm_d = x; s->mm_set = true; }
          else { int c = dec_cmp(x, s->mm_d); if (mx ? c > 0 : c < 0) s->mm_d = x; }
          break;
        }
        case ET::Bytes: {
          const std::string &x = v.is_scalar ? v.s_b : v.vec.b[li];
          if (!s->mm_set) { s->mm_b = x; s->mm_set = true; }
          else if (mx ? x > s->mm_b : x < s->mm_b) s->mm_b = x;
          break;
        }
      }
      return true;
    }
    case COPR_AGG_FIRST:
      if (s->mm_set) return true;
      s->mm_set = true;
      s->mm_null = nul;
      if (!nul) {
        switch (def.in_et) {
          case ET::Int: s->mm_i = v.is_scalar ? v.s_i : v.vec.i[li]; break;
          case ET::Real: s->mm_r = v.is_scalar ? v.s_r : v.vec.r[li]; break;
          case ET::Decimal: s->mm_d = v.is_scalar ? v.s_d : v.vec.d[li]; break;
          case ET::Bytes: s->mm_b = v.is_scalar ? v.s_b : v.vec.b[li]; break;
        }
      }
      return true;
    case COPR_AGG_BIT_AND: case COPR_AGG_BIT_OR: case COPR_AGG_BIT_XOR: {
      if (nul) return true;
      uint64_t x = (uint64_t)(v.is_scalar ? v.s_i : v.vec.i[li]);
      if (def.func == COPR_AGG_BIT_AND) s->bits &= x;
      else if (def.func == COPR_AGG_BIT_OR) s->bits |= x;
      else s->bits ^= x;
      return true;
    }
  }
  FAIL("unknown agg func");
}

/* push_result into output typed vectors.
 * column layout per agg: COUNT/SUM/MAX/MIN/FIRST/BIT* -> 1 col; AVG -> 2
 * (count LongLong UNSIGNED, sum out_ft) — impl_avg.rs:52-60,146-156 */
static void agg_push_result(const AggState &s, const AggDefRt &def,
                            std::vector<TypedVec> *out, size_t *oi) {
  switch (def.func) {
    case COPR_AGG_COUNT:
      (*out)[(*oi)++].push_int((int64_t)s.count);
      return;
    case COPR_AGG_AVG:
      (*out)[(*oi)].push_int((int64_t)s.count); (*oi)++;
      /* fallthrough to sum column */
      if (!s.has_value) { (*out)[(*oi)++].push_null(); return; }
      if (def.in_et == ET::Decimal) (*out)[(*oi)++].push_dec(s.dsum);
      else (*out)[(*oi)++].push_real(s.rsum);
      return;
    case COPR_AGG_SUM:
      if (!s.has_value) { (*out)[(*oi)++].push_null(); return; }
      if (def.in_et == ET::Decimal) (*out)[(*oi)++].push_dec(s.dsum);
      else (*out)[(*oi)++].push_real(s.rsum);
      return;
    case COPR_AGG_MAX: case COPR_AGG_MIN: case COPR_AGG_FIRST: {
      bool present = s.mm_set && !s.mm_null;
      if (!present) { (*out)[(*oi)++].push_null(); return; }
      switch (def.in_et) {
        case ET::Int: (*out)[(*oi)++].push_int(s.mm_i); break;
        case ET::Real: (*out)[(*oi)++].push_real(s.mm_r); break;
        case ET::Decimal: (*out)[(*oi)++].push_dec(s.mm_d); break;
        case ET::Bytes: (*out)[(*oi)++].push_bytes(s.mm_b); break;
      }
      return;
    }
    case COPR_AGG_BIT_AND: case COPR_AGG_BIT_OR: case COPR_AGG_BIT_XOR:
      (*out)[(*oi)++].push_int((int64_t)s.bits);
      return;
  }
}

/* ---------------- response encode ----------------
 * lazy_column.rs:242-256 (raw verbatim / decoded via vector.rs:362) */
static bool encode_output_cell(const Column &col, uint32_t phys_idx,
                               const CoprFieldType &ft, std::vector<uint8_t> *out) {
  if (col.raw) {
    const std::string &rv = col.raw_vals[phys_idx];
    out->insert(out->end(), rv.begin(), rv.end());
    return true;
  }
  const TypedVec &tv = col.dec;
  uint8_t tmp[48];
  if (tv.nulls[phys_idx]) { out->push_back(NIL_FLAG); return true; }
  switch (tv.et) {
    case ET::Int:
      if (is_unsigned(ft)) {                     /* datum_codec.rs:254-258 */
        out->push_back(UINT_FLAG);
        encode_comparable_u64(tmp, (uint64_t)tv.i[phys_idx]);
      } else {                                   /* datum_codec.rs:261-265 */
        out->push_back(INT_FLAG);
        encode_comparable_i64(tmp, tv.i[phys_idx]);
      }
      out->insert(out->end(), tmp, tmp + 8);
      return true;
    case ET::Real:                               /* datum_codec.rs:275-279 */
      out->push_back(FLOAT_FLAG);
      encode_comparable_f64(tmp, tv.r[phys_idx]);
      out->insert(out->end(), tmp, tmp + 8);
      return true;
    case ET::Decimal: {                          /* datum_codec.rs:281-287 */
      out->push_back(DECIMAL_FLAG);
      uint8_t prec, frac;
      dec_prec_and_frac(tv.d[phys_idx], &prec, &frac);
      size_t n = dec_encode(tv.d[phys_idx], prec, frac, tmp);
      out->insert(out->end(), tmp, tmp + n);
      return true;
    }
    case ET::Bytes: {                            /* datum_codec.rs:290-294 */
      out->push_back(COMPACT_BYTES_FLAG);
      uint8_t hdr[10];
      size_t n = encode_var_i64(hdr, (int64_t)tv.b[phys_idx].size());
      out->insert(out->end(), hdr, hdr + n);
      out->insert(out->end(), tv.b[phys_idx].begin(), tv.b[phys_idx].end());
      return true;
    }
  }
  return false;
}

/* ---------------- group key ---------------- */
struct GroupKey {
  bool nul = true;
  ET et = ET::Int;
  int64_t i = 0; double r = 0; std::string b; Decimal d;
  bool operator==(const GroupKey &o) const {
    if (nul != o.nul) return false;
    if (nul) return true;
    switch (et) {
      case ET::Int: return i == o.i;
      case ET::Real: return r == o.r;
      case ET::Bytes: return b == o.b;
      case ET::Decimal: return dec_cmp(d, o.d) == 0;
    }
    return false;
  }
};
struct GroupKeyHash {
  size_t operator()(const GroupKey &k) const {
    if (k.nul) return 0x9e3779b97f4a7c15ull;
    switch (k.et) {
      case ET::Int: return std::hash<int64_t>()(k.i);
      case ET::Real: return std::hash<double>()(k.r);
      case ET::Bytes: return std::hash<std::string>()(k.b);
      case ET::Decimal: return std::hash<std::string>()(dec_to_string(k.d));
    }
    return 0;
  }
};


/* ---------------- TypeChunk response encoding ----------------
 * chunk/column.rs:41-71 (container choice by field type), :1052-1071
 * (wire layout), decimal.rs:2135-2142 (decimal = 40 B struct dump),
 * runner.rs:1188-1225 (one Chunk per executor batch). The pipeline always
 * produces TypeDefault datum rows internally; this post-pass re-encodes
 * them column-wise per chunk. */
struct ChunkCol {
  bool fixed = true;
  uint32_t flen = 8;
  uint32_t length = 0, null_cnt = 0;
  std::vector<uint8_t> bitmap, data;
  std::vector<int64_t> offsets{0};
  bool init(const CoprFieldType &ft) {
    switch (ft.tp) {
      case COPR_TP_TINY: case COPR_TP_SHORT: case COPR_TP_INT24:
      case COPR_TP_LONG: case COPR_TP_LONGLONG: case COPR_TP_YEAR:
      case COPR_TP_DOUBLE: case COPR_TP_DURATION:
        fixed = true; flen = 8; return true;
      case COPR_TP_NEWDECIMAL:
        fixed = true; flen = 40; return true;
      case COPR_TP_VARCHAR: case COPR_TP_STRING: case COPR_TP_VARSTRING:
      case COPR_TP_BLOB:
        fixed = false; return true;
      default:
        return false;
    }
  }
  void bit(bool notnull) {
    if ((length & 7) == 0) bitmap.push_back(0);
    if (notnull) bitmap[length >> 3] |= (uint8_t)(1u << (length & 7));
    else null_cnt++;
  }
  void app_null() {
    bit(false);
    if (fixed) data.insert(data.end(), flen, 0);
    else offsets.push_back(offsets.back());
    length++;
  }
  void app_fixed(const uint8_t *p, uint32_t n) {
    bit(true);
    data.insert(data.end(), p, p + n);
    if (n < flen) data.insert(data.end(), flen - n, 0);
    length++;
  }
  void app_var(const uint8_t *p, size_t n) {
    bit(true);
    data.insert(data.end(), p, p + n);
    offsets.push_back(offsets.back() + (int64_t)n);
    length++;
  }
  void flush(std::vector<uint8_t> *out) {
    uint8_t w[8];
    auto u32le = [&](uint32_t v) {
      w[0] = v; w[1] = v >> 8; w[2] = v >> 16; w[3] = v >> 24;
      out->insert(out->end(), w, w + 4);
    };
    u32le(length);
    u32le(null_cnt);
    if (null_cnt > 0)
      out->insert(out->end(), bitmap.begin(), bitmap.end());
    if (!fixed) {
      for (int64_t v : offsets) {
        uint64_t u = (uint64_t)v;
        for (int i = 0; i < 8; i++) w[i] = (uint8_t)(u >> (8 * i));
        out->insert(out->end(), w, w + 8);
      }
    }
    out->insert(out->end(), data.begin(), data.end());
  }
};

/* parse ONE datum at p and append it to the chunk column; returns bytes
   consumed, 0 on error (datum.rs flags; from_raw_datums semantics) */
static size_t chunk_append_datum(ChunkCol *c, const uint8_t *p, size_t rem) {
  if (!rem) return 0;
  uint8_t flag = p[0];
  uint8_t tmp[8];
  switch (flag) {
    case 0:
      c->app_null();
      return 1;
    case 3: case 4: {                    /* comparable int/uint */
      if (rem < 9) return 0;
      uint64_t u = 0;
      for (int i = 0; i < 8; i++) u = (u << 8) | p[1 + i];
      if (flag == 3) u ^= 0x8000000000000000ull;
      for (int i = 0; i < 8; i++) tmp[i] = (uint8_t)(u >> (8 * i));
      c->app_fixed(tmp, 8);
      return 9;
    }
    case 5: {                            /* comparable f64 */
      if (rem < 9) return 0;
      uint64_t u = 0;
      for (int i = 0; i < 8; i++) u = (u << 8) | p[1 + i];
      if (u & 0x8000000000000000ull) u &= 0x7FFFFFFFFFFFFFFFull;
      else u = ~u;
      for (int i = 0; i < 8; i++) tmp[i] = (uint8_t)(u >> (8 * i));
      c->app_fixed(tmp, 8);
      return 9;
    }
    case 8: case 9: {                    /* var int / var uint */
      uint64_t uv = 0;
      size_t n = 0;
      if (!decode_var_u64(p + 1, rem - 1, &uv, &n)) return 0;
      uint64_t u = uv;
      if (flag == 8) {
        uint64_t half = uv >> 1;
        u = (uv & 1) ? ~half : half;
      }
      for (int i = 0; i < 8; i++) tmp[i] = (uint8_t)(u >> (8 * i));
      c->app_fixed(tmp, 8);
      return 1 + n;
    }
    case 6: {                            /* decimal -> 40 B struct dump */
      Decimal d;
      size_t used = 0;
      if (!dec_decode(p + 1, rem - 1, &d, &used)) return 0;
      uint8_t buf[40];
      buf[0] = d.int_cnt; buf[1] = d.frac_cnt; buf[2] = d.result_frac_cnt;
      buf[3] = d.negative ? 1 : 0;
      for (int i = 0; i < 9; i++) {
        uint32_t w = d.word_buf[i];
        buf[4 + 4 * i] = (uint8_t)w;
        buf[5 + 4 * i] = (uint8_t)(w >> 8);
        buf[6 + 4 * i] = (uint8_t)(w >> 16);
        buf[7 + 4 * i] = (uint8_t)(w >> 24);
      }
      c->app_fixed(buf, 40);
      return 1 + used;
    }
    case 2: {                            /* compact bytes */
      int64_t n;
      size_t nb;
      if (!decode_var_i64(p + 1, rem - 1, &n, &nb)) return 0;
      if (n < 0 || 1 + nb + (uint64_t)n > rem) return 0;
      c->app_var(p + 1 + nb, (size_t)n);
      return 1 + (size_t)nb + (size_t)n;
    }
    case 1: {                            /* memcomparable bytes */
      std::vector<uint8_t> out;
      size_t used = memcmp_decode(p + 1, rem - 1, &out);
      if (!used) return 0;
      c->app_var(out.data(), out.size());
      return 1 + used;
    }
    default:
      return 0;
  }
}

/* re-encode a TypeDefault datum-row response into TypeChunk chunks */
static bool chunk_encode_post(const std::vector<uint8_t> &datum_resp,
                              const std::vector<uint64_t> &rows_per_chunk,
                              const std::vector<CoprFieldType> &out_fts,
                              std::vector<uint8_t> *out) {
  size_t p = 0;
  size_t nc = out_fts.size();
  for (uint64_t nrows : rows_per_chunk) {
    if (!nrows) continue;
    std::vector<ChunkCol> cols(nc);
    for (size_t c = 0; c < nc; c++)
      if (!cols[c].init(out_fts[c])) return false;
    for (uint64_t r = 0; r < nrows; r++) {
      for (size_t c = 0; c < nc; c++) {
        size_t used = chunk_append_datum(&cols[c], datum_resp.data() + p,
                                         datum_resp.size() - p);
        if (!used) return false;
        p += used;
      }
    }
    for (size_t c = 0; c < nc; c++) cols[c].flush(out);
  }
  return p == datum_resp.size();
}

/* ---------------- the pipeline ---------------- */
struct Pipeline {
  const CoprDagRequest *req;
  /* schemas per executor boundary; final = output schema */
  std::vector<CoprFieldType> scan_schema;
  std::vector<CoprFieldType> out_schema;
  ScanState scan;
  /* executor list split */
  const CoprExecutor *scan_exec = nullptr;
  std::vector<const CoprExecutor *> rest;
  /* agg runtime */
  const CoprExecutor *agg_exec = nullptr;   /* at most one agg node supported */
  bool stream_agg = false;    /* BatchStreamAggregationExecutor: groups are
                                 CONTIGUOUS RUNS of equal keys in input order
                                 (stream_aggr_executor.rs:108-117); no re-merge
                                 of non-adjacent equal keys */
  std::vector<AggDefRt> agg_defs;
  /* BatchTopNExecutor (top_n_executor.rs): keep the n smallest rows under
     the order-by comparator (NULL sorts first; desc reverses the whole
     order incl. NULL placement), emitted in sorted order after the source
     drains. Ties are emitted in source-row order (the reference's heap is
     unstable on ties; stable-by-arrival is a deterministic refinement). */
  const CoprExecutor *topn_exec = nullptr;
  uint64_t limit = UINT64_MAX;
};

static bool build_pipeline(const CoprDagRequest *req, Pipeline *pl) {
  pl->req = req;
  if (req->n_executors == 0) FAIL("empty executor list");
  const CoprExecutor &first = req->executors[0];
  if (first.kind != COPR_EXEC_TABLE_SCAN && first.kind != COPR_EXEC_INDEX_SCAN)
    FAIL("first executor must be a scan");
  pl->scan.index_scan = first.kind == COPR_EXEC_INDEX_SCAN;
  pl->scan_exec = &first;
  for (uint32_t i = 0; i < first.n_columns; i++) {
    const CoprColumnInfo &ci = first.columns[i];
    pl->scan.cols.push_back(ci);
    pl->scan_schema.push_back(ci.ft);
    if (ci.pk_handle) pl->scan.handle_indices.push_back((int)i);
    else pl->scan.col_id_index[ci.column_id] = (int)i;
  }
  for (uint32_t e = 1; e < req->n_executors; e++) {
    const CoprExecutor &ex = req->executors[e];
    switch (ex.kind) {
      case COPR_EXEC_SELECTION:
        pl->rest.push_back(&ex);
        break;
      case COPR_EXEC_SIMPLE_AGG: case COPR_EXEC_FAST_HASH_AGG:
      case COPR_EXEC_SLOW_HASH_AGG:
        if (pl->agg_exec) FAIL("only one aggregation node supported");
        pl->agg_exec = &ex;
        break;
      case COPR_EXEC_STREAM_AGG:
        if (pl->agg_exec) FAIL("only one aggregation node supported");
        pl->agg_exec = &ex;
        pl->stream_agg = true;
        break;
      case COPR_EXEC_TOPN:
        if (pl->topn_exec) FAIL("only one TopN node supported");
        if (ex.n_order_by == 0) FAIL("TopN needs order-by expressions");
        pl->topn_exec = &ex;
        break;
      case COPR_EXEC_LIMIT:
        pl->limit = ex.limit;
        break;
      default:
        FAIL("unsupported executor kind");
    }
  }
  /* out schema: no agg -> scan schema; agg -> agg outputs then group-bys
     (util/aggr_executor.rs:137-147 prepares schema: aggr outputs first,
      then FastHashAgg group_by ft appended — fast_hash_aggr_executor.rs:272) */
  if (pl->agg_exec && pl->topn_exec) FAIL("TopN below/above agg unsupported");
  if (!pl->agg_exec) {
    pl->out_schema = pl->scan_schema;
  } else {
    for (uint32_t a = 0; a < pl->agg_exec->n_aggs; a++) {
      const CoprAggDef &ad = pl->agg_exec->aggs[a];
      AggDefRt rt;
      rt.func = ad.func;
      rt.arg = ad.arg;
      rt.out_ft = ad.out_ft;
      /* arg eval type + sum/avg rewrite (util::rewrite_exp_for_sum_avg):
         Int -> Decimal */
      ET arg_et = ET::Int;
      const CoprExprNode &last = ad.arg.nodes[ad.arg.n_nodes - 1];
      CoprFieldType arg_ft = last.ft;
      if (last.kind == COPR_EXPR_COLUMN_REF)
        arg_ft = pl->scan_schema[(size_t)last.i64_val];
      if (!et_of_tp(arg_ft.tp, &arg_et)) FAIL("agg arg type unsupported");
      rt.arg_unsigned = is_unsigned(arg_ft);
      if ((ad.func == COPR_AGG_SUM || ad.func == COPR_AGG_AVG) && arg_et == ET::Int)
        arg_et = ET::Decimal;   /* exact cast at update time */
      rt.in_et = arg_et;
      if (ad.func == COPR_AGG_AVG) {
        CoprFieldType cnt_ft{};
        cnt_ft.tp = COPR_TP_LONGLONG; cnt_ft.flag = COPR_FLAG_UNSIGNED;
        cnt_ft.flen = -1; cnt_ft.decimal = -1; cnt_ft.collate = 63;
        pl->out_schema.push_back(cnt_ft);        /* impl_avg.rs:52-58 */
      }
      pl->out_schema.push_back(ad.out_ft);
      pl->agg_defs.push_back(rt);
    }
    for (uint32_t g = 0; g < pl->agg_exec->n_group_by; g++) {
      const CoprExpr &ge = pl->agg_exec->group_by[g];
      const CoprExprNode &last = ge.nodes[ge.n_nodes - 1];
      CoprFieldType gft = last.ft;
      if (last.kind == COPR_EXPR_COLUMN_REF)
        gft = pl->scan_schema[(size_t)last.i64_val];
      pl->out_schema.push_back(gft);
    }
  }
  return true;
}

static bool run_pipeline(const CoprDagRequest *req,
                         const uint8_t *keys, const uint64_t *key_offs,
                         const uint8_t *vals, const uint64_t *val_offs,
                         uint64_t n_kv, std::vector<uint8_t> *resp,
                         uint64_t *n_out_rows, uint64_t *resume_row) {
  *resume_row = UINT64_MAX;
  Pipeline pl;
  if (!build_pipeline(req, &pl)) return false;

  size_t n_agg_out_cols = 0;
  for (auto &d : pl.agg_defs) n_agg_out_cols += (d.func == COPR_AGG_AVG) ? 2 : 1;

  /* group table: key -> state index (fast_hash_aggr_executor.rs:226-246) */
  std::unordered_map<GroupKey, size_t, GroupKeyHash> groups;
  std::vector<GroupKey> group_keys_in_order;
  std::vector<AggState> states;   /* n_groups * n_aggs, row-major by group */
  bool simple_agg = pl.agg_exec && pl.agg_exec->kind == COPR_EXEC_SIMPLE_AGG;
  if (simple_agg) {
    states.resize(pl.agg_defs.size());
    for (size_t a = 0; a < pl.agg_defs.size(); a++)
      agg_state_init(&states[a], pl.agg_defs[a]);
  }

  uint64_t out_rows = 0;
  uint64_t emitted_rows = 0;
  /* chunk segmentation for TypeChunk (runner.rs:1188-1225: one chunk per
     executor batch; drains emit in 1024-row chunks) */
  std::vector<uint64_t> chunk_rows;

  /* TopN collection: per surviving row, the order keys + the row's encoded
     output cells (top_n_executor.rs keeps whole rows in its heap; we keep
     all candidates and partial-sort at drain -- same results) */
  struct TopRow {
    std::vector<int64_t> kv;
    std::vector<uint8_t> knul;
    std::vector<uint8_t> cells;
    uint64_t arrival;
  };
  std::vector<TopRow> top_rows;
  std::vector<uint8_t> top_desc;      /* per order-by expr */
  std::vector<uint8_t> top_uns;
  if (pl.topn_exec) {
    for (uint32_t ob = 0; ob < pl.topn_exec->n_order_by; ob++) {
      top_desc.push_back(pl.topn_exec->order_desc
                             ? (uint8_t)(pl.topn_exec->order_desc[ob] != 0) : 0);
      const CoprExprNode &last =
          pl.topn_exec->order_by[ob].nodes[pl.topn_exec->order_by[ob].n_nodes - 1];
      CoprFieldType oft = last.ft;
      if (last.kind == COPR_EXPR_COLUMN_REF)
        oft = pl.scan_schema[(size_t)last.i64_val];
      ET oet;
      if (!et_of_tp(oft.tp, &oet) || oet != ET::Int)
        FAIL("TopN order-by supports int columns");
      top_uns.push_back((uint8_t)is_unsigned(oft));
    }
  }

  /* stream agg: run boundary = key differs from the previous row's */
  bool sa_have_prev = false;
  GroupKey sa_prev;

  /* runner loop: batch 32 -> x2 -> 1024 (runner.rs:39,51,986) */
  uint64_t batch_size = 32;
  uint64_t cursor = 0;
  while (cursor < n_kv && (pl.topn_exec || emitted_rows < pl.limit)) {
    uint64_t n = std::min<uint64_t>(batch_size, n_kv - cursor);
    Batch batch;
    batch.cols.resize(pl.scan.cols.size());
    for (size_t i = 0; i < pl.scan.cols.size(); i++) {
      bool is_handle = std::find(pl.scan.handle_indices.begin(),
                                 pl.scan.handle_indices.end(),
                                 (int)i) != pl.scan.handle_indices.end();
      batch.cols[i].raw = !is_handle;
      if (is_handle) batch.cols[i].dec.et = ET::Int;
    }
    for (uint64_t k = 0; k < n; k++) {
      uint64_t idx = cursor + k;
      bool okk = pl.scan.index_scan
          ? scan_process_kv_index(pl.scan, keys + key_offs[idx],
                                  (size_t)(key_offs[idx + 1] - key_offs[idx]),
                                  vals + val_offs[idx],
                                  (size_t)(val_offs[idx + 1] - val_offs[idx]), &batch)
          : scan_process_kv(pl.scan, keys + key_offs[idx],
                            (size_t)(key_offs[idx + 1] - key_offs[idx]),
                            vals + val_offs[idx],
                            (size_t)(val_offs[idx + 1] - val_offs[idx]), &batch);
      if (!okk) return false;
    }
    cursor += n;
    batch.logical_rows.resize(n);
    for (uint64_t k = 0; k < n; k++) batch.logical_rows[k] = (uint32_t)k;

    /* selections */
    for (auto *ex : pl.rest) {
      for (uint32_t c = 0; c < ex->n_conditions && !batch.logical_rows.empty(); c++) {
        StackEntry v;
        if (!eval_rpn(ex->conditions[c], &batch, pl.scan_schema, &v)) return false;
        std::vector<uint32_t> kept;
        if (v.is_scalar) {
          bool b = !v.s_null && v.s_i != 0;      /* as_mysql_bool for Int */
          if (b) kept = batch.logical_rows;
        } else {
          for (uint32_t li = 0; li < batch.logical_rows.size(); li++) {
            bool b = !v.vec.nulls[li] && v.vec.i[li] != 0;
            if (b) kept.push_back(batch.logical_rows[li]);
          }
        }
        batch.logical_rows = std::move(kept);
      }
    }

    if (pl.topn_exec) {
      if (!batch.logical_rows.empty()) {
        size_t n_logical = batch.logical_rows.size();
        std::vector<StackEntry> okeys(pl.topn_exec->n_order_by);
        for (uint32_t ob = 0; ob < pl.topn_exec->n_order_by; ob++)
          if (!eval_rpn(pl.topn_exec->order_by[ob], &batch, pl.scan_schema,
                        &okeys[ob]))
            return false;
        for (uint32_t li = 0; li < n_logical; li++) {
          TopRow tr;
          tr.arrival = top_rows.size();
          for (uint32_t ob = 0; ob < pl.topn_exec->n_order_by; ob++) {
            const StackEntry &v = okeys[ob];
            if (v.is_scalar) {
              tr.knul.push_back((uint8_t)v.s_null);
              tr.kv.push_back(v.s_i);
            } else {
              tr.knul.push_back((uint8_t)(v.vec.nulls[li] != 0));
              tr.kv.push_back(v.vec.nulls[li] ? 0 : v.vec.i[li]);
            }
          }
          uint32_t pi = batch.logical_rows[li];
          std::vector<uint8_t> cells;
          for (uint32_t oo = 0; oo < req->n_output_offsets; oo++) {
            uint32_t off = req->output_offsets[oo];
            if (off >= batch.cols.size()) FAIL("output offset out of range");
            if (!encode_output_cell(batch.cols[off], pi, pl.out_schema[off],
                                    &cells))
              return false;
          }
          tr.cells = std::move(cells);
          top_rows.push_back(std::move(tr));
        }
      }
    } else if (!pl.agg_exec) {
      /* stream rows straight to the response (with LIMIT) */
      uint64_t take = std::min<uint64_t>(batch.logical_rows.size(),
                                         pl.limit - emitted_rows);
      for (uint64_t li = 0; li < take; li++) {
        uint32_t pi = batch.logical_rows[li];
        for (uint32_t oo = 0; oo < req->n_output_offsets; oo++) {
          uint32_t off = req->output_offsets[oo];
          if (off >= batch.cols.size()) FAIL("output offset out of range");
          if (!encode_output_cell(batch.cols[off], pi, pl.out_schema[off], resp))
            return false;
        }
        out_rows++;
      }
      emitted_rows += take;
      if (take) chunk_rows.push_back(take);
      /* paging: stop at the batch boundary where accumulated OUTPUT rows
         reach paging_size (runner.rs:917-921 record_all >= paging_size) */
      if (req->paging_size && emitted_rows >= req->paging_size && cursor < n_kv) {
        *resume_row = cursor;
        break;
      }
    } else if (!batch.logical_rows.empty()) {
      /* evaluate agg args + group keys over the batch */
      size_t n_logical = batch.logical_rows.size();
      std::vector<StackEntry> arg_vals(pl.agg_defs.size());
      for (size_t a = 0; a < pl.agg_defs.size(); a++) {
        if (!eval_rpn(pl.agg_defs[a].arg, &batch, pl.scan_schema, &arg_vals[a]))
          return false;
        /* sum/avg Int->Decimal rewrite: cast values now (exact) */
        if ((pl.agg_defs[a].func == COPR_AGG_SUM || pl.agg_defs[a].func == COPR_AGG_AVG) &&
            pl.agg_defs[a].in_et == ET::Decimal) {
          StackEntry &v = arg_vals[a];
          if (v.is_scalar) {
            if (!v.s_null && v.s_et == ET::Int)
              v.s_d = v.ft_unsigned ? dec_from_u64((uint64_t)v.s_i) : dec_from_i64(v.s_i);
            v.s_et = ET::Decimal;
          } else if (v.vec.et == ET::Int) {
            TypedVec cast; cast.et = ET::Decimal;
            for (size_t li = 0; li < v.vec.size(); li++) {
              if (v.vec.nulls[li]) cast.push_null();
              else cast.push_dec(v.ft_unsigned ? dec_from_u64((uint64_t)v.vec.i[li])
                                               : dec_from_i64(v.vec.i[li]));
            }
            v.vec = std::move(cast);
          }
        }
      }
      if (simple_agg) {
        for (uint32_t li = 0; li < n_logical; li++)
          for (size_t a = 0; a < pl.agg_defs.size(); a++)
            if (!agg_update(&states[a], pl.agg_defs[a], arg_vals[a], li)) return false;
      } else {
        if (pl.agg_exec->n_group_by != 1) FAIL("exactly one group-by supported");
        StackEntry gv;
        if (!eval_rpn(pl.agg_exec->group_by[0], &batch, pl.scan_schema, &gv)) return false;
        for (uint32_t li = 0; li < n_logical; li++) {
          GroupKey key;
          if (gv.is_scalar) {
            key.nul = gv.s_null; key.et = gv.s_et;
            key.i = gv.s_i; key.r = gv.s_r; key.b = gv.s_b; key.d = gv.s_d;
          } else {
            key.et = gv.vec.et;
            key.nul = gv.vec.nulls[li] != 0;
            if (!key.nul) {
              switch (key.et) {
                case ET::Int: key.i = gv.vec.i[li]; break;
                case ET::Real: key.r = gv.vec.r[li]; break;
                case ET::Bytes: key.b = gv.vec.b[li]; break;
                case ET::Decimal: key.d = gv.vec.d[li]; break;
              }
            }
          }
          size_t gi;
          if (pl.stream_agg) {
            /* contiguous-run grouping: new state whenever the key changes */
            if (!sa_have_prev || !(key == sa_prev)) {
              gi = group_keys_in_order.size();
              group_keys_in_order.push_back(key);
              size_t base = states.size();
              states.resize(base + pl.agg_defs.size());
              for (size_t a = 0; a < pl.agg_defs.size(); a++)
                agg_state_init(&states[base + a], pl.agg_defs[a]);
              sa_have_prev = true;
              sa_prev = key;
            } else {
              gi = group_keys_in_order.size() - 1;
            }
          } else {
          auto it = groups.find(key);
          if (it == groups.end()) {
            gi = group_keys_in_order.size();
            groups.emplace(key, gi);
            group_keys_in_order.push_back(key);
            size_t base = states.size();
            states.resize(base + pl.agg_defs.size());
            for (size_t a = 0; a < pl.agg_defs.size(); a++)
              agg_state_init(&states[base + a], pl.agg_defs[a]);
          } else {
            gi = it->second;
          }
          }
          for (size_t a = 0; a < pl.agg_defs.size(); a++)
            if (!agg_update(&states[gi * pl.agg_defs.size() + a], pl.agg_defs[a],
                            arg_vals[a], li))
              return false;
        }
      }
    }

    if (batch_size < 1024) batch_size *= 2;     /* runner.rs:1229-1241 */
    if (batch_size > 1024) batch_size = 1024;
  }

  if (pl.topn_exec) {
    std::vector<size_t> order(top_rows.size());
    for (size_t i = 0; i < order.size(); i++) order[i] = i;
    std::sort(order.begin(), order.end(), [&](size_t x, size_t y) {
      const TopRow &a = top_rows[x], &b = top_rows[y];
      for (size_t k = 0; k < a.kv.size(); k++) {
        int c;
        if (a.knul[k] != b.knul[k]) c = a.knul[k] ? -1 : 1;   /* NULL first */
        else if (a.knul[k]) c = 0;
        else if (top_uns[k]) {
          uint64_t ua = (uint64_t)a.kv[k], ub = (uint64_t)b.kv[k];
          c = ua < ub ? -1 : (ua > ub ? 1 : 0);
        } else {
          c = a.kv[k] < b.kv[k] ? -1 : (a.kv[k] > b.kv[k] ? 1 : 0);
        }
        if (top_desc[k]) c = -c;
        if (c) return c < 0;
      }
      return a.arrival < b.arrival;    /* stable tie-break */
    });
    uint64_t take = std::min<uint64_t>(
        std::min<uint64_t>(pl.topn_exec->limit, pl.limit), order.size());
    for (uint64_t i = 0; i < take; i++) {
      const TopRow &tr = top_rows[order[i]];
      resp->insert(resp->end(), tr.cells.begin(), tr.cells.end());
      out_rows++;
    }
    for (uint64_t left = take; left; ) {
      uint64_t b = std::min<uint64_t>(left, 1024);
      chunk_rows.push_back(b);
      left -= b;
    }
  } else if (pl.agg_exec) {
    /* drain: iterate groups, push states (fast_hash_aggr_executor.rs:393) */
    size_t n_groups = simple_agg ? 1 : group_keys_in_order.size();
    size_t n_out_cols = n_agg_out_cols + (simple_agg ? 0 : 1);
    std::vector<TypedVec> out_cols(n_out_cols);
    {
      size_t oc = 0;
      for (auto &d : pl.agg_defs) {
        if (d.func == COPR_AGG_AVG) { out_cols[oc++].et = ET::Int; out_cols[oc++].et = d.in_et; }
        else if (d.func == COPR_AGG_COUNT || d.func == COPR_AGG_BIT_AND ||
                 d.func == COPR_AGG_BIT_OR || d.func == COPR_AGG_BIT_XOR)
          out_cols[oc++].et = ET::Int;
        else out_cols[oc++].et = d.in_et;
      }
      if (!simple_agg) {
        ET get = ET::Int;
        const CoprFieldType &gft = pl.out_schema[pl.out_schema.size() - 1];
        if (!et_of_tp(gft.tp, &get)) FAIL("bad group ft");
        out_cols[oc].et = get;
      }
    }
    for (size_t g = 0; g < n_groups; g++) {
      size_t oi = 0;
      for (size_t a = 0; a < pl.agg_defs.size(); a++)
        agg_push_result(states[g * pl.agg_defs.size() + a], pl.agg_defs[a], &out_cols, &oi);
      if (!simple_agg) {
        const GroupKey &k = group_keys_in_order[g];
        TypedVec &gcol = out_cols[n_out_cols - 1];
        if (k.nul) gcol.push_null();
        else switch (k.et) {
          case ET::Int: gcol.push_int(k.i); break;
          case ET::Real: gcol.push_real(k.r); break;
          case ET::Bytes: gcol.push_bytes(k.b); break;
          case ET::Decimal: gcol.push_dec(k.d); break;
        }
      }
    }
    /* encode drained rows via output_offsets */
    uint64_t take = std::min<uint64_t>(n_groups, pl.limit);
    for (uint64_t left = take; left; ) {
      uint64_t b = std::min<uint64_t>(left, 1024);
      chunk_rows.push_back(b);
      left -= b;
    }
    for (uint64_t g = 0; g < take; g++) {
      for (uint32_t oo = 0; oo < req->n_output_offsets; oo++) {
        uint32_t off = req->output_offsets[oo];
        if (off >= n_out_cols) FAIL("output offset out of range");
        Column tmp; tmp.raw = false; tmp.dec = out_cols[off];  /* cheap-ish */
        if (!encode_output_cell(tmp, (uint32_t)g, pl.out_schema[off], resp))
          return false;
      }
      out_rows++;
    }
  }

  if (req->encode_type == 1) {
    std::vector<CoprFieldType> fts;
    for (uint32_t oo = 0; oo < req->n_output_offsets; oo++)
      fts.push_back(pl.out_schema[req->output_offsets[oo]]);
    std::vector<uint8_t> chunked;
    if (!chunk_encode_post(*resp, chunk_rows, fts, &chunked))
      FAIL("TypeChunk encode failed");
    resp->swap(chunked);
  }
  *n_out_rows = out_rows;
  return true;
}

}  // namespace

/* ================= C API ================= */
extern "C" {

/* ---- SST data-block decode (oracle restatement) ----
 * RocksDB BlockBasedTable data block (block_builder.cc/block.cc public
 * format; the crate TiKV pins is rust-rocksdb in Cargo.lock): entries of
 * [varint32 shared][varint32 non_shared][varint32 value_len][key delta]
 * [value], restart-point array u32le + count at the tail; keys are
 * InternalKeys whose trailing 8 bytes ((seq<<8)|type) are stripped. */
static bool blk_varint32(const uint8_t *p, size_t rem, uint32_t *v,
                         size_t *n) {
  uint32_t x = 0;
  size_t i = 0;
  int sh = 0;
  while (i < rem && i < 5) {
    uint8_t b = p[i++];
    x |= (uint32_t)(b & 0x7F) << sh;
    sh += 7;
    if (!(b & 0x80)) { *v = x; *n = i; return true; }
  }
  return false;
}

extern "C" int orc_block_parse(const uint8_t *blocks,
                               const uint64_t *block_offs, uint32_t n_blocks,
                               OrcRegion *out) {
  std::vector<uint8_t> keys, vals;
  std::vector<uint64_t> koffs{0}, voffs{0};
  for (uint32_t b = 0; b < n_blocks; b++) {
    const uint8_t *blk = blocks + block_offs[b];
    size_t blen = (size_t)(block_offs[b + 1] - block_offs[b]);
    if (blen < 8) return -1;
    uint32_t nr = (uint32_t)blk[blen - 4] | ((uint32_t)blk[blen - 3] << 8) |
                  ((uint32_t)blk[blen - 2] << 16) |
                  ((uint32_t)blk[blen - 1] << 24);
    if (blen < 4 + (size_t)nr * 4) return -1;
    size_t data_end = blen - 4 - (size_t)nr * 4;
    std::string key;
    size_t pos = 0;
    while (pos < data_end) {
      uint32_t shared, non_shared, vlen;
      size_t n;
      if (!blk_varint32(blk + pos, data_end - pos, &shared, &n)) return -1;
      pos += n;
      if (!blk_varint32(blk + pos, data_end - pos, &non_shared, &n)) return -1;
      pos += n;
      if (!blk_varint32(blk + pos, data_end - pos, &vlen, &n)) return -1;
      pos += n;
      if (pos + non_shared + vlen > data_end || shared > key.size()) return -1;
      key.resize(shared);
      key.append((const char *)(blk + pos), non_shared);
      pos += non_shared;
      if (key.size() < 8) return -1;
      keys.insert(keys.end(), key.begin(), key.end() - 8);
      koffs.push_back(keys.size());
      vals.insert(vals.end(), blk + pos, blk + pos + vlen);
      voffs.push_back(vals.size());
      pos += vlen;
    }
    if (pos != data_end) return -1;
  }
  uint64_t n_kv = koffs.size() - 1;
  out->keys = (uint8_t *)malloc(keys.size() ? keys.size() : 1);
  memcpy(out->keys, keys.data(), keys.size());
  out->vals = (uint8_t *)malloc(vals.size() ? vals.size() : 1);
  memcpy(out->vals, vals.data(), vals.size());
  out->key_offs = (uint64_t *)malloc(koffs.size() * 8);
  memcpy(out->key_offs, koffs.data(), koffs.size() * 8);
  out->val_offs = (uint64_t *)malloc(voffs.size() * 8);
  memcpy(out->val_offs, voffs.data(), voffs.size() * 8);
  out->n_kv = n_kv;
  return 0;
}

int orc_dag_run(const CoprDagRequest *req,
                const uint8_t *keys, const uint64_t *key_offs,
                const uint8_t *vals, const uint64_t *val_offs,
                uint64_t n_kv, OrcResult *out) {
  std::vector<uint8_t> resp;
  uint64_t n_rows = 0, resume = UINT64_MAX;
  if (!run_pipeline(req, keys, key_offs, vals, val_offs, n_kv, &resp, &n_rows,
                    &resume))
    return 1;
  out->data = (uint8_t *)malloc(resp.size() ? resp.size() : 1);
  memcpy(out->data, resp.data(), resp.size());
  out->data_len = resp.size();
  out->n_rows = n_rows;
  out->resume_row = resume;
  return 0;
}

void orc_result_free(OrcResult *r) {
  if (r && r->data) { free(r->data); r->data = nullptr; }
}

/* checksum.rs:59-114: per-KV CRC-64/XZ digest of key||value, XOR-folded */
int orc_checksum(const uint8_t *keys, const uint64_t *key_offs,
                 const uint8_t *vals, const uint64_t *val_offs,
                 uint64_t n_kv,
                 uint64_t *checksum, uint64_t *total_kvs, uint64_t *total_bytes) {
  uint64_t cs = 0, bytes = 0;
  for (uint64_t i = 0; i < n_kv; i++) {
    uint64_t st = crc64_xz_init();
    st = crc64_xz_update(st, keys + key_offs[i], (size_t)(key_offs[i + 1] - key_offs[i]));
    st = crc64_xz_update(st, vals + val_offs[i], (size_t)(val_offs[i + 1] - val_offs[i]));
    cs ^= crc64_xz_finish(st);
    bytes += (key_offs[i + 1] - key_offs[i]) + (val_offs[i + 1] - val_offs[i]);
  }
  *checksum = cs;
  *total_kvs = n_kv;
  *total_bytes = bytes;
  return 0;
}

uint64_t orc_crc64_xz(const uint8_t *p, uint64_t len) { return crc64_xz(p, (size_t)len); }

uint64_t orc_test_memcmp_encode(const uint8_t *src, uint64_t len, int desc, uint8_t *out) {
  return desc ? memcmp_encode_all_desc(src, (size_t)len, out)
              : memcmp_encode_all(src, (size_t)len, out);
}
uint64_t orc_test_memcmp_decode(const uint8_t *src, uint64_t len, uint8_t *out,
                                uint64_t *out_len) {
  std::vector<uint8_t> tmp;
  size_t n = memcmp_decode(src, (size_t)len, &tmp);
  if (n == 0) return 0;
  memcpy(out, tmp.data(), tmp.size());
  *out_len = tmp.size();
  return n;
}
uint64_t orc_test_var_i64_encode(int64_t v, uint8_t out[10]) {
  return encode_var_i64(out, v);
}
int orc_test_var_i64_decode(const uint8_t *p, uint64_t len, int64_t *v, uint64_t *consumed) {
  size_t n;
  if (!decode_var_i64(p, (size_t)len, v, &n)) return 1;
  *consumed = n;
  return 0;
}
void orc_test_row_key(int64_t table_id, int64_t handle, uint8_t out[19]) {
  encode_row_key(table_id, handle, out);
}
int orc_test_int_handle(const uint8_t *key, uint64_t len, int64_t *handle) {
  return decode_int_handle(key, (size_t)len, handle) ? 0 : 1;
}
int orc_test_row_v2_col(const uint8_t *val, uint64_t len, int64_t col_id,
                        int32_t tp, uint32_t ft_flag, uint8_t *out, int *is_null) {
  RowSliceV2 rs;
  *is_null = 0;
  if (!row_v2_parse(val, (size_t)len, &rs)) return -1;
  uint32_t s, e;
  if (row_v2_find(rs, col_id, &s, &e)) {
    std::vector<uint8_t> d;
    if (!row_v2_cell_to_v1_datum(rs.values + s, e - s, tp, ft_flag, &d)) return -1;
    memcpy(out, d.data(), d.size());
    return (int)d.size();
  }
  if (row_v2_is_null(rs, col_id)) { *is_null = 1; return 0; }
  return 0;
}
int orc_test_dec_add_encode(const uint8_t *a, uint64_t alen,
                            const uint8_t *b, uint64_t blen, uint8_t *out) {
  Decimal da, db, sum;
  size_t c;
  if (!dec_decode(a, (size_t)alen, &da, &c)) return -1;
  if (!dec_decode(b, (size_t)blen, &db, &c)) return -1;
  if (dec_add(da, db, &sum) == 2) return -1;
  uint8_t prec, frac;
  dec_prec_and_frac(sum, &prec, &frac);
  return (int)dec_encode(sum, prec, frac, out);
}
int orc_test_dec_from_i64_encode(int64_t v, uint8_t *out) {
  Decimal d = dec_from_i64(v);
  uint8_t prec, frac;
  dec_prec_and_frac(d, &prec, &frac);
  return (int)dec_encode(d, prec, frac, out);
}

/* ---- MVCC write-CF filter ----
 * Key: memcomparable(user_key) || BE(~commit_ts) (types.rs:152-161).
 * Value: [type][varint start_ts][tags 'v' len sv | 'R' | 'F' u64 |
 * 'l' u64+varint | 'S' varint; unknown tag stops] (write.rs:296-361).
 * Visibility (forward.rs:440-515): newest version with commit_ts <= read_ts;
 * Put -> row (short value; default-CF lookup unsupported here), Delete ->
 * skip key, Lock/Rollback (no last_change tag -> LastChange::Unknown,
 * types.rs:721-731) -> next older version. gc_fence unsupported. */
static bool parse_write_rec(const uint8_t *v, size_t len, char *type,
                            const uint8_t **sv, size_t *sv_len,
                            uint64_t *gc_fence, int *lc_not_exist,
                            uint64_t *start_ts = nullptr) {
  *sv = nullptr; *sv_len = 0; *gc_fence = 0; *lc_not_exist = 0;
  if (len < 1) return false;
  char t = (char)v[0];
  if (t != 'P' && t != 'D' && t != 'L' && t != 'R') return false;
  *type = t;
  size_t p = 1;
  uint64_t sts; size_t n;
  if (!decode_var_u64(v + p, len - p, &sts, &n)) return false;
  if (start_ts) *start_ts = sts;
  p += n;
  while (p < len) {
    uint8_t tag = v[p++];
    switch (tag) {
      case 'v': {
        if (p >= len) return false;
        uint8_t l = v[p++];
        if (p + l > len) return false;
        *sv = v + p; *sv_len = l;
        p += l;
        break;
      }
      case 'R': break;                         /* overlapped rollback flag */
      case 'F': {                              /* gc fence: u64 BE */
        if (p + 8 > len) return false;
        uint64_t f = 0;
        for (int b = 0; b < 8; b++) f = (f << 8) | v[p + b];
        *gc_fence = f;
        p += 8;
        break;
      }
      case 'l': {                              /* last_change ts + versions */
        if (p + 8 > len) return false;
        uint64_t lts = 0;
        for (int b = 0; b < 8; b++) lts = (lts << 8) | v[p + b];
        p += 8;
        uint64_t vers; size_t nn;
        if (!decode_var_u64(v + p, len - p, &vers, &nn)) return false;
        p += nn;
        /* LastChange::from_parts (types.rs:721-731): ts==0 && vers>0 =>
           NotExist => key invisible; otherwise iterating older versions is
           semantically equivalent to the reference's seek optimization */
        if (lts == 0 && vers > 0) *lc_not_exist = 1;
        break;
      }
      case 'S': {
        uint64_t vv; size_t nn;
        if (!decode_var_u64(v + p, len - p, &vv, &nn)) return false;
        p += nn;
        break;
      }
      default:
        return true;                           /* unknown tag stops parse */
    }
  }
  return true;
}

static int mvcc_filter_impl(const uint8_t *keys, const uint64_t *key_offs,
                            const uint8_t *vals, const uint64_t *val_offs,
                            uint64_t n_kv,
                            const uint8_t *dkeys, const uint64_t *dkey_offs,
                            const uint8_t *dvals, const uint64_t *dval_offs,
                            uint64_t n_default,
                            uint64_t read_ts, OrcRegion *out) {
  std::vector<uint8_t> okeys, ovals;
  std::vector<uint64_t> okoffs{0}, ovoffs{0};
  /* default-CF lookup at memcomparable(user_key)||BE(~start_ts)
     (forward.rs:433-515 load_data_from_default_cf) */
  auto default_find = [&](const uint8_t *u, size_t ulen,
                          uint64_t start_ts) -> int64_t {
    uint64_t tsd = ~start_ts;
    uint8_t suffix[8];
    for (int b = 0; b < 8; b++) suffix[b] = (uint8_t)(tsd >> (8 * (7 - b)));
    uint64_t lo = 0, hi = n_default;
    while (lo < hi) {
      uint64_t mid = (lo + hi) >> 1;
      const uint8_t *d = dkeys + dkey_offs[mid];
      size_t dlen = (size_t)(dkey_offs[mid + 1] - dkey_offs[mid]);
      size_t tot = ulen + 8;
      int c = 0;
      size_t n = tot < dlen ? tot : dlen;
      for (size_t i2 = 0; i2 < n && c == 0; i2++) {
        uint8_t ub = i2 < ulen ? u[i2] : suffix[i2 - ulen];
        if (ub != d[i2]) c = ub < d[i2] ? -1 : 1;
      }
      if (c == 0) c = tot == dlen ? 0 : (tot < dlen ? -1 : 1);
      if (c == 0) return (int64_t)mid;
      if (c < 0) hi = mid;
      else lo = mid + 1;
    }
    return -1;
  };
  auto uenc = [&](uint64_t i, size_t *len) -> const uint8_t * {
    size_t kl = (size_t)(key_offs[i + 1] - key_offs[i]);
    if (kl < 9) return nullptr;
    *len = kl - 8;
    return keys + key_offs[i];
  };
  auto commit_ts = [&](uint64_t i) -> uint64_t {
    const uint8_t *p = keys + key_offs[i + 1] - 8;
    uint64_t d = 0;
    for (int b = 0; b < 8; b++) d = (d << 8) | p[b];
    return ~d;
  };
  uint64_t i = 0;
  while (i < n_kv) {
    size_t ulen;
    const uint8_t *u = uenc(i, &ulen);
    if (!u) return 1;
    /* walk this key group */
    uint64_t j = i;
    while (j < n_kv) {
      size_t ul2;
      const uint8_t *u2 = uenc(j, &ul2);
      if (!u2 || ul2 != ulen || memcmp(u, u2, ulen) != 0) break;
      uint64_t ts = commit_ts(j);
      if (ts > read_ts) { j++; continue; }
      char type; const uint8_t *sv; size_t svl; uint64_t fence; int lc_ne;
      uint64_t start_ts = 0;
      if (!parse_write_rec(vals + val_offs[j],
                           (size_t)(val_offs[j + 1] - val_offs[j]),
                           &type, &sv, &svl, &fence, &lc_ne, &start_ts))
        return 1;
      /* gc fence pointing within read_ts: key invisible
         (write.rs:425-442 + forward.rs:444-446 break None) */
      if (fence != 0 && fence <= read_ts) break;
      if (type == 'P') {
        if (!sv) {
          if (!dkeys) return 2;                /* no default stream: loud */
          int64_t m = default_find(u, ulen, start_ts);
          if (m < 0) return 1;                 /* corruption: DEFAULT_NOT_FOUND */
          sv = dvals + dval_offs[m];
          svl = (size_t)(dval_offs[m + 1] - dval_offs[m]);
        }
        std::vector<uint8_t> raw;
        if (!memcmp_decode(u, ulen, &raw)) return 1;
        okeys.insert(okeys.end(), raw.begin(), raw.end());
        okoffs.push_back(okeys.size());
        ovals.insert(ovals.end(), sv, sv + svl);
        ovoffs.push_back(ovals.size());
        break;
      }
      if (type == 'D') break;
      if (lc_ne) break;                        /* LastChange::NotExist */
      j++;                                     /* Lock/Rollback: older */
    }
    /* advance to the next user key */
    while (i < n_kv) {
      size_t ul2;
      const uint8_t *u2 = uenc(i, &ul2);
      if (!u2) return 1;
      if (ul2 != ulen || memcmp(u, u2, ulen) != 0) break;
      i++;
    }
  }
  uint64_t n = okoffs.size() - 1;
  out->n_kv = n;
  out->keys = (uint8_t *)malloc(okeys.size() ? okeys.size() : 1);
  memcpy(out->keys, okeys.data(), okeys.size());
  out->vals = (uint8_t *)malloc(ovals.size() ? ovals.size() : 1);
  memcpy(out->vals, ovals.data(), ovals.size());
  out->key_offs = (uint64_t *)malloc((n + 1) * 8);
  memcpy(out->key_offs, okoffs.data(), (n + 1) * 8);
  out->val_offs = (uint64_t *)malloc((n + 1) * 8);
  memcpy(out->val_offs, ovoffs.data(), (n + 1) * 8);
  return 0;
}

extern "C" int orc_mvcc_filter(const uint8_t *keys, const uint64_t *key_offs,
                               const uint8_t *vals, const uint64_t *val_offs,
                               uint64_t n_kv, uint64_t read_ts, OrcRegion *out) {
  return mvcc_filter_impl(keys, key_offs, vals, val_offs, n_kv,
                          nullptr, nullptr, nullptr, nullptr, 0, read_ts, out);
}

extern "C" int orc_mvcc_filter2(const uint8_t *keys, const uint64_t *key_offs,
                                const uint8_t *vals, const uint64_t *val_offs,
                                uint64_t n_kv,
                                const uint8_t *dkeys, const uint64_t *dkey_offs,
                                const uint8_t *dvals, const uint64_t *dval_offs,
                                uint64_t n_default,
                                uint64_t read_ts, OrcRegion *out) {
  return mvcc_filter_impl(keys, key_offs, vals, val_offs, n_kv,
                          dkeys, dkey_offs, dvals, dval_offs, n_default,
                          read_ts, out);
}

extern "C" void orc_region_free(OrcRegion *r) {
  if (!r) return;
  free(r->keys); free(r->key_offs); free(r->vals); free(r->val_offs);
  memset(r, 0, sizeof(*r));
}

/* ---- whole-SST walk (oracle restatement of the BlockBasedTable file
 * layer: RocksDB format.cc / block_based_table_reader.cc public format;
 * TiKV consumes SSTs via rust-rocksdb, engine_iterator.rs:12).
 * Footer (format_version 1..5): last 53 bytes =
 *   [checksum_type u8][metaindex handle][index handle][pad to 40]
 *   [format_version u32le][magic u64le], magic 0x88e241b785f4cff7.
 * Per-block trailer: [compression u8][checksum u32le]; checksum_type 1 =
 * masked crc32c(contents || compression byte) (util/crc32c.h mask).
 * Index values are plain BlockHandles (varint64 offset + size). */

static uint32_t o_crc32c(const uint8_t *p, size_t n) {
  uint32_t c = 0xFFFFFFFFu;
  for (size_t i = 0; i < n; i++) {
    c ^= p[i];
    for (int k = 0; k < 8; k++) c = (c >> 1) ^ ((c & 1) ? 0x82F63B78u : 0);
  }
  return c ^ 0xFFFFFFFFu;
}

static bool o_var64(const uint8_t *p, size_t rem, uint64_t *v, size_t *n) {
  uint64_t x = 0;
  size_t i = 0;
  int sh = 0;
  while (i < rem && i < 10) {
    uint8_t b = p[i++];
    x |= (uint64_t)(b & 0x7F) << sh;
    sh += 7;
    if (!(b & 0x80)) { *v = x; *n = i; return true; }
  }
  return false;
}

static bool o_decompress(uint8_t type, const uint8_t *p, size_t len,
                         std::vector<uint8_t> *out) {
  /* compress_format_version 2: varint32 raw size + payload */
  uint32_t raw;
  size_t n;
  if (!blk_varint32(p, len, &raw, &n)) return false;
  size_t base = out->size();
  out->resize(base + raw);
  static void *l4 = dlopen("liblz4.so.1", RTLD_NOW);
  static void *lz = dlopen("libzstd.so.1", RTLD_NOW);
  if (type == 4 || type == 5) {
    typedef int (*fn)(const char *, char *, int, int);
    static fn d = l4 ? (fn)dlsym(l4, "LZ4_decompress_safe") : nullptr;
    if (!d) return false;
    int r = d((const char *)(p + n), (char *)(out->data() + base),
              (int)(len - n), (int)raw);
    return r >= 0 && (uint32_t)r == raw;
  }
  if (type == 7) {
    typedef size_t (*fn)(void *, size_t, const void *, size_t);
    typedef unsigned (*efn)(size_t);
    static fn d = lz ? (fn)dlsym(lz, "ZSTD_decompress") : nullptr;
    static efn ie = lz ? (efn)dlsym(lz, "ZSTD_isError") : nullptr;
    if (!d || !ie) return false;
    size_t r = d(out->data() + base, raw, p + n, len - n);
    return !ie(r) && r == raw;
  }
  return false;
}

/* -1 malformed/checksum, -2 unsupported shape, 0 ok */
extern "C" int orc_sst_parse(const uint8_t *f, uint64_t len, OrcRegion *out) {
  if (len < 53) return -1;
  uint64_t magic = 0;
  for (int i = 7; i >= 0; i--) magic = (magic << 8) | f[len - 8 + i];
  if (magic != 0x88e241b785f4cff7ull) return -1;
  uint32_t ver = (uint32_t)f[len - 12] | ((uint32_t)f[len - 11] << 8) |
                 ((uint32_t)f[len - 10] << 16) | ((uint32_t)f[len - 9] << 24);
  if (ver < 1 || ver > 5) return -2;
  const uint8_t *fp = f + len - 53;
  uint8_t cks = fp[0];
  uint64_t h[4];
  size_t pos = 1, n;
  for (int i = 0; i < 4; i++) {
    if (!o_var64(fp + pos, 41 - pos, &h[i], &n)) return -1;
    pos += n;
  }
  uint64_t i_off = h[2], i_sz = h[3];
  auto trailer_ok = [&](uint64_t off, uint64_t sz, uint8_t *type) {
    if (off + sz + 5 > len) return false;
    *type = f[off + sz];
    if (cks == 1) {
      uint32_t stored = (uint32_t)f[off + sz + 1] |
                        ((uint32_t)f[off + sz + 2] << 8) |
                        ((uint32_t)f[off + sz + 3] << 16) |
                        ((uint32_t)f[off + sz + 4] << 24);
      uint32_t c = o_crc32c(f + off, sz + 1);
      if (((c >> 15) | (c << 17)) + 0xa282ead8u != stored) return false;
    }
    return true;
  };
  uint8_t itype;
  if (!trailer_ok(i_off, i_sz, &itype)) return -1;
  std::vector<uint8_t> ibuf;
  const uint8_t *ib = f + i_off;
  size_t iblen = (size_t)i_sz;
  if (itype != 0) {
    if (!o_decompress(itype, f + i_off, (size_t)i_sz, &ibuf)) return -1;
    ib = ibuf.data();
    iblen = ibuf.size();
  }
  if (iblen < 8) return -1;
  uint32_t nr = (uint32_t)ib[iblen - 4] | ((uint32_t)ib[iblen - 3] << 8) |
                ((uint32_t)ib[iblen - 2] << 16) |
                ((uint32_t)ib[iblen - 1] << 24);
  if (iblen < 4 + (size_t)nr * 4) return -1;
  size_t dend = iblen - 4 - (size_t)nr * 4;
  std::vector<uint8_t> cat;
  std::vector<uint64_t> offs{0};
  std::string key;
  pos = 0;
  while (pos < dend) {
    uint32_t sh, ns, vl;
    if (!blk_varint32(ib + pos, dend - pos, &sh, &n)) return -1;
    pos += n;
    if (!blk_varint32(ib + pos, dend - pos, &ns, &n)) return -1;
    pos += n;
    if (!blk_varint32(ib + pos, dend - pos, &vl, &n)) return -1;
    pos += n;
    if (pos + ns + vl > dend || sh > key.size()) return -1;
    key.resize(sh);
    key.append((const char *)(ib + pos), ns);
    pos += ns;
    uint64_t b_off, b_sz;
    size_t n2;
    if (!o_var64(ib + pos, vl, &b_off, &n) ||
        !o_var64(ib + pos + n, vl - n, &b_sz, &n2))
      return -2;
    pos += vl;
    uint8_t bt;
    if (!trailer_ok(b_off, b_sz, &bt)) return -1;
    if (bt == 0) {
      cat.insert(cat.end(), f + b_off, f + b_off + b_sz);
    } else {
      if (!o_decompress(bt, f + b_off, (size_t)b_sz, &cat)) return -1;
    }
    offs.push_back(cat.size());
  }
  if (pos != dend || offs.size() < 2) return -1;
  return orc_block_parse(cat.data(), offs.data(), (uint32_t)(offs.size() - 1),
                         out);
}

}  // extern "C"
