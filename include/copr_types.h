/* copr_types.h — plain-C descriptor types for the coprocessor DAG boundary.
 *
 * These structs are the decoded form of the tipb::DagRequest executor
 * descriptors that TiKV's BatchExecutorsRunner::from_request consumes
 * (reference: components/tidb_query_executors/src/runner.rs:307 build_executors;
 * executor kinds runner.rs:199-234). The Rust shim inside TiKV translates the
 * already-protobuf-decoded tipb types into these structs;
 * the engine never parses protobuf.
 *
 * tipb itself (git tipb @ 5f9928e per the reference Cargo.lock) is not vendored
 * in the reference tree; symbolic names below mirror tipb's enum variant names,
 * and field-type codes are the public MySQL protocol type codes that
 * tidb_query_datatype::FieldTypeTp restates.
 */
#ifndef COPR_TYPES_H
#define COPR_TYPES_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- field types (MySQL protocol codes; tidb_query_datatype/src/def/field_type.rs) ---- */
enum CoprFieldTp {
  COPR_TP_DECIMAL     = 0,
  COPR_TP_TINY        = 1,
  COPR_TP_SHORT       = 2,
  COPR_TP_LONG        = 3,
  COPR_TP_FLOAT       = 4,
  COPR_TP_DOUBLE      = 5,
  COPR_TP_NULL        = 6,
  COPR_TP_TIMESTAMP   = 7,
  COPR_TP_LONGLONG    = 8,
  COPR_TP_INT24       = 9,
  COPR_TP_DATE        = 10,
  COPR_TP_DURATION    = 11,
  COPR_TP_DATETIME    = 12,
  COPR_TP_YEAR        = 13,
  COPR_TP_VARCHAR     = 15,
  COPR_TP_JSON        = 0xf5,      /* 245 */
  COPR_TP_NEWDECIMAL  = 0xf6,      /* 246 */
  COPR_TP_BLOB        = 0xfc,      /* 252 */
  COPR_TP_VARSTRING   = 0xfd,      /* 253 */
  COPR_TP_STRING      = 0xfe       /* 254 */
};

/* MySQL column flag bits (subset; tidb_query_datatype::FieldTypeFlag) */
enum CoprFieldFlag {
  COPR_FLAG_NOT_NULL  = 1u << 0,
  COPR_FLAG_PRI_KEY   = 1u << 1,
  COPR_FLAG_UNSIGNED  = 1u << 5
};

typedef struct CoprFieldType {
  int32_t  tp;        /* CoprFieldTp */
  uint32_t flag;      /* CoprFieldFlag bits */
  int32_t  flen;      /* display length; -1 = unspecified */
  int32_t  decimal;   /* frac digits; -1 = unspecified    */
  int32_t  collate;   /* collation id; 63 = binary        */
} CoprFieldType;

/* ---- column info (mirrors tipb::ColumnInfo as used by
 *      table_scan_executor.rs:89-208 / index_scan_executor.rs) ---- */
typedef struct CoprColumnInfo {
  int64_t       column_id;
  CoprFieldType ft;
  int32_t       pk_handle;          /* column is the int primary-key handle */
  const uint8_t *default_val;       /* datum-encoded default, may be NULL   */
  uint32_t      default_val_len;
} CoprColumnInfo;

/* ---- RPN expressions ----
 * The engine consumes expressions in RPN (postorder) form — the same form
 * RpnExpressionBuilder::build_from_expr_tree produces from the tipb tree
 * (tidb_query_expr/src/types/expr_builder.rs; dispatch lib.rs:435). A tipb
 * Expr tree maps to this by a postorder walk (INTEGRATION.md shows the shim).
 */
enum CoprExprNodeKind {
  COPR_EXPR_COLUMN_REF = 0,     /* operand: column offset in source schema */
  COPR_EXPR_CONST_NULL = 1,
  COPR_EXPR_CONST_INT  = 2,     /* i64 payload  */
  COPR_EXPR_CONST_UINT = 3,     /* u64 payload  */
  COPR_EXPR_CONST_REAL = 4,     /* f64 payload  */
  COPR_EXPR_CONST_BYTES = 5,    /* bytes payload */
  COPR_EXPR_CONST_DECIMAL = 6,  /* datum-payload-encoded decimal bytes */
  COPR_EXPR_SCALAR_FUNC = 7     /* sig identifies the function */
};

/* Scalar function signatures (mirrors tipb::ScalarFuncSig variant names;
 * the subset map_expr_node_to_rpn_func supports natively here —
 * tidb_query_expr/src/lib.rs:435+. Int/Uint variants are selected by the
 * operands' UNSIGNED flags exactly as map_int_sig does (lib.rs:234). */
enum CoprScalarSig {
  COPR_SIG_LT_INT = 1, COPR_SIG_LE_INT, COPR_SIG_GT_INT, COPR_SIG_GE_INT,
  COPR_SIG_EQ_INT, COPR_SIG_NE_INT,
  COPR_SIG_LT_REAL, COPR_SIG_LE_REAL, COPR_SIG_GT_REAL, COPR_SIG_GE_REAL,
  COPR_SIG_EQ_REAL, COPR_SIG_NE_REAL,
  COPR_SIG_LOGICAL_AND, COPR_SIG_LOGICAL_OR, COPR_SIG_UNARY_NOT,
  COPR_SIG_PLUS_INT, COPR_SIG_MINUS_INT, COPR_SIG_MULTIPLY_INT,
  COPR_SIG_INT_IS_NULL, COPR_SIG_INT_IS_TRUE, COPR_SIG_INT_IS_FALSE
};

typedef struct CoprExprNode {
  int32_t       kind;        /* CoprExprNodeKind */
  int32_t       sig;         /* CoprScalarSig when kind==SCALAR_FUNC */
  int32_t       n_args;      /* arity when kind==SCALAR_FUNC */
  CoprFieldType ft;          /* node's field type (tipb Expr.field_type) */
  int64_t       i64_val;     /* CONST_INT/CONST_UINT payload, or column offset */
  double        f64_val;     /* CONST_REAL payload */
  const uint8_t *bytes_val;  /* CONST_BYTES / CONST_DECIMAL payload */
  uint32_t      bytes_len;
} CoprExprNode;

typedef struct CoprExpr {
  const CoprExprNode *nodes; /* RPN order */
  uint32_t            n_nodes;
} CoprExpr;

/* ---- aggregate definitions (mirrors tipb ExprType agg variants;
 *      parser: tidb_query_aggr/src/parser.rs AllAggrDefinitionParser) ---- */
enum CoprAggFunc {
  COPR_AGG_COUNT = 0,   /* impl_count.rs */
  COPR_AGG_SUM,         /* impl_sum.rs: int input rewritten to Decimal
                           (util::rewrite_exp_for_sum_avg) */
  COPR_AGG_AVG,         /* impl_avg.rs: emits (count, sum) — no division on
                           TiKV, impl_avg.rs:146-156 */
  COPR_AGG_MAX,         /* impl_max_min.rs */
  COPR_AGG_MIN,
  COPR_AGG_FIRST,       /* impl_first.rs */
  COPR_AGG_BIT_AND,     /* impl_bit_op.rs */
  COPR_AGG_BIT_OR,
  COPR_AGG_BIT_XOR
};

typedef struct CoprAggDef {
  int32_t       func;      /* CoprAggFunc */
  CoprExpr      arg;       /* the (single) argument expression */
  CoprFieldType out_ft;    /* tipb root Expr.field_type of the aggregate */
} CoprAggDef;

/* ---- executors (mirrors tipb::Executor; build_executors runner.rs:307) ---- */
enum CoprExecKind {
  COPR_EXEC_TABLE_SCAN = 0,   /* BatchTableScanExecutor   */
  COPR_EXEC_INDEX_SCAN,       /* BatchIndexScanExecutor   */
  COPR_EXEC_SELECTION,        /* BatchSelectionExecutor   */
  COPR_EXEC_SIMPLE_AGG,       /* BatchSimpleAggregationExecutor */
  COPR_EXEC_FAST_HASH_AGG,    /* BatchFastHashAggregationExecutor */
  COPR_EXEC_SLOW_HASH_AGG,    /* BatchSlowHashAggregationExecutor */
  COPR_EXEC_STREAM_AGG,       /* BatchStreamAggregationExecutor */
  COPR_EXEC_LIMIT,            /* BatchLimitExecutor       */
  COPR_EXEC_TOPN,             /* BatchTopNExecutor        */
  COPR_EXEC_PROJECTION        /* BatchProjectionExecutor  */
};

typedef struct CoprExecutor {
  int32_t kind;   /* CoprExecKind */
  /* TABLE_SCAN / INDEX_SCAN */
  const CoprColumnInfo *columns;
  uint32_t              n_columns;
  int32_t               desc;            /* scan order (0 = forward) */
  /* SELECTION: conditions (ANDed, each a predicate — selection_executor.rs:86) */
  const CoprExpr *conditions;
  uint32_t        n_conditions;
  /* aggregations */
  const CoprExpr *group_by;     /* group-by expressions */
  uint32_t        n_group_by;
  const CoprAggDef *aggs;
  uint32_t          n_aggs;
  /* LIMIT / TOPN */
  uint64_t limit;
  /* TOPN: order-by expressions (top_n_executor.rs) with per-expression
     descending flags (NULL sorts first; desc reverses the whole order) */
  const CoprExpr *order_by;
  const int32_t  *order_desc;
  uint32_t        n_order_by;
} CoprExecutor;

typedef struct CoprDagRequest {
  const CoprExecutor *executors;   /* root-last order, exactly like the tipb
                                      executors list (leaf scan first) */
  uint32_t            n_executors;
  const uint32_t     *output_offsets;
  uint32_t            n_output_offsets;
  /* EvalConfig subset that the BASELINE configs depend on
     (tidb_query_datatype/src/expr/ctx.rs:65-118) */
  uint64_t flags;                  /* Flag bitset */
  int32_t  div_precision_increment;
  /* paging (runner.rs:92-126): 0 = disabled. Only valid with a single
     region per request (the reference pages per coprocessor task = per
     Region); copr_dag_run returns COPR_ERR_UNSUPPORTED for paging_size != 0
     with n_regions > 1. */
  uint64_t paging_size;
  /* response encoding (tipb EncodeType, runner.rs:1188-1225):
     0 = TypeDefault (datum rows), 1 = TypeChunk (column-oriented chunks:
     per batch, per output column: u32le length, u32le null_cnt,
     [bitmap ceil(len/8) if null_cnt>0, bit set = NOT NULL],
     [i64le var_offsets x (len+1) for var-size], data -- chunk/column.rs:
     1052-1071; ints 8B LE, decimals the 40B struct dump
     decimal.rs:2135-2142, bytes raw payload) */
  int32_t encode_type;
} CoprDagRequest;

/* ---- results ---- */
typedef struct CoprExecSummary {   /* ExecSummary, execute_stats.rs:8 */
  uint64_t num_produced_rows;
  uint64_t num_iterations;
  uint64_t time_processed_ns;
} CoprExecSummary;

typedef struct CoprSelectResult {
  /* SelectResponse chunk payload: datum-encoded rows restricted to
     output_offsets (runner.rs:1188 TypeDefault arm). Owned by the engine;
     free with the matching *_result_free. */
  uint8_t  *data;
  uint64_t  data_len;
  uint64_t  n_rows;
  CoprExecSummary *summaries;      /* one per executor slot */
  uint32_t         n_summaries;
  /* paging resume range (IntervalRange, scanner.rs:209): row index the scan
     stopped at; UINT64_MAX = drained */
  uint64_t resume_row;
} CoprSelectResult;

#ifdef __cplusplus
}
#endif
#endif /* COPR_TYPES_H */
