/* Shared POD types for the arroyo-amd C ABI.
 *
 * These mirror the configuration and batch contract of the reference's
 * windowed-aggregate operators (ArroyoSystems/arroyo):
 *   - AmdWindowConfig <-> api::SlidingWindowAggregateOperator /
 *     api::TumblingWindowAggregateOperator protobufs decoded at
 *     crates/arroyo-worker/src/arrow/sliding_aggregating_window.rs:449-528 and
 *     tumbling_aggregating_window.rs:111-201 (width/slide micros, input
 *     schema with key columns first, partial/final aggregation plans).  The
 *     serialized DataFusion plans are replaced by an explicit aggregate spec
 *     (op + input column), which is the information content of the plans for
 *     the supported aggregate set.
 *   - batch columns <-> Arrow RecordBatch per ArroyoSchema
 *     (crates/arroyo-rpc/src/df.rs:24-30: schema + timestamp_index +
 *     key_indices); key columns first, `_timestamp` (ns) last.
 *
 * Round-1 scope: fixed-width i64 key (0 or 1 key columns), i64 value
 * columns, non-nullable.  Timestamps are u64 nanoseconds since the epoch;
 * the end-of-stream watermark is UINT64_MAX (arroyo-worker watermark
 * generator on_close, watermark_generator.rs:131-148).
 */
#ifndef ARROYO_AMD_TYPES_H
#define ARROYO_AMD_TYPES_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum AmdAggOp {
    AMD_AGG_COUNT = 0, /* partial: count        final: sum of counts  */
    AMD_AGG_SUM   = 1, /* partial: sum (i64)    final: sum            */
    AMD_AGG_MIN   = 2, /* partial: min          final: min            */
    AMD_AGG_MAX   = 3, /* partial: max          final: max            */
    AMD_AGG_AVG   = 4, /* partial: (count,sum f64)  final: sum/divide */
    AMD_AGG_COUNT_DISTINCT = 5, /* updating aggregate only: exact count of
                                   distinct values under append+retract */
    /* round 2, updating aggregate only (every_aggregate.sql coverage):
     * retractable co-moment states.  n is the key's live row count (the
     * aggregated column is non-null in this ABI), state = (sum(x) f64,
     * sum(x^2) f64); sample variants emit NaN when n < 2 (the reference
     * emits SQL NULL there). */
    AMD_AGG_STDDEV = 6,
    AMD_AGG_STDDEV_POP = 7,
    AMD_AGG_VAR = 8,
    AMD_AGG_VAR_POP = 9,
    AMD_AGG_BIT_XOR = 10, /* xor is its own inverse: retract == append */
    /* two-argument co-moment family (updating aggregate only): state =
     * (sum y, sum x, sum xy, sum y^2, sum x^2) f64; the FIRST argument
     * (SQL's Y) is agg_col, the second (X) agg_col2.  Sample variants and
     * zero-variance cases emit NaN where the reference emits SQL NULL. */
    AMD_AGG_COVAR_POP = 11,
    AMD_AGG_COVAR_SAMP = 12,
    AMD_AGG_CORR = 13,
    AMD_AGG_REGR_SLOPE = 14,
    AMD_AGG_REGR_INTERCEPT = 15,
    AMD_AGG_REGR_R2 = 16,
    AMD_AGG_REGR_AVGX = 17,
    AMD_AGG_REGR_AVGY = 18,
    AMD_AGG_REGR_COUNT = 19,   /* i64 output */
    AMD_AGG_REGR_SXX = 20,
    AMD_AGG_REGR_SYY = 21,
    AMD_AGG_REGR_SXY = 22,
    /* retractable bit aggregates: 64 per-bit u32 set-counts (32 state
     * words); AND = bit set in every live row, OR = in any.  bool_and /
     * bool_or are these over a 0/1 column. */
    AMD_AGG_BIT_AND = 23,
    AMD_AGG_BIT_OR = 24
};

#define AMD_MAX_AGGS 8

typedef struct {
    uint64_t width_nanos;
    uint64_t slide_nanos;     /* ignored when is_tumbling */
    int32_t  is_tumbling;     /* 1: TumblingAggregatingWindowFunc semantics */
    int32_t  n_keys;          /* 0 or 1 (i64 key columns, first) */
    int32_t  n_aggs;
    int32_t  agg_ops[AMD_MAX_AGGS];
    int32_t  agg_col[AMD_MAX_AGGS];  /* value-column index; -1 for COUNT(*) */
    int32_t  n_value_cols;    /* value columns between keys and _timestamp */
    uint32_t log2_capacity;   /* hash slots per pane (GPU path) */
    uint32_t ring_panes;      /* live-pane ring size (GPU path, power of 2) */
    int32_t  device;          /* HIP device ordinal (GPU path) */
    int32_t  emit_to_host;    /* GPU path: 1 = outputs copied to host memory */
    int32_t  val_is_f64[8];   /* value column v carries f64 bit patterns in
                                 its i64 plane (Arrow Float64 column);
                                 SUM/MIN/MAX/AVG over it aggregate as
                                 doubles and the output column is f64 */
} AmdWindowConfig;

/* Instant (windowed stream-stream) join configuration.  Mirrors
 * api::JoinOperator as decoded by InstantJoinConstructor
 * (crates/arroyo-worker/src/arrow/instant_join.rs:372-412): two keyed input
 * schemas and a join plan.  The serialized DataFusion HashJoinExec is
 * replaced by its information content for the supported shape: equi-join on
 * the (single i64) key column within each exact `_timestamp` instant
 * (rows are routed to per-instant execs, instant_join.rs:109-172; fired in
 * timestamp order when the watermark passes, :256-283).  n_keys=0 means the
 * join condition is the window/instant itself (e.g. windowed_inner_join.sql:
 * ON dropoffs.window = pickups.window) and each instant emits the cross
 * product of its sides.  Input batch columns per side: [key?, vals...,
 * _timestamp]; output: [key?, left vals..., right vals..., _timestamp].
 * join_type selects the reference's JoinType (planner plan/join.rs maps
 * LEFT/RIGHT/FULL onto the same per-instant HashJoinExec): non-inner
 * output gains two trailing presence columns [left_present, right_present]
 * (the C ABI's stand-in for Arrow validity bitmaps; absent side's value
 * columns are zero-filled).  Unmatched rows emit once, at instant fire. */
#define AMD_JOIN_INNER 0
#define AMD_JOIN_LEFT  1
#define AMD_JOIN_RIGHT 2
#define AMD_JOIN_FULL  3
typedef struct {
    int32_t  n_keys;            /* 0 or 1 */
    int32_t  n_left_vals;
    int32_t  n_right_vals;
    uint32_t log2_rows_cap;     /* per-instant per-side row capacity (GPU) */
    uint32_t instants;          /* live-instant slots (GPU, power of two) */
    uint32_t log2_out_cap;      /* output rows per fire (GPU) */
    int32_t  device;
    int32_t  emit_to_host;
    int32_t  join_type;         /* AMD_JOIN_* (default inner) */
} AmdJoinConfig;

/* Session (gap) window aggregate configuration.  Mirrors
 * api::SessionWindowAggregateOperator as decoded by
 * SessionAggregatingWindowConstructor
 * (crates/arroyo-worker/src/arrow/session_aggregating_window.rs:707-772):
 * gap_micros + input schema (keys first) + final aggregation plan.  The
 * serialized DataFusion plan is replaced by the explicit aggregate spec, as
 * for AmdWindowConfig.  Semantics (restated in oracle/arroyo_oracle.c and
 * arroyo_amd/csrc/session.hip): per key, rows sorted by time form maximal
 * runs where each next timestamp is strictly within (previous max + gap)
 * (ActiveSession::add_batch :424-495); a session fires when
 * data_end + gap < watermark (KeyComputingHolder::watermark_update
 * :559-608); output window = [min_ts, max_ts + gap), _timestamp = end - 1
 * (to_record_batch :316-380); rows with ts < watermark are dropped
 * (process_batch :849-874, gt_eq filter). */
typedef struct {
    int32_t  n_keys;            /* 0 or 1 */
    int32_t  n_value_cols;
    int32_t  n_aggs;
    int32_t  agg_ops[AMD_MAX_AGGS];  /* incl. AMD_AGG_COUNT_DISTINCT (one
                                        per op on the GPU path): exact
                                        per-session distinct count via
                                        single-writer hash regions */
    int32_t  agg_col[AMD_MAX_AGGS];
    uint64_t gap_nanos;
    uint32_t log2_capacity;     /* key slots in the session store (GPU) */
    uint32_t max_sessions;      /* live sessions held inline per key (GPU) */
    uint32_t log2_batch_capacity; /* per-batch pre-aggregation table (GPU) */
    uint32_t log2_out_cap;      /* output rows per watermark (GPU) */
    uint32_t log2_distinct;     /* per-session distinct-set capacity (GPU) */
    uint32_t log2_cd_regions;   /* distinct-set region pool (GPU) */
    int32_t  device;
    int32_t  emit_to_host;
} AmdSessionConfig;

/* Non-windowed (TTL'd) stream-stream join configuration.  Mirrors
 * api::JoinOperator as decoded by JoinWithExpirationConstructor
 * (crates/arroyo-worker/src/arrow/join_with_expiration.rs:213-267): two
 * keyed schemas + ttl_micros + join plan.  Semantics: each incoming batch
 * is inserted into its side's per-key state (KeyTimeView,
 * crates/arroyo-state/src/tables/expiring_time_key_map.rs:932-1050) and
 * immediately joined against the OTHER side's stored rows for its keys
 * (process_left/process_right :42-108) — inner equi-join on the i64 key,
 * output _timestamp = max(left._timestamp, right._timestamp)
 * (post_join_timestamp_projection,
 * crates/arroyo-planner/src/plan/join.rs:121-191).  The live in-memory view
 * never evicts during a run (TTL filters state only on restore,
 * table_manager.rs:533-570 passes the watermark to get_view); restore drops
 * rows with ts < watermark - ttl.
 *
 * join_type (AMD_JOIN_*) selects the reference's JoinType for updating
 * joins (the planner passes the SQL join type into the DataFusion join
 * inside JoinWithExpiration, plan/join.rs:326-379); non-inner output gains
 * trailing [left_present, right_present] columns (Arrow-validity stand-in)
 * and, because a later match must retract an earlier null-padded row, a
 * trailing is_retract column.  updating=1 means inputs carry retractions:
 * per-side layout becomes [key, vals..., is_retract, _timestamp], the
 * output gains the trailing is_retract column, and emissions maintain the
 * join incrementally (append/retract pairs as either side's multiset
 * changes).  CPU oracle only in round 1: the GPU library rejects
 * join_type != 0 or updating != 0 at create (no silent fallback). */
typedef struct {
    int32_t  n_keys;            /* 1 (i64 equi-join key) */
    int32_t  n_left_vals;
    int32_t  n_right_vals;
    uint64_t ttl_nanos;
    uint32_t log2_capacity;     /* key slots per side (GPU) */
    uint32_t log2_rows_cap;     /* stored-row pool per side (GPU) */
    uint32_t log2_out_cap;      /* output rows per process_batch (GPU) */
    int32_t  device;
    int32_t  emit_to_host;
    int32_t  join_type;         /* AMD_JOIN_* (default inner) */
    int32_t  updating;          /* 1 = inputs carry is_retract */
} AmdExpJoinConfig;

/* Updating (non-windowed) aggregate configuration.  Mirrors
 * api::UpdatingAggregateOperator as decoded by
 * IncrementalAggregatingFunc's constructor
 * (crates/arroyo-worker/src/arrow/incremental_aggregator.rs:1035-1193):
 * aggregate exprs + flush interval + ttl.  Semantics (restated in
 * oracle/arroyo_oracle.c and arroyo_amd/csrc/updagg.hip): per key,
 * retractable accumulators updated row by row (appends and, when the input
 * carries _updating_meta.is_retract, retractions); on flush
 * (handle_tick -> flush :637-737) every key touched since the last flush
 * emits retract(previously emitted value) + append(new value) -- the
 * retract omitted for first-time keys, the append omitted (retract only)
 * when the key's rows have all been retracted, and the whole pair skipped
 * when the value did not change.  COUNT/SUM/AVG retract by subtraction
 * (the reference's Sliding accumulators); COUNT DISTINCT keeps an exact
 * per-key value multiset (the reference's Batch accumulator,
 * IncrementalState::Batch :84-174); MIN/MAX are append-only here (the
 * reference re-aggregates a stored multiset on demand; unsupported-retract
 * is a loud error).  Flush cadence is driven by the caller (the reference's
 * tick timer). */
typedef struct {
    int32_t  n_keys;            /* 0 or 1 */
    int32_t  n_value_cols;
    int32_t  n_aggs;
    int32_t  agg_col2[AMD_MAX_AGGS]; /* second argument (X) of the
                                        co-moment family; -1 otherwise */
    int32_t  agg_ops[AMD_MAX_AGGS];
    int32_t  agg_col[AMD_MAX_AGGS];
    uint32_t log2_capacity;     /* key slots (GPU) */
    uint32_t log2_nodes;        /* distinct-value node pool (GPU) */
    uint32_t log2_out_cap;
    int32_t  device;
    int32_t  emit_to_host;
} AmdUpdatingConfig;

/* SQL window-function (ROW_NUMBER) configuration.  Mirrors
 * api::WindowFunctionOperator as decoded by WindowFunctionConstructor
 * (crates/arroyo-worker/src/arrow/window_fn.rs:180-273): a
 * BoundedWindowAggExec run per exact `_timestamp` instant.  Semantics
 * (filter_and_split_batches :52-93, handle_watermark :220-246): rows are
 * buffered per instant (late rows ts < watermark silently filtered); when
 * the watermark passes an instant it fires in timestamp order, computing
 * ROW_NUMBER() OVER (PARTITION BY part_col ORDER BY order cols) per row;
 * rows with row_number > limit are dropped (the reference plans the
 * downstream filter separately; fusing it here saves materialising the
 * full ranking).  Output columns: input columns + trailing row_number. */
typedef struct {
    int32_t  n_cols;          /* input columns incl. trailing _timestamp */
    int32_t  part_col;        /* partition column index; -1 = whole instant */
    int32_t  n_order;         /* 1 or 2 ORDER BY columns */
    int32_t  order_col[2];
    int32_t  order_desc[2];
    int64_t  limit;           /* keep row_number <= limit; 0 = keep all */
    uint32_t log2_rows_cap;   /* per-instant row capacity (GPU) */
    uint32_t instants;        /* live-instant slots (GPU, power of two) */
    uint32_t log2_out_cap;
    int32_t  device;
    int32_t  emit_to_host;
} AmdWindowFnConfig;

/* Stateless map/filter/projection configuration.  Replaces the reference's
 * expression operators (crates/arroyo-worker/src/arrow/mod.rs):
 * ValueExecutionOperator :48, ProjectionOperator :99, KeyExecutionOperator
 * :180 -- each runs a serialized DataFusion expression plan over every
 * batch.  Here the expression plan's information content is a small
 * register program evaluated per row: registers r0..r(n_in_cols-1) are
 * preloaded with the input columns, instructions write higher registers,
 * `out_reg` names the emitted columns (KeyExecutionOperator = key exprs
 * first), and `filter_reg` (when >= 0) keeps rows whose register is
 * nonzero, preserving row order like the reference's filter kernels.
 * Integer division/modulo by zero is a loud error (DataFusion errors too);
 * f64 ops follow IEEE.  i64 values and f64 bit patterns share the
 * register file; AMD_MOP_I2F / F2I convert. */
enum AmdMapOp {
    AMD_MOP_CONST = 0,  /* dst = imm */
    AMD_MOP_ADD, AMD_MOP_SUB, AMD_MOP_MUL, AMD_MOP_DIV, AMD_MOP_MOD,
    AMD_MOP_EQ, AMD_MOP_NE, AMD_MOP_LT, AMD_MOP_LE, AMD_MOP_GT, AMD_MOP_GE,
    AMD_MOP_AND, AMD_MOP_OR, AMD_MOP_NOT,
    AMD_MOP_I2F, AMD_MOP_F2I,    /* value <-> f64 bit pattern (truncating) */
    AMD_MOP_FADD, AMD_MOP_FSUB, AMD_MOP_FMUL, AMD_MOP_FDIV
};

#define AMD_MAP_MAX_PROG 64
#define AMD_MAP_MAX_REGS 32
#define AMD_MAP_MAX_OUT  16

typedef struct {
    int32_t op;     /* AmdMapOp */
    int32_t a, b;   /* source registers (b unused for unary/CONST) */
    int32_t dst;
    int64_t imm;    /* CONST value (f64 ops: bit pattern) */
} AmdMapInstr;

typedef struct {
    int32_t n_in_cols;          /* input columns incl. trailing _timestamp */
    int32_t n_prog;
    AmdMapInstr prog[AMD_MAP_MAX_PROG];
    int32_t n_out;
    int32_t out_reg[AMD_MAP_MAX_OUT];
    int32_t out_is_f64[AMD_MAP_MAX_OUT];
    int32_t filter_reg;         /* -1 = no filter */
    int32_t device;
    int32_t emit_to_host;
} AmdMapConfig;

/* Output batch, allocated by the callee; free with *_free_out.
 * Column order: [key (if n_keys)], agg outputs (one column per agg),
 * window_start, window_end, _timestamp.  All columns are 8-byte elements;
 * is_f64[i] marks double columns (AVG outputs).  on_device=1 means `cols`
 * are HIP device pointers.
 *
 * validity: Arrow-layout per-column validity bitmaps (LSB order: bit
 * (r & 7) of byte (r >> 3) set = row r valid), or NULL when every column
 * is fully valid.  validity[c] == NULL marks column c all-valid.  The
 * outer-join operators set bitmaps on the null-padded value columns (the
 * Arrow RecordBatch contract of ArroyoSchema, crates/arroyo-rpc/src/
 * df.rs:24-30); the presence columns remain alongside for callers that
 * prefer flags. */
typedef struct {
    int64_t   n_rows;
    int32_t   n_cols;
    void    **cols;
    int32_t  *is_f64;
    int32_t   on_device;
    uint8_t **validity;
} AmdOutBatch;

#ifdef __cplusplus
}
#endif

#endif /* ARROYO_AMD_TYPES_H */
