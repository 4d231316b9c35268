/* arroyo-amd C ABI: the drop-in boundary for the MI355X-native execution
 * path of Arroyo's windowed-aggregate operators.
 *
 * Each entry point replaces one method of the reference's `ArrowOperator`
 * trait (ArroyoSystems/arroyo, crates/arroyo-operator/src/operator.rs), with
 * the same ownership and threading contract: one thread of execution per
 * handle, operator receives owned immutable batches, emits owned batches.
 * The Rust host would bind these over FFI (see INTEGRATION.md for the
 * ready-to-paste binding) exactly where OperatorConstructor::with_config and
 * the operator methods are invoked today:
 *
 *   arroyo_amd_create           <-> OperatorConstructor::with_config
 *                                   (operator.rs:56-63; config contents from
 *                                   api::SlidingWindowAggregateOperator,
 *                                   sliding_aggregating_window.rs:449-528)
 *   arroyo_amd_process_batch    <-> ArrowOperator::process_batch
 *                                   (operator.rs:1183;
 *                                   sliding_aggregating_window.rs:598-674)
 *   arroyo_amd_handle_watermark <-> ArrowOperator::handle_watermark
 *                                   (operator.rs:1208; sliding :676-691);
 *                                   emitted batches = Collector::collect
 *                                   calls (context.rs:491-494)
 *   arroyo_amd_checkpoint_drain <-> ArrowOperator::handle_checkpoint
 *                                   (operator.rs:1218; sliding :693-737)
 *   arroyo_amd_restore          <-> ArrowOperator::on_start restore
 *                                   (operator.rs:1167; sliding :556-595)
 *   arroyo_amd_destroy          <-> operator drop
 *   arroyo_amd_partition        <-> ArrowCollector::repartition +
 *                                   server_for_hash_array
 *                                   (context.rs:506-560,
 *                                   arroyo-operator/src/lib.rs:30-41) --
 *                                   device-side partitioning for the keyed
 *                                   shuffle; the exchange itself is RCCL
 *                                   all-to-all over xGMI at harness level.
 *
 * Batches are Arrow-layout columns (8-byte fixed-width values buffers, key
 * columns first, `_timestamp` in nanoseconds last -- ArroyoSchema,
 * arroyo-rpc/src/df.rs:24-30).  Timestamps/watermarks are u64 ns since the
 * epoch; the end-of-stream watermark is UINT64_MAX.  Errors: non-zero return
 * + arroyo_amd_last_error(handle).
 */
#ifndef ARROYO_AMD_H
#define ARROYO_AMD_H

#include "arroyo_amd_types.h"

#ifdef __cplusplus
extern "C" {
#endif

/* Create a sliding/tumbling window-aggregate operator on cfg->device.
 * Returns NULL on failure (message via arroyo_amd_last_error(NULL)).
 * Requires a HIP device: there is no CPU fallback. */
void *arroyo_amd_create(const AmdWindowConfig *cfg);

/* Ingest one batch of host-memory columns
 * [keys..., values..., _timestamp], each `n_rows` i64 values. */
int arroyo_amd_process_batch(void *h, const int64_t *const *cols,
                             int32_t n_cols, int64_t n_rows);

/* Same, with columns already resident in device HBM; ts_offset is added to
 * every timestamp (synthetic-stream ring replay). */
int arroyo_amd_process_batch_device(void *h, const int64_t *const *dcols,
                                    int32_t n_cols, int64_t n_rows,
                                    uint64_t ts_offset);

/* Submit `reps` device-resident batches in one call (one kernel launch per
 * batch).  contiguous=1: batch k = rows [k*n_rows,(k+1)*n_rows) of dcols;
 * contiguous=0: the same rows replayed.  ts_offset advances by ts_step per
 * batch.  Keeps per-batch FFI overhead off the hot loop. */
int arroyo_amd_process_batches_device(void *h, const int64_t *const *dcols,
                                      int32_t n_cols, int64_t n_rows,
                                      int32_t reps, int32_t contiguous,
                                      uint64_t ts_offset0, uint64_t ts_step);

/* Streaming-read bandwidth calibration over two n-row i64 device columns;
 * returns GB/s (measurement tooling). */
double arroyo_amd_stream_gbps(const void *d_a, const void *d_b, int64_t n,
                              int iters);

/* Synchronize the operator's internal stream: callers that reuse
 * device buffers handed to process_batch_device (e.g. the N>1 bench's
 * RCCL exchange buffers) must fence the consuming kernels first. */
int arroyo_amd_sync(void *h);

/* Advance the watermark; fires every window the reference would fire, in
 * order, and returns the emitted rows (all fired windows concatenated;
 * column order [key?, aggs..., window_start, window_end, _timestamp]).
 * `out` may be NULL to defer collection to a later call. */
int arroyo_amd_handle_watermark(void *h, uint64_t watermark_nanos,
                                AmdOutBatch *out);

/* Batched form of handle_watermark for a group of watermarks with NO
 * process_batch between them (the periodic WatermarkGenerator emits such
 * runs whenever the source idles; the bench's fused periods do too).
 * Equivalent to calling handle_watermark(wms[i]) in order, but the
 * device status (pane tags, min non-late bin) is read ONCE for the whole
 * group — each per-watermark read costs a stream sync (~15 us idle). */
int arroyo_amd_handle_watermarks(void *h, const uint64_t *wms, int32_t n,
                                 AmdOutBatch *out);

/* Epoch pipelining: lets a harness submit the NEXT period's batches
 * before folding this period's watermarks, hiding the fold's host
 * latency behind the next period's kernels.  Sequence:
 *   submit(g); mark_epoch(); set_filter_watermark(last wm of group g);
 *   submit(g+1); handle_watermarks_epoch(group g's wms, out);
 * mark_epoch snapshots the device status at period g's stream point
 * (depth 2).  set_filter_watermark pre-advances the ingest late-drop
 * cutoff to the watermark the deferred group will establish, so the
 * next period's rows are filtered exactly as in the sequential order;
 * emissions are bit-identical to sequential handle_watermark calls. */
int arroyo_amd_set_filter_watermark(void *h, uint64_t watermark_nanos);
int arroyo_amd_mark_epoch(void *h);
int arroyo_amd_handle_watermarks_epoch(void *h, const uint64_t *wms,
                                       int32_t n, AmdOutBatch *out);

/* Drain open panes' partial states for a checkpoint barrier.  Columns:
 * [key?, partial state words (AVG takes 2)..., bin _timestamp]. */
int arroyo_amd_checkpoint_drain(void *h, AmdOutBatch *out);

/* Restore checkpointed partial states (layout as checkpoint_drain). */
int arroyo_amd_restore(void *h, const int64_t *const *cols, int32_t n_cols,
                       int64_t n_rows, int has_watermark,
                       uint64_t watermark_nanos);

void arroyo_amd_free_out(AmdOutBatch *out);
void arroyo_amd_destroy(void *h);
const char *arroyo_amd_last_error(void *h);

/* Dominant-kernel timing for the measurement harness: total k_update time
 * (HIP events on the operator's stream), rows processed, launches, and rows
 * emitted device-side since the last call. */
int arroyo_amd_perf(void *h, double *update_ms, int64_t *rows,
                    int64_t *launches, int64_t *emitted_device_rows);

/* Device-side key-hash partitioning (shuffle): writes rows regrouped into
 * per-partition contiguous segments and per-partition counts. */
int arroyo_amd_partition(const int64_t *d_keys, const int64_t *d_vals,
                         const int64_t *d_ts, int64_t n, uint32_t n_parts,
                         int64_t *d_out_keys, int64_t *d_out_vals,
                         int64_t *d_out_ts, uint64_t *h_counts);

/* ---- instant (windowed stream-stream) join ----------------------------
 * Replaces InstantJoin (crates/arroyo-worker/src/arrow/instant_join.rs)
 * behind the same ArrowOperator surface:
 *   join_create          <-> InstantJoinConstructor::with_config (:372-412)
 *   join_process_batch   <-> process_batch_index left/right routing
 *                            (:241-255 -> process_side :109-172); side 0 =
 *                            left, 1 = right; cols [key?, vals, _timestamp]
 *   join_handle_watermark<-> handle_watermark (:256-283): fires every
 *                            instant < wm in timestamp order; out columns
 *                            [key?, left vals..., right vals..., _timestamp]
 *   join_checkpoint_drain<-> handle_checkpoint (:285-303): one side's
 *                            buffered rows
 *   join_restore          = process_batch (on_start re-processes the
 *                            drained batches, :205-230)                  */
void *arroyo_amd_join_create(const AmdJoinConfig *cfg);
int arroyo_amd_join_process_batch(void *h, int32_t side,
                                  const int64_t *const *cols, int32_t n_cols,
                                  int64_t n_rows);
int arroyo_amd_join_process_batch_device(void *h, int32_t side,
                                         const int64_t *const *dcols,
                                         int32_t n_cols, int64_t n_rows);
int arroyo_amd_join_handle_watermark(void *h, uint64_t watermark_nanos,
                                     AmdOutBatch *out);
int arroyo_amd_join_checkpoint_drain(void *h, int32_t side, AmdOutBatch *out);
void arroyo_amd_join_destroy(void *h);
const char *arroyo_amd_join_last_error(void *h);

/* ---- session (gap) window aggregate -----------------------------------
 * Replaces SessionAggregatingWindowFunc
 * (crates/arroyo-worker/src/arrow/session_aggregating_window.rs) behind the
 * same ArrowOperator surface:
 *   session_create           <-> SessionAggregatingWindowConstructor
 *                                ::with_config (:707-772)
 *   session_process_batch    <-> process_batch (:849-893): late rows
 *                                (ts < watermark) silently dropped, rest
 *                                buffered into per-key session state
 *   session_handle_watermark <-> handle_watermark -> advance (:76-98,
 *                                :894-903): fires every session with
 *                                data_end + gap < watermark; out columns
 *                                [key?, aggs..., window_start, window_end,
 *                                _timestamp = window_end - 1]
 *   session_checkpoint_drain <-> handle_checkpoint (:905-921): live session
 *                                partial states [key?, state words...,
 *                                data_start, data_end]
 *   session_restore          <-> on_start (:803-846): reload drained
 *                                partial sessions
 */
void *arroyo_amd_session_create(const AmdSessionConfig *cfg);
int arroyo_amd_session_process_batch(void *h, const int64_t *const *cols,
                                     int32_t n_cols, int64_t n_rows);
int arroyo_amd_session_process_batch_device(void *h,
                                            const int64_t *const *dcols,
                                            int32_t n_cols, int64_t n_rows,
                                            uint64_t ts_offset);
int arroyo_amd_session_handle_watermark(void *h, uint64_t watermark_nanos,
                                        AmdOutBatch *out);
int arroyo_amd_session_checkpoint_drain(void *h, AmdOutBatch *out);
/* COUNT DISTINCT value stream for checkpoints: one row per live
 * (session, value): [key?, session_start, value]; restore after
 * session_restore (which zeroes the CD words it rebuilds) */
int arroyo_amd_session_drain_values(void *h, AmdOutBatch *out);
int arroyo_amd_session_restore_values(void *h, const int64_t *const *cols,
                                      int32_t n_cols, int64_t n_rows);
int arroyo_amd_session_restore(void *h, const int64_t *const *cols,
                               int32_t n_cols, int64_t n_rows);
void arroyo_amd_session_destroy(void *h);
const char *arroyo_amd_session_last_error(void *h);

/* ---- non-windowed (TTL'd) stream-stream join --------------------------
 * Replaces JoinWithExpiration
 * (crates/arroyo-worker/src/arrow/join_with_expiration.rs) behind the same
 * ArrowOperator surface:
 *   expjoin_create          <-> JoinWithExpirationConstructor::with_config
 *                               (:213-267)
 *   expjoin_process_batch   <-> process_batch_index (:162-180) ->
 *                               process_left/process_right (:42-108):
 *                               inserts the batch into its side's per-key
 *                               state AND returns the joined output rows
 *                               [key, left vals..., right vals...,
 *                               _timestamp = max(l_ts, r_ts)] against the
 *                               other side's stored rows
 *   expjoin_handle_watermark:   records the watermark (live state never
 *                               evicts during a run, table_manager.rs:533;
 *                               TTL applies on restore)
 *   expjoin_expire          :   explicit eviction of rows with
 *                               ts < watermark - ttl (bounded-memory mode;
 *                               matches what the reference's state layer
 *                               drops across checkpoint/restore)
 *   expjoin_checkpoint_drain<-> tables() flush: one side's stored rows
 *   expjoin_restore         <-> on_start: reload rows, dropping those with
 *                               ts < watermark - ttl (get_key_time_table
 *                               restore filter)
 */
void *arroyo_amd_expjoin_create(const AmdExpJoinConfig *cfg);
int arroyo_amd_expjoin_process_batch(void *h, int32_t side,
                                     const int64_t *const *cols,
                                     int32_t n_cols, int64_t n_rows,
                                     AmdOutBatch *out);
int arroyo_amd_expjoin_process_batch_device(void *h, int32_t side,
                                            const int64_t *const *dcols,
                                            int32_t n_cols, int64_t n_rows);
int arroyo_amd_expjoin_collect(void *h, AmdOutBatch *out);
int arroyo_amd_expjoin_handle_watermark(void *h, uint64_t watermark_nanos);

/* Device-resident consumption: report and reset the accumulated match
 * count without host copies (the next stage consumes the device match
 * columns in place). */
int arroyo_amd_expjoin_match_count(void *h, int64_t *n_matches);
int arroyo_amd_expjoin_expire(void *h);
int arroyo_amd_expjoin_checkpoint_drain(void *h, int32_t side,
                                        AmdOutBatch *out);
int arroyo_amd_expjoin_restore(void *h, int32_t side,
                               const int64_t *const *cols, int32_t n_cols,
                               int64_t n_rows, int has_watermark,
                               uint64_t watermark_nanos);
void arroyo_amd_expjoin_destroy(void *h);
const char *arroyo_amd_expjoin_last_error(void *h);

/* ---- updating (non-windowed) aggregate --------------------------------
 * Replaces IncrementalAggregatingFunc
 * (crates/arroyo-worker/src/arrow/incremental_aggregator.rs) behind the
 * same ArrowOperator surface:
 *   updagg_create        <-> constructor (:1035-1193)
 *   updagg_process_batch <-> process_batch (:931-949) ->
 *                            keyed_aggregate/global_aggregate (:778-884);
 *                            cols [key?, vals..., is_retract] where
 *                            is_retract mirrors _updating_meta.is_retract
 *                            (get_retracts :740-758)
 *   updagg_flush         <-> handle_tick/on_close -> flush (:637-737):
 *                            out columns [key?, agg outputs...,
 *                            is_retract]
 *   updagg_checkpoint_drain(which): which=0 scalar accumulator states
 *                            (the "a" table, checkpoint_sliding :271-340),
 *                            which=1 distinct-value multiset rows
 *                            [key?, agg_index, value, net_count] (the "b"
 *                            table, checkpoint_batch :342-418)
 *   updagg_restore       <-> on_start -> initialize (:446-599)
 */
void *arroyo_amd_updagg_create(const AmdUpdatingConfig *cfg);
int arroyo_amd_updagg_process_batch(void *h, const int64_t *const *cols,
                                    int32_t n_cols, int64_t n_rows);
int arroyo_amd_updagg_process_batch_device(void *h,
                                           const int64_t *const *dcols,
                                           int32_t n_cols, int64_t n_rows);
int arroyo_amd_updagg_flush(void *h, AmdOutBatch *out);
/* TTL eviction of keys idle for more than `idle_flushes` flush epochs
 * (the reference's UpdatingCache::time_out, wall-clock there); emits the
 * retract rows */
int arroyo_amd_updagg_expire(void *h, int64_t idle_flushes, AmdOutBatch *out);
int arroyo_amd_updagg_checkpoint_drain(void *h, int32_t which,
                                       AmdOutBatch *out);
int arroyo_amd_updagg_restore(void *h, int32_t which,
                              const int64_t *const *cols, int32_t n_cols,
                              int64_t n_rows);
void arroyo_amd_updagg_destroy(void *h);
const char *arroyo_amd_updagg_last_error(void *h);

/* ---- SQL window function (ROW_NUMBER per instant) ---------------------
 * Replaces WindowFunctionOperator
 * (crates/arroyo-worker/src/arrow/window_fn.rs) behind the same
 * ArrowOperator surface:
 *   windowfn_create           <-> WindowFunctionConstructor (:180-273)
 *   windowfn_process_batch    <-> process_batch (:275-300): buffers rows
 *                                 per exact instant; late rows silently
 *                                 filtered (filter_and_split_batches)
 *   windowfn_handle_watermark <-> handle_watermark (:220-246): fires
 *                                 instants < wm in timestamp order; out
 *                                 columns [input cols..., row_number]
 *   windowfn_checkpoint_drain <-> handle_checkpoint (:302-324): buffered
 *                                 rows; restore = process_batch
 */
void *arroyo_amd_windowfn_create(const AmdWindowFnConfig *cfg);
int arroyo_amd_windowfn_process_batch(void *h, const int64_t *const *cols,
                                      int32_t n_cols, int64_t n_rows);
int arroyo_amd_windowfn_process_batch_device(void *h,
                                             const int64_t *const *dcols,
                                             int32_t n_cols, int64_t n_rows);
int arroyo_amd_windowfn_restore(void *h, const int64_t *const *cols,
                                int32_t n_cols, int64_t n_rows);
int arroyo_amd_windowfn_handle_watermark(void *h, uint64_t watermark_nanos,
                                         AmdOutBatch *out);
int arroyo_amd_windowfn_checkpoint_drain(void *h, AmdOutBatch *out);
void arroyo_amd_windowfn_destroy(void *h);
const char *arroyo_amd_windowfn_last_error(void *h);

/* ---- stateless map / filter / projection ------------------------------
 * Replaces ValueExecutionOperator / ProjectionOperator /
 * KeyExecutionOperator (crates/arroyo-worker/src/arrow/mod.rs:48-243):
 *   map_create        <-> the operators' constructors (:70-97, :127-178,
 *                         :213-243): the serialized expression plan becomes
 *                         the register program in AmdMapConfig
 *   map_process_batch <-> process_batch (:56-68, :112-125, :196-211):
 *                         stateless, emits the projected (and filtered,
 *                         order-preserving) rows immediately
 * No state: nothing to checkpoint (the reference's tables() are empty). */
void *arroyo_amd_map_create(const AmdMapConfig *cfg);
int arroyo_amd_map_process_batch(void *h, const int64_t *const *cols,
                                 int32_t n_cols, int64_t n_rows,
                                 AmdOutBatch *out);
int arroyo_amd_map_process_batch_device(void *h, const int64_t *const *dcols,
                                        int32_t n_cols, int64_t n_rows,
                                        const int64_t **d_out_cols,
                                        int64_t *n_out_rows);
void arroyo_amd_map_destroy(void *h);
const char *arroyo_amd_map_last_error(void *h);

#ifdef __cplusplus
}
#endif

#endif /* ARROYO_AMD_H */
