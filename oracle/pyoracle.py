"""Pure-Python restatement of Arroyo's sliding/tumbling window-aggregate
operator semantics.  TEST INFRASTRUCTURE ONLY: this module is the
cross-checker used to validate the C oracle (oracle/arroyo_oracle.c) and to
generate golden fixtures; nothing in the product path may import it.

Restated from (all paths relative to the reference repo ArroyoSystems/arroyo):
  - sliding window operator:
      crates/arroyo-worker/src/arrow/sliding_aggregating_window.rs
      bin_start :90-99, should_advance :102-113, advance :115-210,
      process_batch :598-674 (late-data drop :631-633), state machine :63-73
  - tumbling window operator:
      crates/arroyo-worker/src/arrow/tumbling_aggregating_window.rs
      bin_start :65-74, process_batch :250-319, handle_watermark :321-392
  - state table (for get_min_time / expire semantics):
      crates/arroyo-state/src/tables/expiring_time_key_map.rs :826-929
      (flush retention cutoff :856-860, expire_timestamp exact-key :886-898,
       get_min_time :919-928); retention = window width
      (timestamp_table_config call at sliding_aggregating_window.rs:739-752)
  - watermark generator:
      crates/arroyo-worker/src/arrow/watermark_generator.rs :150-196
      (emit when batch max_ts - last_emitted_at > period(1s); value = min over
       batch of (expr); default expr = _timestamp - 1s,
       crates/arroyo-planner/src/rewriters.rs:48-86); final watermark at
      EndOfData = from_nanos(u64::MAX) (:131-148)
  - final projection (window struct + output timestamp):
      crates/arroyo-planner/src/extension/aggregate.rs :292-390
      window = (bin_start, bin_start+width); _timestamp = bin_start+width-1ns
  - partial/final aggregate split (COUNT partial=count/final=sum, MIN/MAX/SUM
      partial=running/final=same, AVG partial=(count,sum f64)/final=divide):
      crates/arroyo-planner/src/builder.rs:135-199 (DataFusion 48.0.1
      partial/final AggregateExec semantics; dependency not vendored in the
      reference -- pinned instead by the reference's own golden vectors, see
      tests/golden/).

Pinned against the reference's golden vectors (tests/test_oracle_golden.py):
  sliding_window_end, hourly_by_event_type, tight_watermark,
  most_active_driver_last_hour.

Timestamps are u64 nanoseconds since the unix epoch; the end-of-stream
watermark is 2**64-1 (matching from_nanos(u64::MAX)).
"""

NS = 10**9
U64MAX = 2**64 - 1

COUNT, SUM, MIN, MAX, AVG = 0, 1, 2, 3, 4


def bin_start(ts, width):
    # sliding_aggregating_window.rs:90-99 (nanos -= nanos % slide)
    if width == 0:
        return ts
    return ts - ts % width


class Agg:
    """Aggregate spec: list of (op, value_col_index)."""

    def __init__(self, specs):
        self.specs = specs

    def init(self):
        st = []
        for k, _ in self.specs:
            if k == COUNT:
                st.append(0)
            elif k == AVG:
                st.append([0, 0.0])
            else:
                st.append(None)
        return st

    def update(self, st, row):
        for i, (k, c) in enumerate(self.specs):
            if k == COUNT:
                st[i] += 1
            elif k == MIN:
                v = row[c]
                st[i] = v if st[i] is None else min(st[i], v)
            elif k == MAX:
                v = row[c]
                st[i] = v if st[i] is None else max(st[i], v)
            elif k == SUM:
                v = row[c]
                st[i] = v if st[i] is None else st[i] + v
            elif k == AVG:
                st[i][0] += 1
                st[i][1] += float(row[c])

    def merge(self, a, b):
        for i, (k, _) in enumerate(self.specs):
            if k == COUNT:
                a[i] += b[i]
            elif k == MIN:
                a[i] = b[i] if a[i] is None else (a[i] if b[i] is None else min(a[i], b[i]))
            elif k == MAX:
                a[i] = b[i] if a[i] is None else (a[i] if b[i] is None else max(a[i], b[i]))
            elif k == SUM:
                a[i] = b[i] if a[i] is None else (a[i] if b[i] is None else a[i] + b[i])
            elif k == AVG:
                a[i][0] += b[i][0]
                a[i][1] += b[i][1]

    def copy_state(self, st):
        return [list(x) if isinstance(x, list) else x for x in st]

    def finalize(self, st):
        out = []
        for i, (k, _) in enumerate(self.specs):
            if k == AVG:
                out.append(st[i][1] / st[i][0] if st[i][0] else None)
            else:
                out.append(st[i])
        return out


class SlidingWindow:
    """sliding_aggregating_window.rs state machine.

    Emitted rows: (key_tuple, [finalized aggs], window_start, window_end,
    out_timestamp) where window=(E-width, E), out_timestamp=E-1ns.
    """

    NO_DATA, BUFFERED, IN_MEMORY = 0, 1, 2

    def __init__(self, width, slide, key_cols, agg):
        self.width = width
        self.slide = slide
        self.key_cols = key_cols
        self.agg = agg
        self.open = {}      # bin -> {key: partial state}   (= execs)
        self.closed = {}    # bin -> {key: partial state}   (= TieredRecordBatchHolder)
        self.table = set()  # bins in the state table       (= ExpiringTimeKeyView keys)
        self.state = self.NO_DATA
        self.earliest = None
        self.next = None
        self.out = []
        self.wm = None      # last present watermark

    def bs(self, ts):
        return bin_start(ts, self.slide)

    def process_batch(self, rows):
        # The reference sorts the batch by bin and walks partition ranges
        # (:610-625); per-row processing is equivalent: the late check and the
        # state transitions commute across rows of one batch.
        for row in rows:
            b = self.bs(row[-1])
            if self.wm is not None and b < self.bs(self.wm):
                continue  # late-data drop :631-633
            if self.state == self.NO_DATA:
                self.state, self.earliest = self.BUFFERED, b
            elif self.state == self.BUFFERED:
                self.earliest = min(self.earliest, b)
            tbl = self.open.setdefault(b, {})
            k = tuple(row[c] for c in self.key_cols)
            st = tbl.get(k)
            if st is None:
                st = self.agg.init()
                tbl[k] = st
            self.agg.update(st, row)

    def should_advance(self, wm):
        # :102-113
        wb = self.bs(wm)
        if self.state == self.NO_DATA:
            return False
        base = self.earliest if self.state == self.BUFFERED else self.next
        return base + self.slide <= wb

    def handle_watermark(self, wm):
        self.wm = wm
        while self.should_advance(wm):
            self.advance()

    def advance(self):
        # :115-210
        b = self.earliest if self.state == self.BUFFERED else self.next
        E = b + self.slide
        # partial_table.flush(Some(bin_end)): retention(=width) cutoff :131
        self.table = {x for x in self.table if x >= E - self.width}
        if b in self.open:
            pane = self.open.pop(b)
            tgt = self.closed.setdefault(b, {})
            for k, st in pane.items():
                if k in tgt:
                    self.agg.merge(tgt[k], st)
                else:
                    tgt[k] = st
            self.table.add(b)  # partial_table.insert :151
        self.table.discard(E - self.width + self.slide)  # expire_timestamp :160
        merged = {}
        for pb in sorted(self.closed):
            if E - self.width <= pb < E:  # batches_for_interval :161-167
                for k, st in self.closed[pb].items():
                    if k in merged:
                        self.agg.merge(merged[k], st)
                    else:
                        merged[k] = self.agg.copy_state(st)
        # delete_before(bin_end + slide - width) :173-174
        self.closed = {pb: v for pb, v in self.closed.items()
                       if pb >= E + self.slide - self.width}
        for k, st in merged.items():
            self.out.append((k, self.agg.finalize(st), E - self.width, E, E - 1))
        # state transition :176-187
        if not self.closed:
            if self.table:
                self.state = self.BUFFERED
                self.earliest = self.bs(min(self.table))
            else:
                self.state = self.NO_DATA
        else:
            self.state = self.IN_MEMORY
            self.next = E

    def checkpoint_drain(self):
        """handle_checkpoint :693-737: open bins' partial states are written
        into the state table (affects get_min_time / the BUFFERED state)."""
        drained = []
        for b in sorted(self.open):
            self.table.add(b)
            for k, st in self.open[b].items():
                drained.append((b, k, self.agg.copy_state(st)))
        return drained


class TumblingWindow:
    """tumbling_aggregating_window.rs; fires every bin strictly below the
    watermark's bin (:327-334)."""

    def __init__(self, width, key_cols, agg):
        self.width = width
        self.key_cols = key_cols
        self.agg = agg
        self.open = {}
        self.out = []
        self.wm = None

    def bs(self, ts):
        return bin_start(ts, self.width)

    def process_batch(self, rows):
        for row in rows:
            b = self.bs(row[-1])
            if self.wm is not None and b < self.bs(self.wm):
                continue  # :282-291
            tbl = self.open.setdefault(b, {})
            k = tuple(row[c] for c in self.key_cols)
            st = tbl.get(k)
            if st is None:
                st = self.agg.init()
                tbl[k] = st
            self.agg.update(st, row)

    def handle_watermark(self, wm):
        self.wm = wm
        wb = self.bs(wm)
        while self.open:
            b = min(self.open)
            if b >= wb:
                break
            pane = self.open.pop(b)
            for k, st in pane.items():
                self.out.append((k, self.agg.finalize(st), b, b + self.width,
                                 b + self.width - 1))


class WatermarkGen:
    """watermark_generator.rs:150-196. expression = ts - lateness."""

    def __init__(self, lateness_ns, interval_ns=NS):
        self.lateness = lateness_ns
        self.interval = interval_ns
        self.last_emitted_at = 0
        self.idle = False

    def on_batch(self, ts_list):
        mx = max(ts_list)
        wm = min(t - self.lateness for t in ts_list)
        if self.idle or mx - self.last_emitted_at > self.interval:
            self.last_emitted_at = mx
            self.idle = False
            return wm
        return None


def run_pipeline(rows, op, lateness_ns, batch_size=32):
    """Single-subtask harness: source batches -> watermark generator ->
    operator, then the EndOfData final watermark (u64::MAX).  Mirrors the
    smoke-test setup (source_batch_size=32, smoke_tests.rs:53); batch is
    forwarded before its watermark (watermark_generator.rs:152-155)."""
    wg = WatermarkGen(lateness_ns)
    for i in range(0, len(rows), batch_size):
        batch = rows[i:i + batch_size]
        op.process_batch(batch)
        wm = wg.on_batch([r[-1] for r in batch])
        if wm is not None:
            op.handle_watermark(wm)
    op.handle_watermark(U64MAX)
    return op.out
