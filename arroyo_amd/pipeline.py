"""Single-subtask pipeline harness: source batching -> watermark generator ->
window operator -> collected outputs.

Mirrors the reference's per-task event loop and watermark plumbing
(crates/arroyo-operator/src/operator.rs:982-1062 `operator_run_behavior`;
crates/arroyo-worker/src/arrow/watermark_generator.rs:150-196): each data
batch is forwarded to the operator, then the watermark generator may emit a
watermark derived from it (the batch precedes its watermark); when the source
ends, a final watermark of u64::MAX is broadcast
(watermark_generator.rs:131-148).
"""
import numpy as np

NS = 10**9
U64MAX = 2**64 - 1


class WatermarkGen:
    """watermark_generator.rs:150-196; expression = _timestamp - lateness
    (default lateness 1s: arroyo-planner/src/rewriters.rs:48-86)."""

    def __init__(self, lateness_ns=NS, interval_ns=NS):
        self.lateness = lateness_ns
        self.interval = interval_ns
        self.last_emitted_at = 0
        self.idle = False

    def on_batch(self, ts_col):
        mx = int(ts_col.max())
        wm = int(ts_col.min()) - self.lateness
        if self.idle or mx - self.last_emitted_at > self.interval:
            self.last_emitted_at = mx
            self.idle = False
            return wm
        return None


def run_stream(op, batches, lateness_ns=NS, final_watermark=True):
    """Drive `op` (a cabi.WindowOp or oracle op) over `batches` (each a list
    of np.int64 columns, `_timestamp` last).  Returns list of emitted output
    column-sets (one per watermark that produced rows)."""
    wg = WatermarkGen(lateness_ns)
    outs = []
    for cols in batches:
        op.process_batch(cols)
        wm = wg.on_batch(cols[-1])
        if wm is not None:
            out = op.handle_watermark(wm)
            if out and len(out[0]):
                outs.append(out)
    if final_watermark:
        out = op.handle_watermark(U64MAX)
        if out and len(out[0]):
            outs.append(out)
    return outs


def concat_outputs(outs):
    """Concatenate emitted batches into one column set."""
    if not outs:
        return None
    n_cols = len(outs[0])
    return [np.concatenate([o[i] for o in outs]) for i in range(n_cols)]


def batches_from_columns(cols, batch_size):
    n = len(cols[-1])
    return [[c[i:i + batch_size] for c in cols] for i in range(0, n, batch_size)]
