"""ctypes bindings for the arroyo-amd C ABI (include/arroyo_amd_types.h).

Generic over the implementing library: the product HIP library
(libarroyo_amd.so, prefix ``arroyo_amd_``) and the CPU oracle
(oracle/liboracle.so, prefix ``oracle_``) export the same function set; tests
drive both through :class:`WindowOp` and compare.
"""
import ctypes

import numpy as np

NS = 10**9
U64MAX = 2**64 - 1

COUNT, SUM, MIN, MAX, AVG = 0, 1, 2, 3, 4


class AmdWindowConfig(ctypes.Structure):
    _fields_ = [
        ("width_nanos", ctypes.c_uint64),
        ("slide_nanos", ctypes.c_uint64),
        ("is_tumbling", ctypes.c_int32),
        ("n_keys", ctypes.c_int32),
        ("n_aggs", ctypes.c_int32),
        ("agg_ops", ctypes.c_int32 * 8),
        ("agg_col", ctypes.c_int32 * 8),
        ("n_value_cols", ctypes.c_int32),
        ("log2_capacity", ctypes.c_uint32),
        ("ring_panes", ctypes.c_uint32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
        ("val_is_f64", ctypes.c_int32 * 8),
    ]


class AmdOutBatch(ctypes.Structure):
    _fields_ = [
        ("n_rows", ctypes.c_int64),
        ("n_cols", ctypes.c_int32),
        ("cols", ctypes.POINTER(ctypes.c_void_p)),
        ("is_f64", ctypes.POINTER(ctypes.c_int32)),
        ("on_device", ctypes.c_int32),
        ("validity", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
    ]


def make_config(width_ns, slide_ns, aggs, n_keys=1, n_value_cols=0,
                is_tumbling=False, log2_capacity=20, ring_panes=64,
                device=0, emit_to_host=True, val_is_f64=()):
    """aggs: list of (op, value_col_index); value_col_index -1 for COUNT(*).
    val_is_f64: indices of value columns holding f64 bit patterns."""
    cfg = AmdWindowConfig()
    cfg.width_nanos = width_ns
    cfg.slide_nanos = width_ns if is_tumbling else slide_ns
    cfg.is_tumbling = 1 if is_tumbling else 0
    cfg.n_keys = n_keys
    cfg.n_aggs = len(aggs)
    for i, (op, col) in enumerate(aggs):
        cfg.agg_ops[i] = op
        cfg.agg_col[i] = col
    cfg.n_value_cols = n_value_cols
    for v in val_is_f64:
        cfg.val_is_f64[v] = 1
    cfg.log2_capacity = log2_capacity
    cfg.ring_panes = ring_panes
    cfg.device = device
    cfg.emit_to_host = 1 if emit_to_host else 0
    return cfg


def bind(lib, prefix):
    fn = {}
    g = lambda n: getattr(lib, prefix + n)
    fn["create"] = g("create")
    fn["create"].restype = ctypes.c_void_p
    fn["create"].argtypes = [ctypes.POINTER(AmdWindowConfig)]
    fn["process_batch"] = g("process_batch")
    fn["process_batch"].restype = ctypes.c_int
    fn["process_batch"].argtypes = [ctypes.c_void_p,
                                    ctypes.POINTER(ctypes.c_void_p),
                                    ctypes.c_int32, ctypes.c_int64]
    fn["handle_watermark"] = g("handle_watermark")
    fn["handle_watermark"].restype = ctypes.c_int
    fn["handle_watermark"].argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                       ctypes.POINTER(AmdOutBatch)]
    try:
        # batched variant (HIP library only; the oracle loops in Python)
        fn["handle_watermarks"] = g("handle_watermarks")
        fn["handle_watermarks"].restype = ctypes.c_int
        fn["handle_watermarks"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint64), ctypes.c_int32,
            ctypes.POINTER(AmdOutBatch)]
        # epoch pipelining (HIP library only)
        fn["set_filter_watermark"] = g("set_filter_watermark")
        fn["set_filter_watermark"].restype = ctypes.c_int
        fn["set_filter_watermark"].argtypes = [ctypes.c_void_p,
                                               ctypes.c_uint64]
        fn["mark_epoch"] = g("mark_epoch")
        fn["mark_epoch"].restype = ctypes.c_int
        fn["mark_epoch"].argtypes = [ctypes.c_void_p]
        fn["handle_watermarks_epoch"] = g("handle_watermarks_epoch")
        fn["handle_watermarks_epoch"].restype = ctypes.c_int
        fn["handle_watermarks_epoch"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint64), ctypes.c_int32,
            ctypes.POINTER(AmdOutBatch)]
    except AttributeError:
        pass
    fn["checkpoint_drain"] = g("checkpoint_drain")
    fn["checkpoint_drain"].restype = ctypes.c_int
    fn["checkpoint_drain"].argtypes = [ctypes.c_void_p,
                                       ctypes.POINTER(AmdOutBatch)]
    fn["free_out"] = g("free_out")
    fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
    fn["destroy"] = g("destroy")
    fn["destroy"].argtypes = [ctypes.c_void_p]
    fn["last_error"] = g("last_error")
    fn["last_error"].restype = ctypes.c_char_p
    fn["last_error"].argtypes = [ctypes.c_void_p]
    return fn


def out_validity(out):
    """Decode the Arrow validity bitmaps of an AmdOutBatch into per-column
    numpy bool masks (None = column all-valid)."""
    n = out.n_rows
    masks = []
    if not out.validity:
        return [None] * out.n_cols
    for i in range(out.n_cols):
        bm = out.validity[i]
        if not bm:
            masks.append(None)
            continue
        nbytes = (n + 7) // 8
        raw = np.frombuffer(
            bytes(ctypes.cast(bm, ctypes.POINTER(
                ctypes.c_uint8 * max(nbytes, 1))).contents),
            dtype=np.uint8)
        masks.append(np.unpackbits(raw, bitorder="little")[:n].astype(bool))
    return masks


def _out_to_numpy(out):
    """Copy an AmdOutBatch (host memory) into numpy arrays."""
    n = out.n_rows
    cols = []
    for i in range(out.n_cols):
        dt = np.float64 if out.is_f64[i] else np.int64
        if n == 0:
            cols.append(np.empty(0, dtype=dt))
            continue
        buf = ctypes.cast(out.cols[i], ctypes.POINTER(ctypes.c_int64 * n))
        arr = np.frombuffer(bytearray(bytes(buf.contents)), dtype=dt).copy()
        cols.append(arr)
    return cols


class AmdJoinConfig(ctypes.Structure):
    _fields_ = [
        ("n_keys", ctypes.c_int32),
        ("n_left_vals", ctypes.c_int32),
        ("n_right_vals", ctypes.c_int32),
        ("log2_rows_cap", ctypes.c_uint32),
        ("instants", ctypes.c_uint32),
        ("log2_out_cap", ctypes.c_uint32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
        ("join_type", ctypes.c_int32),
    ]


JOIN_INNER, JOIN_LEFT, JOIN_RIGHT, JOIN_FULL = 0, 1, 2, 3


def make_join_config(n_keys=1, n_left_vals=0, n_right_vals=0,
                     log2_rows_cap=15, instants=128, log2_out_cap=20,
                     device=0, emit_to_host=True, join_type=JOIN_INNER):
    cfg = AmdJoinConfig()
    cfg.n_keys = n_keys
    cfg.n_left_vals = n_left_vals
    cfg.n_right_vals = n_right_vals
    cfg.log2_rows_cap = log2_rows_cap
    cfg.instants = instants
    cfg.log2_out_cap = log2_out_cap
    cfg.device = device
    cfg.emit_to_host = 1 if emit_to_host else 0
    cfg.join_type = join_type
    return cfg


def bind_join(lib, prefix):
    fn = {}
    g = lambda n: getattr(lib, prefix + "join_" + n)
    fn["create"] = g("create")
    fn["create"].restype = ctypes.c_void_p
    fn["create"].argtypes = [ctypes.POINTER(AmdJoinConfig)]
    fn["process_batch"] = g("process_batch")
    fn["process_batch"].restype = ctypes.c_int
    fn["process_batch"].argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                    ctypes.POINTER(ctypes.c_void_p),
                                    ctypes.c_int32, ctypes.c_int64]
    fn["handle_watermark"] = g("handle_watermark")
    fn["handle_watermark"].restype = ctypes.c_int
    fn["handle_watermark"].argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                       ctypes.POINTER(AmdOutBatch)]
    fn["checkpoint_drain"] = g("checkpoint_drain")
    fn["checkpoint_drain"].restype = ctypes.c_int
    fn["checkpoint_drain"].argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                       ctypes.POINTER(AmdOutBatch)]
    fn["free_out"] = getattr(lib, prefix + "free_out")
    fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
    fn["destroy"] = g("destroy")
    fn["destroy"].argtypes = [ctypes.c_void_p]
    fn["last_error"] = g("last_error")
    fn["last_error"].restype = ctypes.c_char_p
    fn["last_error"].argtypes = [ctypes.c_void_p]
    return fn


class JoinOp:
    """One instant-join operator instance behind the C ABI (left = side 0,
    right = side 1), mirroring InstantJoin's ArrowOperator surface
    (crates/arroyo-worker/src/arrow/instant_join.rs)."""

    LEFT, RIGHT = 0, 1

    def __init__(self, lib, prefix, cfg):
        self._fn = bind_join(lib, prefix)
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(f"{prefix}join_create failed")

    def process_batch(self, side, cols):
        n_rows = len(cols[0]) if cols else 0
        arr = (ctypes.c_void_p * len(cols))()
        keep = []
        for i, c in enumerate(cols):
            c = np.ascontiguousarray(c, dtype=np.int64)
            keep.append(c)
            arr[i] = c.ctypes.data_as(ctypes.c_void_p).value
        rc = self._fn["process_batch"](self._h, side, arr, len(cols), n_rows)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        return keep

    def handle_watermark(self, wm):
        out = AmdOutBatch()
        rc = self._fn["handle_watermark"](self._h, wm, ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def checkpoint_drain(self, side):
        out = AmdOutBatch()
        rc = self._fn["checkpoint_drain"](self._h, side, ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def restore(self, side, cols):
        """on_start re-processes drained batches (instant_join.rs:205-230)."""
        if cols and len(cols[0]):
            self.process_batch(side, cols)

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class WindowOp:
    """One window-aggregate operator instance behind the C ABI.

    Mirrors the ArrowOperator calling convention
    (crates/arroyo-operator/src/operator.rs:1144-1257): process_batch
    ingests an owned batch; handle_watermark may emit result batches;
    single-threaded per handle."""

    def __init__(self, lib, prefix, cfg):
        self._fn = bind(lib, prefix)
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(f"{prefix}create failed (invalid config)")

    def process_batch(self, cols):
        """cols: list of np.int64 arrays [keys..., values..., _timestamp]."""
        n_rows = len(cols[0]) if cols else 0
        arr = (ctypes.c_void_p * len(cols))()
        keep = []
        for i, c in enumerate(cols):
            c = np.ascontiguousarray(c, dtype=np.int64)
            keep.append(c)
            arr[i] = c.ctypes.data_as(ctypes.c_void_p).value
        rc = self._fn["process_batch"](self._h, arr, len(cols), n_rows)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        return keep  # keep references alive through the call

    def handle_watermark(self, wm):
        out = AmdOutBatch()
        rc = self._fn["handle_watermark"](self._h, wm, ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def handle_watermarks(self, wms):
        """Batched watermarks with no rows between them: one device-status
        round trip serves the whole group.  Semantically identical to
        handle_watermark in order; returns all emissions concatenated.
        Falls back to a Python loop where the library lacks the entry
        point (the oracle)."""
        if "handle_watermarks" not in self._fn:
            outs = [self.handle_watermark(w) for w in wms]
            outs = [o for o in outs if o and len(o[0])]
            if not outs:
                return []
            return [np.concatenate([o[c] for o in outs])
                    for c in range(len(outs[0]))]
        out = AmdOutBatch()
        arr = (ctypes.c_uint64 * len(wms))(*wms)
        rc = self._fn["handle_watermarks"](self._h, arr, len(wms),
                                           ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def set_filter_watermark(self, wm):
        rc = self._fn["set_filter_watermark"](self._h, wm)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def mark_epoch(self):
        rc = self._fn["mark_epoch"](self._h)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def handle_watermarks_epoch(self, wms):
        out = AmdOutBatch()
        arr = (ctypes.c_uint64 * len(wms))(*wms)
        rc = self._fn["handle_watermarks_epoch"](self._h, arr, len(wms),
                                                 ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def checkpoint_drain(self):
        out = AmdOutBatch()
        rc = self._fn["checkpoint_drain"](self._h, ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class AmdSessionConfig(ctypes.Structure):
    _fields_ = [
        ("n_keys", ctypes.c_int32),
        ("n_value_cols", ctypes.c_int32),
        ("n_aggs", ctypes.c_int32),
        ("agg_ops", ctypes.c_int32 * 8),
        ("agg_col", ctypes.c_int32 * 8),
        ("gap_nanos", ctypes.c_uint64),
        ("log2_capacity", ctypes.c_uint32),
        ("max_sessions", ctypes.c_uint32),
        ("log2_batch_capacity", ctypes.c_uint32),
        ("log2_out_cap", ctypes.c_uint32),
        ("log2_distinct", ctypes.c_uint32),
        ("log2_cd_regions", ctypes.c_uint32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
    ]


def make_session_config(gap_ns, aggs, n_keys=1, n_value_cols=0,
                        log2_capacity=16, max_sessions=8,
                        log2_batch_capacity=14, log2_out_cap=20,
                        log2_distinct=10, log2_cd_regions=18,
                        device=0, emit_to_host=True):
    cfg = AmdSessionConfig()
    cfg.n_keys = n_keys
    cfg.n_value_cols = n_value_cols
    cfg.n_aggs = len(aggs)
    for i, (op, col) in enumerate(aggs):
        cfg.agg_ops[i] = op
        cfg.agg_col[i] = col
    cfg.gap_nanos = gap_ns
    cfg.log2_capacity = log2_capacity
    cfg.max_sessions = max_sessions
    cfg.log2_batch_capacity = log2_batch_capacity
    cfg.log2_out_cap = log2_out_cap
    cfg.log2_distinct = log2_distinct
    cfg.log2_cd_regions = log2_cd_regions
    cfg.device = device
    cfg.emit_to_host = 1 if emit_to_host else 0
    return cfg


class AmdExpJoinConfig(ctypes.Structure):
    _fields_ = [
        ("n_keys", ctypes.c_int32),
        ("n_left_vals", ctypes.c_int32),
        ("n_right_vals", ctypes.c_int32),
        ("ttl_nanos", ctypes.c_uint64),
        ("log2_capacity", ctypes.c_uint32),
        ("log2_rows_cap", ctypes.c_uint32),
        ("log2_out_cap", ctypes.c_uint32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
        ("join_type", ctypes.c_int32),
        ("updating", ctypes.c_int32),
    ]


def make_expjoin_config(ttl_ns, n_left_vals=0, n_right_vals=0,
                        log2_capacity=16, log2_rows_cap=20, log2_out_cap=20,
                        device=0, emit_to_host=True, join_type=0,
                        updating=False):
    cfg = AmdExpJoinConfig()
    cfg.n_keys = 1
    cfg.n_left_vals = n_left_vals
    cfg.n_right_vals = n_right_vals
    cfg.ttl_nanos = ttl_ns
    cfg.log2_capacity = log2_capacity
    cfg.log2_rows_cap = log2_rows_cap
    cfg.log2_out_cap = log2_out_cap
    cfg.device = device
    cfg.emit_to_host = 1 if emit_to_host else 0
    cfg.join_type = join_type
    cfg.updating = 1 if updating else 0
    return cfg


def _cols_to_ptrs(cols):
    """Contiguous int64 copies + a void* array over them."""
    keep = [np.ascontiguousarray(c, dtype=np.int64) for c in cols]
    arr = (ctypes.c_void_p * len(keep))(
        *[c.ctypes.data_as(ctypes.c_void_p).value for c in keep])
    return keep, arr


class SessionOp:
    """One session-window aggregate operator behind the C ABI, mirroring
    SessionAggregatingWindowFunc's ArrowOperator surface
    (crates/arroyo-worker/src/arrow/session_aggregating_window.rs)."""

    def __init__(self, lib, prefix, cfg):
        p = prefix + "session_"
        g = lambda n: getattr(lib, p + n)
        self._fn = {}
        self._fn["_lib_prefix"] = (lib, prefix)
        self._fn["create"] = g("create")
        self._fn["create"].restype = ctypes.c_void_p
        self._fn["create"].argtypes = [ctypes.POINTER(AmdSessionConfig)]
        self._fn["process_batch"] = g("process_batch")
        self._fn["process_batch"].restype = ctypes.c_int
        self._fn["process_batch"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64]
        self._fn["handle_watermark"] = g("handle_watermark")
        self._fn["handle_watermark"].restype = ctypes.c_int
        self._fn["handle_watermark"].argtypes = [
            ctypes.c_void_p, ctypes.c_uint64, ctypes.POINTER(AmdOutBatch)]
        self._fn["checkpoint_drain"] = g("checkpoint_drain")
        self._fn["checkpoint_drain"].restype = ctypes.c_int
        self._fn["checkpoint_drain"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(AmdOutBatch)]
        self._fn["restore"] = g("restore")
        self._fn["restore"].restype = ctypes.c_int
        self._fn["restore"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64]
        self._fn["free_out"] = getattr(lib, prefix + "free_out")
        self._fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
        self._fn["destroy"] = g("destroy")
        self._fn["destroy"].argtypes = [ctypes.c_void_p]
        self._fn["last_error"] = g("last_error")
        self._fn["last_error"].restype = ctypes.c_char_p
        self._fn["last_error"].argtypes = [ctypes.c_void_p]
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(f"{p}create failed: "
                               f"{self._fn['last_error'](None)}")

    def _check(self, rc):
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def process_batch(self, cols):
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        self._check(self._fn["process_batch"](self._h, arr, len(keep),
                                              n_rows))

    def handle_watermark(self, wm):
        out = AmdOutBatch()
        self._check(self._fn["handle_watermark"](self._h, wm,
                                                 ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def checkpoint_drain(self):
        out = AmdOutBatch()
        self._check(self._fn["checkpoint_drain"](self._h, ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def restore(self, cols):
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        self._check(self._fn["restore"](self._h, arr, len(keep), n_rows))

    def _bind_values(self):
        # GPU-only extension (the oracle drains raw rows instead)
        if "drain_values" not in self._fn:
            g = self._fn["_lib_prefix"]
            dv = getattr(g[0], g[1] + "session_drain_values")
            dv.restype = ctypes.c_int
            dv.argtypes = [ctypes.c_void_p, ctypes.POINTER(AmdOutBatch)]
            rv = getattr(g[0], g[1] + "session_restore_values")
            rv.restype = ctypes.c_int
            rv.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p),
                           ctypes.c_int32, ctypes.c_int64]
            self._fn["drain_values"] = dv
            self._fn["restore_values"] = rv

    def drain_values(self):
        self._bind_values()
        out = AmdOutBatch()
        self._check(self._fn["drain_values"](self._h, ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def restore_values(self, cols):
        self._bind_values()
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        self._check(self._fn["restore_values"](self._h, arr, len(keep),
                                               n_rows))

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class ExpJoinOp:
    """One TTL'd (non-windowed) join operator behind the C ABI, mirroring
    JoinWithExpiration's ArrowOperator surface
    (crates/arroyo-worker/src/arrow/join_with_expiration.rs)."""

    LEFT, RIGHT = 0, 1

    def __init__(self, lib, prefix, cfg):
        p = prefix + "expjoin_"
        g = lambda n: getattr(lib, p + n)
        self._fn = {}
        self._fn["create"] = g("create")
        self._fn["create"].restype = ctypes.c_void_p
        self._fn["create"].argtypes = [ctypes.POINTER(AmdExpJoinConfig)]
        self._fn["process_batch"] = g("process_batch")
        self._fn["process_batch"].restype = ctypes.c_int
        self._fn["process_batch"].argtypes = [
            ctypes.c_void_p, ctypes.c_int32,
            ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32, ctypes.c_int64,
            ctypes.POINTER(AmdOutBatch)]
        self._fn["handle_watermark"] = g("handle_watermark")
        self._fn["handle_watermark"].restype = ctypes.c_int
        self._fn["handle_watermark"].argtypes = [ctypes.c_void_p,
                                                 ctypes.c_uint64]
        self._fn["expire"] = g("expire")
        self._fn["expire"].restype = ctypes.c_int
        self._fn["expire"].argtypes = [ctypes.c_void_p]
        self._fn["_lib_prefix"] = (lib, prefix)
        self._fn["checkpoint_drain"] = g("checkpoint_drain")
        self._fn["checkpoint_drain"].restype = ctypes.c_int
        self._fn["checkpoint_drain"].argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.POINTER(AmdOutBatch)]
        self._fn["restore"] = g("restore")
        self._fn["restore"].restype = ctypes.c_int
        self._fn["restore"].argtypes = [
            ctypes.c_void_p, ctypes.c_int32,
            ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32, ctypes.c_int64,
            ctypes.c_int, ctypes.c_uint64]
        self._fn["free_out"] = getattr(lib, prefix + "free_out")
        self._fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
        self._fn["destroy"] = g("destroy")
        self._fn["destroy"].argtypes = [ctypes.c_void_p]
        self._fn["last_error"] = g("last_error")
        self._fn["last_error"].restype = ctypes.c_char_p
        self._fn["last_error"].argtypes = [ctypes.c_void_p]
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(f"{p}create failed: "
                               f"{self._fn['last_error'](None)}")

    def _check(self, rc):
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def process_batch(self, side, cols):
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        out = AmdOutBatch()
        self._check(self._fn["process_batch"](self._h, side, arr, len(keep),
                                              n_rows, ctypes.byref(out)))
        res = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return res

    def handle_watermark(self, wm):
        self._check(self._fn["handle_watermark"](self._h, wm))

    def _bind_device(self):
        # GPU-only extension (the oracle has no device-resident path)
        if "process_batch_device" not in self._fn:
            lib, prefix = self._fn["_lib_prefix"]
            pd = getattr(lib, prefix + "expjoin_process_batch_device")
            pd.restype = ctypes.c_int
            pd.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                           ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
                           ctypes.c_int64]
            co = getattr(lib, prefix + "expjoin_collect")
            co.restype = ctypes.c_int
            co.argtypes = [ctypes.c_void_p, ctypes.POINTER(AmdOutBatch)]
            self._fn["process_batch_device"] = pd
            self._fn["collect"] = co

    def process_batch_device(self, side, dptrs, n_rows):
        """Device-resident ingest: dptrs are raw device addresses of the
        same [key, vals..., ts] columns process_batch takes.  Matches
        accumulate on the device until collect()."""
        self._bind_device()
        arr = (ctypes.c_void_p * len(dptrs))(*dptrs)
        self._check(self._fn["process_batch_device"](
            self._h, side, arr, len(dptrs), n_rows))

    def collect(self):
        """Drain device-accumulated match rows to host numpy columns."""
        self._bind_device()
        out = AmdOutBatch()
        self._check(self._fn["collect"](self._h, ctypes.byref(out)))
        res = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return res

    def expire(self):
        self._check(self._fn["expire"](self._h))

    def checkpoint_drain(self, side):
        out = AmdOutBatch()
        self._check(self._fn["checkpoint_drain"](self._h, side,
                                                 ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def restore(self, side, cols, watermark=None):
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        self._check(self._fn["restore"](
            self._h, side, arr, len(keep), n_rows,
            1 if watermark is not None else 0,
            watermark if watermark is not None else 0))

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


COUNT_DISTINCT = 5
STDDEV = 6
STDDEV_POP = 7
VAR = 8
VAR_POP = 9
BIT_XOR = 10
COVAR_POP = 11
COVAR_SAMP = 12
CORR = 13
REGR_SLOPE = 14
REGR_INTERCEPT = 15
REGR_R2 = 16
REGR_AVGX = 17
REGR_AVGY = 18
REGR_COUNT = 19
REGR_SXX = 20
REGR_SYY = 21
REGR_SXY = 22
BIT_AND = 23
BIT_OR = 24


class AmdUpdatingConfig(ctypes.Structure):
    _fields_ = [
        ("n_keys", ctypes.c_int32),
        ("n_value_cols", ctypes.c_int32),
        ("n_aggs", ctypes.c_int32),
        ("agg_col2", ctypes.c_int32 * 8),
        ("agg_ops", ctypes.c_int32 * 8),
        ("agg_col", ctypes.c_int32 * 8),
        ("log2_capacity", ctypes.c_uint32),
        ("log2_nodes", ctypes.c_uint32),
        ("log2_out_cap", ctypes.c_uint32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
    ]


def make_updagg_config(aggs, n_keys=1, n_value_cols=0, log2_capacity=16,
                       log2_nodes=20, log2_out_cap=20, device=0,
                       emit_to_host=True):
    cfg = AmdUpdatingConfig()
    cfg.n_keys = n_keys
    cfg.n_value_cols = n_value_cols
    cfg.n_aggs = len(aggs)
    for i, spec in enumerate(aggs):
        op, col = spec[0], spec[1]
        cfg.agg_ops[i] = op
        cfg.agg_col[i] = col
        cfg.agg_col2[i] = spec[2] if len(spec) > 2 else -1
    cfg.log2_capacity = log2_capacity
    cfg.log2_nodes = log2_nodes
    cfg.log2_out_cap = log2_out_cap
    cfg.device = device
    cfg.emit_to_host = 1 if emit_to_host else 0
    return cfg


class UpdAggOp:
    """One updating (non-windowed) aggregate operator behind the C ABI,
    mirroring IncrementalAggregatingFunc's ArrowOperator surface
    (crates/arroyo-worker/src/arrow/incremental_aggregator.rs)."""

    def __init__(self, lib, prefix, cfg):
        p = prefix + "updagg_"
        g = lambda n: getattr(lib, p + n)
        self._fn = {}
        self._fn["create"] = g("create")
        self._fn["create"].restype = ctypes.c_void_p
        self._fn["create"].argtypes = [ctypes.POINTER(AmdUpdatingConfig)]
        self._fn["process_batch"] = g("process_batch")
        self._fn["process_batch"].restype = ctypes.c_int
        self._fn["process_batch"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64]
        self._fn["flush"] = g("flush")
        self._fn["flush"].restype = ctypes.c_int
        self._fn["flush"].argtypes = [ctypes.c_void_p,
                                      ctypes.POINTER(AmdOutBatch)]
        self._fn["expire"] = g("expire")
        self._fn["expire"].restype = ctypes.c_int
        self._fn["expire"].argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                       ctypes.POINTER(AmdOutBatch)]
        self._fn["_lib_prefix"] = (lib, prefix)
        self._fn["checkpoint_drain"] = g("checkpoint_drain")
        self._fn["checkpoint_drain"].restype = ctypes.c_int
        self._fn["checkpoint_drain"].argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.POINTER(AmdOutBatch)]
        self._fn["restore"] = g("restore")
        self._fn["restore"].restype = ctypes.c_int
        self._fn["restore"].argtypes = [
            ctypes.c_void_p, ctypes.c_int32,
            ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32, ctypes.c_int64]
        self._fn["free_out"] = getattr(lib, prefix + "free_out")
        self._fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
        self._fn["destroy"] = g("destroy")
        self._fn["destroy"].argtypes = [ctypes.c_void_p]
        self._fn["last_error"] = g("last_error")
        self._fn["last_error"].restype = ctypes.c_char_p
        self._fn["last_error"].argtypes = [ctypes.c_void_p]
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(
                f"{p}create failed: "
                f"{self._fn['last_error'](None).decode()}")

    def _check(self, rc):
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def process_batch(self, cols):
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        self._check(self._fn["process_batch"](self._h, arr, len(keep),
                                              n_rows))

    def process_batch_device(self, dptrs, n_rows):
        """Device-resident ingest (GPU-only extension): dptrs are raw
        device addresses of the [keys..., vals..., is_retract] columns."""
        if "process_batch_device" not in self._fn:
            lib, prefix = self._fn["_lib_prefix"]
            pd = getattr(lib, prefix + "updagg_process_batch_device")
            pd.restype = ctypes.c_int
            pd.argtypes = [ctypes.c_void_p,
                           ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
                           ctypes.c_int64]
            self._fn["process_batch_device"] = pd
        arr = (ctypes.c_void_p * len(dptrs))(*dptrs)
        self._check(self._fn["process_batch_device"](self._h, arr,
                                                     len(dptrs), n_rows))

    def flush(self):
        out = AmdOutBatch()
        self._check(self._fn["flush"](self._h, ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def expire(self, idle_flushes):
        out = AmdOutBatch()
        self._check(self._fn["expire"](self._h, idle_flushes,
                                       ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def checkpoint_drain(self, which):
        out = AmdOutBatch()
        self._check(self._fn["checkpoint_drain"](self._h, which,
                                                 ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def restore(self, which, cols):
        keep, arr = _cols_to_ptrs(cols)
        n_rows = len(keep[0]) if keep else 0
        self._check(self._fn["restore"](self._h, which, arr, len(keep),
                                        n_rows))

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class AmdWindowFnConfig(ctypes.Structure):
    _fields_ = [
        ("n_cols", ctypes.c_int32),
        ("part_col", ctypes.c_int32),
        ("n_order", ctypes.c_int32),
        ("order_col", ctypes.c_int32 * 2),
        ("order_desc", ctypes.c_int32 * 2),
        ("limit", ctypes.c_int64),
        ("log2_rows_cap", ctypes.c_uint32),
        ("instants", ctypes.c_uint32),
        ("log2_out_cap", ctypes.c_uint32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
    ]


def make_windowfn_config(n_cols, part_col, order, limit=0, log2_rows_cap=15,
                         instants=128, log2_out_cap=20, device=0,
                         emit_to_host=True):
    """order: list of (col, desc) pairs, max 2."""
    cfg = AmdWindowFnConfig()
    cfg.n_cols = n_cols
    cfg.part_col = part_col
    cfg.n_order = len(order)
    for i, (col, desc) in enumerate(order):
        cfg.order_col[i] = col
        cfg.order_desc[i] = 1 if desc else 0
    cfg.limit = limit
    cfg.log2_rows_cap = log2_rows_cap
    cfg.instants = instants
    cfg.log2_out_cap = log2_out_cap
    cfg.device = device
    cfg.emit_to_host = 1 if emit_to_host else 0
    return cfg


class WindowFnOp:
    """One ROW_NUMBER window-function operator behind the C ABI, mirroring
    WindowFunctionOperator's ArrowOperator surface
    (crates/arroyo-worker/src/arrow/window_fn.rs)."""

    def __init__(self, lib, prefix, cfg):
        p = prefix + "windowfn_"
        g = lambda n: getattr(lib, p + n)
        self._fn = {}
        self._fn["create"] = g("create")
        self._fn["create"].restype = ctypes.c_void_p
        self._fn["create"].argtypes = [ctypes.POINTER(AmdWindowFnConfig)]
        self._fn["process_batch"] = g("process_batch")
        self._fn["process_batch"].restype = ctypes.c_int
        self._fn["process_batch"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64]
        self._fn["handle_watermark"] = g("handle_watermark")
        self._fn["handle_watermark"].restype = ctypes.c_int
        self._fn["handle_watermark"].argtypes = [
            ctypes.c_void_p, ctypes.c_uint64, ctypes.POINTER(AmdOutBatch)]
        self._fn["restore"] = g("restore")
        self._fn["restore"].restype = ctypes.c_int
        self._fn["restore"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64]
        self._fn["checkpoint_drain"] = g("checkpoint_drain")
        self._fn["checkpoint_drain"].restype = ctypes.c_int
        self._fn["checkpoint_drain"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(AmdOutBatch)]
        self._fn["free_out"] = getattr(lib, prefix + "free_out")
        self._fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
        self._fn["destroy"] = g("destroy")
        self._fn["destroy"].argtypes = [ctypes.c_void_p]
        self._fn["last_error"] = g("last_error")
        self._fn["last_error"].restype = ctypes.c_char_p
        self._fn["last_error"].argtypes = [ctypes.c_void_p]
        self._fn["_lib_prefix"] = (lib, prefix)
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(
                f"{p}create failed: "
                f"{self._fn['last_error'](None).decode()}")

    def _check(self, rc):
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def process_batch(self, cols):
        keep, arr = _cols_to_ptrs(cols)
        self._check(self._fn["process_batch"](
            self._h, arr, len(keep), len(keep[0]) if keep else 0))

    def process_batch_device(self, dptrs, n_rows):
        """Device-resident ingest (GPU-only extension): dptrs are raw
        device addresses of the same columns process_batch takes."""
        if "process_batch_device" not in self._fn:
            lib, prefix = self._fn["_lib_prefix"]
            pd = getattr(lib, prefix + "windowfn_process_batch_device")
            pd.restype = ctypes.c_int
            pd.argtypes = [ctypes.c_void_p,
                           ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
                           ctypes.c_int64]
            self._fn["process_batch_device"] = pd
        arr = (ctypes.c_void_p * len(dptrs))(*dptrs)
        self._check(self._fn["process_batch_device"](self._h, arr,
                                                     len(dptrs), n_rows))

    def restore(self, cols):
        keep, arr = _cols_to_ptrs(cols)
        self._check(self._fn["restore"](
            self._h, arr, len(keep), len(keep[0]) if keep else 0))

    def handle_watermark(self, wm):
        out = AmdOutBatch()
        self._check(self._fn["handle_watermark"](self._h, wm,
                                                 ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def checkpoint_drain(self):
        out = AmdOutBatch()
        self._check(self._fn["checkpoint_drain"](self._h, ctypes.byref(out)))
        cols = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return cols

    def restore(self, cols):
        if cols and len(cols[0]):
            self.process_batch(cols)

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


MOP_CONST, MOP_ADD, MOP_SUB, MOP_MUL, MOP_DIV, MOP_MOD = range(6)
MOP_EQ, MOP_NE, MOP_LT, MOP_LE, MOP_GT, MOP_GE = range(6, 12)
MOP_AND, MOP_OR, MOP_NOT, MOP_I2F, MOP_F2I = range(12, 17)
MOP_FADD, MOP_FSUB, MOP_FMUL, MOP_FDIV = range(17, 21)


class AmdMapInstr(ctypes.Structure):
    _fields_ = [("op", ctypes.c_int32), ("a", ctypes.c_int32),
                ("b", ctypes.c_int32), ("dst", ctypes.c_int32),
                ("imm", ctypes.c_int64)]


class AmdMapConfig(ctypes.Structure):
    _fields_ = [
        ("n_in_cols", ctypes.c_int32),
        ("n_prog", ctypes.c_int32),
        ("prog", AmdMapInstr * 64),
        ("n_out", ctypes.c_int32),
        ("out_reg", ctypes.c_int32 * 16),
        ("out_is_f64", ctypes.c_int32 * 16),
        ("filter_reg", ctypes.c_int32),
        ("device", ctypes.c_int32),
        ("emit_to_host", ctypes.c_int32),
    ]


def make_map_config(n_in_cols, prog, out_reg, out_is_f64=None,
                    filter_reg=-1, device=0):
    """prog: list of (op, a, b, dst[, imm]) tuples.  imm may be a float for
    MOP_CONST feeding f64 ops (stored as the f64 bit pattern)."""
    import struct
    cfg = AmdMapConfig()
    cfg.n_in_cols = n_in_cols
    cfg.n_prog = len(prog)
    for i, ins in enumerate(prog):
        op, a, b, dst = ins[:4]
        imm = ins[4] if len(ins) > 4 else 0
        if isinstance(imm, float):
            imm = struct.unpack("<q", struct.pack("<d", imm))[0]
        cfg.prog[i].op = op
        cfg.prog[i].a = a
        cfg.prog[i].b = b
        cfg.prog[i].dst = dst
        cfg.prog[i].imm = imm
    cfg.n_out = len(out_reg)
    for i, r in enumerate(out_reg):
        cfg.out_reg[i] = r
        cfg.out_is_f64[i] = 1 if (out_is_f64 and out_is_f64[i]) else 0
    cfg.filter_reg = filter_reg
    cfg.device = device
    cfg.emit_to_host = 1
    return cfg


class MapOp:
    """One stateless map/filter/projection operator behind the C ABI,
    mirroring the expression operators' ArrowOperator surface
    (crates/arroyo-worker/src/arrow/mod.rs:48-243)."""

    def __init__(self, lib, prefix, cfg):
        p = prefix + "map_"
        g = lambda n: getattr(lib, p + n)
        self._fn = {}
        self._fn["create"] = g("create")
        self._fn["create"].restype = ctypes.c_void_p
        self._fn["create"].argtypes = [ctypes.POINTER(AmdMapConfig)]
        self._fn["process_batch"] = g("process_batch")
        self._fn["process_batch"].restype = ctypes.c_int
        self._fn["process_batch"].argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64, ctypes.POINTER(AmdOutBatch)]
        self._fn["free_out"] = getattr(lib, prefix + "free_out")
        self._fn["free_out"].argtypes = [ctypes.POINTER(AmdOutBatch)]
        self._fn["destroy"] = g("destroy")
        self._fn["destroy"].argtypes = [ctypes.c_void_p]
        self._fn["last_error"] = g("last_error")
        self._fn["last_error"].restype = ctypes.c_char_p
        self._fn["last_error"].argtypes = [ctypes.c_void_p]
        self._fn["_lib_prefix"] = (lib, prefix)
        self.cfg = cfg
        self._h = self._fn["create"](ctypes.byref(cfg))
        if not self._h:
            raise RuntimeError(
                f"{p}create failed: "
                f"{self._fn['last_error'](None).decode()}")

    def process_batch_device(self, dptrs, n_rows):
        """Device-resident map/filter (GPU-only extension): input device
        addresses in, (device output addresses, surviving row count) out.
        The returned addresses are owned by the operator and valid until
        the next process_batch* call."""
        if "process_batch_device" not in self._fn:
            lib, prefix = self._fn["_lib_prefix"]
            pd = getattr(lib, prefix + "map_process_batch_device")
            pd.restype = ctypes.c_int
            pd.argtypes = [ctypes.c_void_p,
                           ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
                           ctypes.c_int64, ctypes.POINTER(ctypes.c_void_p),
                           ctypes.POINTER(ctypes.c_int64)]
            self._fn["process_batch_device"] = pd
        arr = (ctypes.c_void_p * len(dptrs))(*dptrs)
        outp = (ctypes.c_void_p * self.cfg.n_out)()
        n_out = ctypes.c_int64(0)
        rc = self._fn["process_batch_device"](self._h, arr, len(dptrs),
                                              n_rows, outp,
                                              ctypes.byref(n_out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        return [outp[i] for i in range(self.cfg.n_out)], int(n_out.value)

    def process_batch(self, cols):
        keep, arr = _cols_to_ptrs(cols)
        out = AmdOutBatch()
        rc = self._fn["process_batch"](self._h, arr, len(keep),
                                       len(keep[0]) if keep else 0,
                                       ctypes.byref(out))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        res = _out_to_numpy(out)
        self._fn["free_out"](ctypes.byref(out))
        return res

    def close(self):
        if self._h:
            self._fn["destroy"](self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
