"""Loader for the MI355X product library (libarroyo_amd.so).

The HIP extension is built in-tree by __graft_entry__.build() (or lazily here
with hipcc if the source is newer).  There is NO CPU fallback: on a machine
with a GPU, operator creation either runs the HIP path or raises.
"""
import ctypes
import os
import subprocess

import numpy as np

from arroyo_amd.cabi import WindowOp

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libarroyo_amd.so")
_SRCS = [os.path.join(_DIR, "csrc", f)
         for f in ("arroyo_amd.hip", "session.hip", "expjoin.hip",
                   "updagg.hip", "windowfn.hip", "mapop.hip")]

_lib = None

HIPCC_CMD = ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             "-shared", "-fvisibility=hidden", "-parallel-jobs=6",
             "-o", _SO] + _SRCS


def build(force=False):
    if force or not os.path.exists(_SO) or \
            os.path.getmtime(_SO) < max(os.path.getmtime(s) for s in _SRCS):
        proc = subprocess.run(HIPCC_CMD, cwd=_DIR,
                              capture_output=True, text=True)
        if proc.returncode != 0:
            raise RuntimeError(
                f"hipcc failed (rc {proc.returncode}):\n{proc.stderr}")
    return _SO


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            build()
        _lib = ctypes.CDLL(_SO)
        _lib.arroyo_amd_perf.restype = ctypes.c_int
        _lib.arroyo_amd_perf.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_double),
            ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int64),
            ctypes.POINTER(ctypes.c_int64)]
        _lib.arroyo_amd_process_batch_device.restype = ctypes.c_int
        _lib.arroyo_amd_process_batch_device.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64, ctypes.c_uint64]
        _lib.arroyo_amd_restore.restype = ctypes.c_int
        _lib.arroyo_amd_restore.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64, ctypes.c_int, ctypes.c_uint64]
        _lib.arroyo_amd_process_batches_device.restype = ctypes.c_int
        _lib.arroyo_amd_process_batches_device.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.c_int64, ctypes.c_int32, ctypes.c_int32, ctypes.c_uint64,
            ctypes.c_uint64]
        _lib.arroyo_amd_stream_gbps.restype = ctypes.c_double
        _lib.arroyo_amd_stream_gbps.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64, ctypes.c_int]
        _lib.arroyo_amd_sync.restype = ctypes.c_int
        _lib.arroyo_amd_sync.argtypes = [ctypes.c_void_p]
        _lib.arroyo_amd_set_filter_watermark.restype = ctypes.c_int
        _lib.arroyo_amd_set_filter_watermark.argtypes = [
            ctypes.c_void_p, ctypes.c_uint64]
        _lib.arroyo_amd_mark_epoch.restype = ctypes.c_int
        _lib.arroyo_amd_mark_epoch.argtypes = [ctypes.c_void_p]
        _lib.arroyo_amd_partition.restype = ctypes.c_int
        _lib.arroyo_amd_partition.argtypes = [ctypes.c_void_p] * 3 + [
            ctypes.c_int64, ctypes.c_uint32] + [ctypes.c_void_p] * 3 + [
            ctypes.POINTER(ctypes.c_uint64)]
    return _lib


class GpuWindowOp(WindowOp):
    """Product-path window operator (HIP, gfx950)."""

    def __init__(self, cfg):
        super().__init__(lib(), "arroyo_amd_", cfg)

    def process_batch_device(self, dev_ptrs, n_rows, ts_offset=0):
        """dev_ptrs: list of device pointers (ints) to i64 columns."""
        arr = (ctypes.c_void_p * len(dev_ptrs))(*dev_ptrs)
        rc = lib().arroyo_amd_process_batch_device(
            self._h, arr, len(dev_ptrs), n_rows, ts_offset)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def process_batches_device(self, dev_ptrs, n_rows, reps, contiguous=True,
                               ts_offset0=0, ts_step=0):
        arr = (ctypes.c_void_p * len(dev_ptrs))(*dev_ptrs)
        rc = lib().arroyo_amd_process_batches_device(
            self._h, arr, len(dev_ptrs), n_rows, reps,
            1 if contiguous else 0, ts_offset0, ts_step)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def restore(self, cols, watermark=None):
        cols = [np.ascontiguousarray(c, dtype=np.int64) for c in cols]
        n_rows = len(cols[0]) if cols else 0
        arr = (ctypes.c_void_p * len(cols))(
            *[c.ctypes.data_as(ctypes.c_void_p).value for c in cols])
        rc = lib().arroyo_amd_restore(
            self._h, arr, len(cols), n_rows,
            1 if watermark is not None else 0,
            watermark if watermark is not None else 0)
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())

    def perf(self):
        ms = ctypes.c_double()
        rows = ctypes.c_int64()
        launches = ctypes.c_int64()
        emitted = ctypes.c_int64()
        rc = lib().arroyo_amd_perf(self._h, ctypes.byref(ms),
                                   ctypes.byref(rows), ctypes.byref(launches),
                                   ctypes.byref(emitted))
        if rc != 0:
            raise RuntimeError(self._fn["last_error"](self._h).decode())
        return {"update_ms": ms.value, "rows": rows.value,
                "launches": launches.value,
                "emitted_device_rows": emitted.value}


def make_op(cfg):
    return GpuWindowOp(cfg)


def make_join_op(cfg):
    from arroyo_amd.cabi import JoinOp
    return JoinOp(lib(), "arroyo_amd_", cfg)


def partition_device(d_keys, d_vals, d_ts, n, n_parts, d_out_keys, d_out_vals,
                     d_out_ts):
    counts = (ctypes.c_uint64 * n_parts)()
    rc = lib().arroyo_amd_partition(
        d_keys, d_vals, d_ts, n, n_parts, d_out_keys, d_out_vals, d_out_ts,
        counts)
    if rc != 0:
        raise RuntimeError("arroyo_amd_partition failed")
    return list(counts)


def make_session_op(cfg):
    from arroyo_amd.cabi import SessionOp
    return SessionOp(lib(), "arroyo_amd_", cfg)


def make_expjoin_op(cfg):
    from arroyo_amd.cabi import ExpJoinOp
    return ExpJoinOp(lib(), "arroyo_amd_", cfg)


def make_updagg_op(cfg):
    from arroyo_amd.cabi import UpdAggOp
    return UpdAggOp(lib(), "arroyo_amd_", cfg)


def make_windowfn_op(cfg):
    from arroyo_amd.cabi import WindowFnOp
    return WindowFnOp(lib(), "arroyo_amd_", cfg)


def make_map_op(cfg):
    from arroyo_amd.cabi import MapOp
    return MapOp(lib(), "arroyo_amd_", cfg)
