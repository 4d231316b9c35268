"""Loader for the CPU oracle library (test infrastructure + CPU baseline
only; see oracle/arroyo_oracle.c header).  Builds liboracle.so with gcc on
first use if absent."""
import ctypes
import os
import subprocess

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")
_SRC = os.path.join(_DIR, "arroyo_oracle.c")

_lib = None


def build(force=False):
    if force or not os.path.exists(_SO) or \
            os.path.getmtime(_SO) < os.path.getmtime(_SRC):
        subprocess.run(
            ["gcc", "-O2", "-g", "-shared", "-fPIC", "-fvisibility=hidden",
             "-o", _SO, _SRC, "-lm"],
            check=True, cwd=_DIR)
    return _SO


def lib():
    global _lib
    if _lib is None:
        _lib = ctypes.CDLL(build())
    return _lib


def make_op(cfg):
    from arroyo_amd.cabi import WindowOp
    return WindowOp(lib(), "oracle_", cfg)


def make_join_op(cfg):
    from arroyo_amd.cabi import JoinOp
    return JoinOp(lib(), "oracle_", cfg)


def make_session_op(cfg):
    from arroyo_amd.cabi import SessionOp
    return SessionOp(lib(), "oracle_", cfg)


def make_expjoin_op(cfg):
    from arroyo_amd.cabi import ExpJoinOp
    return ExpJoinOp(lib(), "oracle_", cfg)


def make_updagg_op(cfg):
    from arroyo_amd.cabi import UpdAggOp
    return UpdAggOp(lib(), "oracle_", cfg)


def make_windowfn_op(cfg):
    from arroyo_amd.cabi import WindowFnOp
    return WindowFnOp(lib(), "oracle_", cfg)


def make_map_op(cfg):
    from arroyo_amd.cabi import MapOp
    return MapOp(lib(), "oracle_", cfg)
