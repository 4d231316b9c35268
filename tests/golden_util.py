"""Helpers for comparing operator outputs with the reference's golden
vectors (tests/golden/, extracted by oracle/gen_golden.py)."""
import json
import os
from datetime import datetime as dt, timezone


GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")
NS = 10**9


def load_inputs():
    with open(os.path.join(GOLDEN, "inputs.json")) as f:
        return json.load(f)


def load_golden(name):
    with open(os.path.join(GOLDEN, f"{name}.golden.json")) as f:
        return json.load(f)


def fmt_ts(ns):
    s, frac = divmod(int(ns), NS)
    t = dt.fromtimestamp(s, tz=timezone.utc)
    base = t.strftime("%Y-%m-%dT%H:%M:%S")
    if frac:
        digits = "%09d" % frac
        # the reference prints fractional seconds at milli/micro/nano
        # precision (chrono %.f): strip trailing zeros in groups of three
        while digits.endswith("000"):
            digits = digits[:-3]
        base += "." + digits
    return base


def canon(rows):
    """Sorted canonical JSON lines for order-insensitive comparison
    (matches the reference smoke tests' sorted line-exact check,
    arroyo-sql-testing/src/smoke_tests.rs:619-692)."""
    return sorted(json.dumps(r, sort_keys=True) for r in rows)


def assert_rows_match(got_rows, golden_rows):
    g, w = canon(got_rows), canon(golden_rows)
    assert len(g) == len(w), f"row count {len(g)} != golden {len(w)}"
    for a, b in zip(g, w):
        assert a == b, f"mismatch:\n got  {a}\n want {b}"
