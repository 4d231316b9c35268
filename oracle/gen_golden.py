#!/usr/bin/env python3
"""Generate golden parity fixtures under tests/golden/ from the reference
checkout at /root/reference (ArroyoSystems/arroyo).

Run in the build container (where /root/reference is mounted); the committed
outputs travel to GPU boxes where the reference does not exist.  The fixtures
are the reference's own end-to-end test vectors for the windowed-aggregate
operator (crates/arroyo-sql-testing: inputs/ + golden_outputs/), re-encoded as
columnar JSON with integer nanosecond timestamps (the wire shape our C-ABI
takes).  Queries covered and what they pin:

  sliding_window_end           hop(2s,10s) COUNT/MIN/MAX, unkeyed   (impulse)
  hourly_by_event_type         tumble(1h) COUNT GROUP BY event_type (cars)
  tight_watermark              tumble(1h) COUNT, watermark = ts      (cars)
  most_active_driver_last_hour hop(1min,1h) COUNT GROUP BY driver_id (cars)
                               + top-1-per-window reduction done in the test

String group keys (event_type) are dictionary-encoded to i64 ids here; the
mapping is stored in the fixture so tests can map back before comparing with
the golden output.
"""
import json
import os
from datetime import datetime as dt, timezone

REF = "/root/reference/crates/arroyo-sql-testing"
OUT = os.environ.get(
    "GOLDEN_OUT",
    os.path.join(os.path.dirname(__file__), "..", "tests", "golden"))

NS = 10**9


def parse_ns(s):
    s = s.replace("Z", "+00:00")
    if "+" not in s[10:]:
        s += "+00:00"
    t = dt.fromisoformat(s)
    epoch = dt(1970, 1, 1, tzinfo=timezone.utc)
    d = t - epoch
    return (d.days * 86400 + d.seconds) * NS + d.microseconds * 1000


def load_rows(path):
    return [json.loads(line) for line in open(path) if line.strip()]


def main():
    os.makedirs(OUT, exist_ok=True)

    impulse = load_rows(f"{REF}/inputs/impulse.json")
    cars = load_rows(f"{REF}/inputs/cars.json")

    event_types = sorted({r["event_type"] for r in cars})
    etype_id = {e: i for i, e in enumerate(event_types)}

    fixtures = {
        "impulse": {
            "ts": [parse_ns(r["timestamp"]) for r in impulse],
            "counter": [r["counter"] for r in impulse],
        },
        "cars": {
            "ts": [parse_ns(r["timestamp"]) for r in cars],
            "driver_id": [r["driver_id"] for r in cars],
            "event_type_id": [etype_id[r["event_type"]] for r in cars],
            "event_type_dict": event_types,
        },
    }

    # Debezium updating-aggregate fixture: aggregate_updates.json is a
    # debezium_json stream (ops c/u/d over orders with pk=id) feeding
    # debezium_agg.sql's non-windowed GROUP BY (IncrementalAggregatingFunc).
    # Encode to columns: op (0 append / 1 retract), product id, customer id,
    # quantity -- an update becomes retract(before)+append(after).
    agg_upd = load_rows(f"{REF}/inputs/aggregate_updates.json")
    products, customers = {}, {}

    def pid(name):
        return products.setdefault(name, len(products))

    def cid(name):
        return customers.setdefault(name, len(customers))

    ops, prods, custs, qtys = [], [], [], []

    def emit(rec, retract):
        ops.append(1 if retract else 0)
        prods.append(pid(rec["product_name"]))
        custs.append(cid(rec["customer_name"]))
        qtys.append(rec["quantity"])

    for r in agg_upd:
        if r["op"] == "c":
            emit(r["after"], False)
        elif r["op"] == "u":
            emit(r["before"], True)
            emit(r["after"], False)
        elif r["op"] == "d":
            emit(r["before"], True)
    fixtures["aggregate_updates"] = {
        "op": ops, "product": prods, "customer": custs, "quantity": qtys,
        "product_dict": [k for k, _ in
                         sorted(products.items(), key=lambda kv: kv[1])],
    }

    for name in ("sliding_window_end", "hourly_by_event_type",
                 "tight_watermark", "most_active_driver_last_hour",
                 "windowed_inner_join", "session_window",
                 "global_session_window", "updating_inner_join",
                 "debezium_agg", "filter_updating_aggregates",
                 "aggregates", "grouped_aggregates", "every_aggregate",
                 "month_loose_watermark", "reinvoke_window_function",
                 "active_drivers", "windowed_outer_join",
                 "updating_left_join", "updating_right_join",
                 "updating_full_join", "offset_impulse_join"):
        rows = load_rows(f"{REF}/golden_outputs/{name}.json")
        with open(f"{OUT}/{name}.golden.json", "w") as f:
            json.dump(rows, f)

    with open(f"{OUT}/inputs.json", "w") as f:
        json.dump(fixtures, f)

    print("wrote fixtures to", os.path.abspath(OUT))


if __name__ == "__main__":
    main()
