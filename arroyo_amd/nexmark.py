"""Synthetic nexmark-shaped bid stream generator.

Vectorized restatement of the reference generator's bid shape
(ArroyoSystems/arroyo, crates/arroyo-connectors/src/nexmark/operator.rs):
  - event proportions person:auction:bid = 1:3:46 (NexmarkConfig :392-394),
    so bid i maps to event id ~ i*50/46 and the number of auctions created
    by event e is ~ 3e/50 (next_base0_auction_id semantics);
  - 50% of bids go to the current "hot" auction = last auction id rounded
    down to a multiple of 100 (next_bid :728-733, hot_auction_ratio=2 :398);
  - the rest are uniform over the ~100 in-flight auctions
    (num_inflight_auctions :367,:402);
  - event time advances uniformly at `events_per_sec` (event-time rate).

Simplifications vs the reference (stated per SURVEY.md §8d): the per-event
SmallRng stream is replaced by a seeded PCG64, and the auction-id adjustment
for out-of-order delays is dropped (we generate in-order event time).
"""
import numpy as np

NS = 10**9
FIRST_AUCTION_ID = 1000
HOT_RATIO = 100
IN_FLIGHT = 100
BID_PROPORTION = 46
TOTAL_PROPORTION = 50
AUCTION_PROPORTION = 3


def bids(n, events_per_sec=1_000_000, t0=1_600_000_000 * NS, seed=42,
         with_price=False):
    """Returns columns (auction i64, [price i64,] _timestamp i64 ns)."""
    rng = np.random.default_rng(seed)
    i = np.arange(n, dtype=np.int64)
    event_id = (i * TOTAL_PROPORTION) // BID_PROPORTION
    last_auction = (event_id * AUCTION_PROPORTION) // TOTAL_PROPORTION
    hot = rng.random(n) < 0.5
    uniform = last_auction - rng.integers(0, IN_FLIGHT + 1, size=n)
    auction = np.where(hot, (last_auction // HOT_RATIO) * HOT_RATIO,
                       np.maximum(uniform, 0)) + FIRST_AUCTION_ID
    ts = t0 + (event_id * NS) // events_per_sec
    cols = [auction.astype(np.int64)]
    if with_price:
        price = rng.integers(1, 10_000, size=n, dtype=np.int64)
        cols.append(price)
    cols.append(ts.astype(np.int64))
    return cols
