"""Canonical feature registry.

Replaces the reference's implicit feature registry — the SQL `join_statement`
assembled by DESCRIBE-ing the main table and every VIEW (reference
create_database.py:193-258) — with an explicit, ordered registry of the same
108 columns, in the same order, with the value ranges recorded in the
reference `norm_params` artifact.

Column groups, in canonical order (matching the reference main table DDL at
create_database.py:29-67 plus the VIEW columns appended at :240-241):

  7  bid_{0..6}_size      order-book bid sizes
  6  bid_{1..6}           bid price distance from best bid
  7  ask_{0..6}_size      order-book ask sizes
  6  ask_{1..6}           ask price distance from best ask
  2  bids_ord_WA, asks_ord_WA
  1  vol_imbalance
  1  delta
  1  micro_price
  1  spread
  1  session_start
  8  day_1..4, week_1..4  one-hot day-of-week / week-of-month
  1  VIX
  6  1_open 2_high 3_low 4_close 5_volume wick_prct
 12  COT fields
 39  13 events x {Actual, Prev_actual_diff, Forc_actual_diff}
  2  upper_BB_dist lower_BB_dist
  2  vol_MA6 vol_MA20
  1  price_MA20
  1  delta_MA12
  1  stoch
  1  ATR
  1  price_change
 ---
108
"""
from typing import List, Tuple

from .config import (ASK_LEVELS, BID_LEVELS, DELTA_MA_PERIODS,
                     EVENT_LIST_REPL, EVENT_VALUES, PRICE_MA_PERIODS,
                     VOLUME_MA_PERIODS)
from .feature_ranges import REFERENCE_RANGES


def _build_names() -> List[str]:
    names: List[str] = []
    names += [f"sd.bid_{i}_size" for i in range(BID_LEVELS)]
    names += [f"sd.bid_{i}" for i in range(1, BID_LEVELS)]
    names += [f"sd.ask_{i}_size" for i in range(ASK_LEVELS)]
    names += [f"sd.ask_{i}" for i in range(1, ASK_LEVELS)]
    names += ["sd.bids_ord_WA", "sd.asks_ord_WA", "sd.vol_imbalance",
              "sd.delta", "sd.micro_price", "sd.spread", "sd.session_start"]
    names += [f"sd.day_{i}" for i in range(1, 5)]
    names += [f"sd.week_{i}" for i in range(1, 5)]
    names += ["sd.VIX"]
    names += ["sd.1_open", "sd.2_high", "sd.3_low", "sd.4_close",
              "sd.5_volume", "sd.wick_prct"]
    for side in ("Asset", "Leveraged"):
        for direction in ("long", "short"):
            names += [f"sd.{side}_{direction}_pos",
                      f"sd.{side}_{direction}_pos_change",
                      f"sd.{side}_{direction}_open_int"]
    for event in EVENT_LIST_REPL:
        for value in EVENT_VALUES:
            names.append(f"sd.{event}_{value}")
    names += ["bb.upper_BB_dist", "bb.lower_BB_dist"]
    names += [f"vol.vol_MA{p}" for p in VOLUME_MA_PERIODS]
    names += [f"p.price_MA{p}" for p in PRICE_MA_PERIODS]
    names += [f"d.delta_MA{p}" for p in DELTA_MA_PERIODS]
    names += ["so.stoch", "ATR.ATR", "pc.price_change"]
    return names


FEATURE_NAMES: List[str] = _build_names()
N_FEATURES: int = len(FEATURE_NAMES)
TARGET_NAMES: List[str] = ["up1", "up2", "down1", "down2"]

assert FEATURE_NAMES == list(REFERENCE_RANGES.keys()), (
    "feature registry drifted from the reference norm_params column order")


def feature_range(name: str) -> Tuple[float, float]:
    return REFERENCE_RANGES[name]


def index_of(name: str) -> int:
    return FEATURE_NAMES.index(name)


# Frequently-used column indices.
IDX_CLOSE = FEATURE_NAMES.index("sd.4_close")
IDX_OPEN = FEATURE_NAMES.index("sd.1_open")
IDX_HIGH = FEATURE_NAMES.index("sd.2_high")
IDX_LOW = FEATURE_NAMES.index("sd.3_low")
IDX_VOLUME = FEATURE_NAMES.index("sd.5_volume")
IDX_ATR = FEATURE_NAMES.index("ATR.ATR")

# Order-book size columns share one global MIN/MAX per side within a chunk
# (reference sql_pytorch_dataloader.py:119-144).
BID_SIZE_IDX = [FEATURE_NAMES.index(f"sd.bid_{i}_size") for i in range(BID_LEVELS)]
ASK_SIZE_IDX = [FEATURE_NAMES.index(f"sd.ask_{i}_size") for i in range(ASK_LEVELS)]
