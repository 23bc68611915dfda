"""Stream assembler — the spark_consumer.py equivalent.

Consumes the five raw topics (deep, volume, vix, cot, ind) from the
in-process bus and rebuilds the joined 108-feature rows the reference
produced with PySpark Structured Streaming + MariaDB:

- per-bar join keyed by the 5-minute-floored timestamp (the reference's
  interval stream-stream joins with 3-minute tolerance,
  spark_consumer.py:434-477); vix joins per bar, cot forward-fills between
  weekly reports, ind applies only at its event bar (zeros otherwise, like
  the reference table's IFNULL(...,0) fetch, sql_pytorch_dataloader.py:229);
- microstructure features from the raw book snapshot: size-weighted order
  distances (spark_consumer.py:320-340), volume imbalance (:342-347), delta
  (:349-353), micro-price (:355-364), spread (:366-368), per-level price
  distances from best (:370-400), wick % (:187-193), day/week one-hots and
  session_start (:402-432);
- the windowed "SQL VIEW" features the reference computed in MariaDB
  (create_database.py:76-164): vol/price/delta MAs, Bollinger distances,
  stochastic oscillator, ATR, price_change — identical formulas via the
  shared rolling helpers of the generator;
- a `predict_timestamp` signal per assembled row (spark_consumer.py:490-502).

Threading contract: handlers run synchronously in the publisher's thread
(the in-process bus delivers on publish). Ingestion is single-writer;
call `table()` after the feed drains (or under external synchronization)
— matching the reference, where the Spark job owns all mutation and the
dataloader reads the warehouse afterwards.

`table()` returns the assembled (n, 108) float32 tensor in canonical
registry order — bit-comparable (to float tolerance) with the
`SyntheticMarket` table the raw feeds were derived from, which is exactly
the round-trip `tests/test_stream_assembler.py` checks.
"""
import math
from typing import Dict, List, Optional

import torch

from ..config import (ASK_LEVELS, BID_LEVELS, EVENT_LIST_REPL, EVENT_VALUES)
from ..features import FEATURE_NAMES, N_FEATURES
from ..data.generator import (_rolling_max, _rolling_mean, _rolling_min,
                              _rolling_std)
from .bus import MessageBus

BARS_PER_DAY = 78


class StreamAssembler:
    def __init__(self, bus: MessageBus, freq: float = 300.0, t0: float = 0.0,
                 emit_signal: bool = True):
        self.bus = bus
        self.freq = freq
        self.t0 = t0
        self.emit_signal = emit_signal
        self._col = {name: i for i, name in enumerate(FEATURE_NAMES)}
        self._rows: Dict[int, List[float]] = {}   # bar index -> row values
        self._order: List[int] = []
        self._last_vix: Optional[float] = None
        self._last_cot: Optional[dict] = None
        bus.topic("deep").subscribe(self._on_deep)
        bus.topic("volume").subscribe(self._on_volume)
        bus.topic("vix").subscribe(self._on_vix)
        bus.topic("cot").subscribe(self._on_cot)
        bus.topic("ind").subscribe(self._on_ind)

    # ------------------------------------------------------------------ #

    def _bar(self, ts: float) -> int:
        """5-minute floor of the message timestamp -> bar index
        (the reference's window(col('Timestamp'), '5 minutes') floor)."""
        return int(math.floor((ts - self.t0) / self.freq + 1e-9))

    def _row(self, ts: float) -> List[float]:
        i = self._bar(ts)
        if i not in self._rows:
            self._rows[i] = [0.0] * N_FEATURES
            self._order.append(i)
        return self._rows[i]

    def _set(self, row: List[float], name: str, v: float) -> None:
        row[self._col[name]] = float(v)

    # ------------------------------ handlers --------------------------- #

    def _on_deep(self, msg: dict) -> None:
        row = self._row(msg["Timestamp"])
        s = self._set
        best_bid, best_ask = msg["bid_0"], msg["ask_0"]
        bid_sizes = [msg[f"bid_{k}_size"] for k in range(BID_LEVELS)]
        ask_sizes = [msg[f"ask_{k}_size"] for k in range(ASK_LEVELS)]
        bid_rel = [best_bid - msg[f"bid_{k}"] for k in range(1, BID_LEVELS)]
        ask_rel = [msg[f"ask_{k}"] - best_ask for k in range(1, ASK_LEVELS)]

        for k in range(BID_LEVELS):
            s(row, f"sd.bid_{k}_size", bid_sizes[k])
        for k in range(ASK_LEVELS):
            s(row, f"sd.ask_{k}_size", ask_sizes[k])
        for k in range(1, BID_LEVELS):
            s(row, f"sd.bid_{k}", bid_rel[k - 1])
        for k in range(1, ASK_LEVELS):
            s(row, f"sd.ask_{k}", ask_rel[k - 1])

        # size-weighted order-distance averages (spark_consumer.py:320-340);
        # level-0 distance is 0, weights are the level-1.. sizes
        bw = sum(bid_sizes[1:])
        aw = sum(ask_sizes[1:])
        s(row, "sd.bids_ord_WA",
          sum(r * w for r, w in zip(bid_rel, bid_sizes[1:])) / max(bw, 1.0))
        s(row, "sd.asks_ord_WA",
          sum(r * w for r, w in zip(ask_rel, ask_sizes[1:])) / max(aw, 1.0))

        tot_bid, tot_ask = sum(bid_sizes), sum(ask_sizes)
        s(row, "sd.vol_imbalance",
          (tot_bid - tot_ask) / max(tot_bid + tot_ask, 1.0))
        s(row, "sd.delta", tot_bid - tot_ask)
        b0, a0 = bid_sizes[0], ask_sizes[0]
        s(row, "sd.micro_price",
          (best_bid * a0 + best_ask * b0) / max(a0 + b0, 1.0))
        s(row, "sd.spread", best_bid - best_ask)   # stored negated

        # calendar one-hots + session_start (spark_consumer.py:402-432)
        i = self._bar(msg["Timestamp"])
        day_idx = (i // BARS_PER_DAY) % 5
        week_idx = ((i // BARS_PER_DAY) // 5) % 4
        s(row, "sd.session_start", 1.0 if i % BARS_PER_DAY == 0 else 0.0)
        for d in range(1, 5):
            s(row, f"sd.day_{d}", 1.0 if day_idx == d else 0.0)
        for w in range(1, 5):
            s(row, f"sd.week_{w}", 1.0 if week_idx == (w - 1) else 0.0)

        # forward-filled joins (the interval-join tolerance of
        # spark_consumer.py:434-477 collapses to last-seen over virtual time)
        if self._last_vix is not None:
            s(row, "sd.VIX", self._last_vix)
        if self._last_cot is not None:
            for k, v in self._last_cot.items():
                if k != "Timestamp":
                    s(row, f"sd.{k}", v)

        if self.emit_signal:
            self.bus.publish("predict_timestamp",
                             {"Timestamp": msg["Timestamp"]})

    def _on_volume(self, msg: dict) -> None:
        row = self._row(msg["Timestamp"])
        s = self._set
        o, h, lo, c = (msg["1_open"], msg["2_high"], msg["3_low"],
                       msg["4_close"])
        for name in ("1_open", "2_high", "3_low", "4_close", "5_volume"):
            s(row, f"sd.{name}", msg[name])
        # wick % (spark_consumer.py:187-193)
        body_hi, body_lo = max(o, c), min(o, c)
        s(row, "sd.wick_prct",
          ((h - body_hi) - (body_lo - lo)) / max(h - lo, 1e-6))

    def _on_vix(self, msg: dict) -> None:
        self._last_vix = msg["VIX"]
        row = self._rows.get(self._bar(msg["Timestamp"]))
        if row is not None:
            self._set(row, "sd.VIX", msg["VIX"])

    def _on_cot(self, msg: dict) -> None:
        self._last_cot = msg
        row = self._rows.get(self._bar(msg["Timestamp"]))
        if row is not None:
            for k, v in msg.items():
                if k != "Timestamp":
                    self._set(row, f"sd.{k}", v)

    def _on_ind(self, msg: dict) -> None:
        row = self._row(msg["Timestamp"])
        for event in EVENT_LIST_REPL:
            for value in EVENT_VALUES:
                self._set(row, f"sd.{event}_{value}", msg[event][value])

    # ------------------------------------------------------------------ #

    def table(self) -> torch.Tensor:
        """Assembled (n, 108) float32 table with the windowed SQL-VIEW
        features computed over the completed rows."""
        idx = sorted(self._order)
        X = torch.tensor([self._rows[i] for i in idx], dtype=torch.float32)
        if X.numel() == 0:
            return X.reshape(0, N_FEATURES)
        c = self._col
        close = X[:, c["sd.4_close"]]
        high = X[:, c["sd.2_high"]]
        low = X[:, c["sd.3_low"]]

        X[:, c["vol.vol_MA6"]] = _rolling_mean(X[:, c["sd.5_volume"]], 6)
        X[:, c["vol.vol_MA20"]] = _rolling_mean(X[:, c["sd.5_volume"]], 20)
        X[:, c["p.price_MA20"]] = _rolling_mean(close, 20)
        X[:, c["d.delta_MA12"]] = _rolling_mean(X[:, c["sd.delta"]], 12)

        bb_avg = _rolling_mean(close, 20)
        bb_std = _rolling_std(close, 20)
        X[:, c["bb.upper_BB_dist"]] = (bb_avg + 2 * bb_std) - close
        X[:, c["bb.lower_BB_dist"]] = close - (bb_avg - 2 * bb_std)

        min15 = _rolling_min(close, 15)
        max15 = _rolling_max(close, 15)
        X[:, c["so.stoch"]] = (close - min15) / torch.clamp(
            max15 - min15, min=1e-6)
        X[:, c["ATR.ATR"]] = _rolling_mean(high - low, 15)

        pc = torch.zeros(len(idx))
        pc[1:] = close[1:] - close[:-1]
        X[:, c["pc.price_change"]] = pc
        return X
