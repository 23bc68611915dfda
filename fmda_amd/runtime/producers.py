"""Per-topic synthetic feed producers.

The reference acquisition layer publishes five raw Kafka topics (reference
config.py:15): `deep` (IEX DEEP order-book snapshots, getMarketData.py:82-137
reshaped to flat bid_i/ask_i keys at :117-127), `volume` (Alpha Vantage
intraday OHLCV with sanitized `1_open`..`5_volume` keys, getMarketData.py:139,
change_keys :10), `vix` ({'VIX', 'Timestamp'} from the VIX spider pipeline,
vix_spider.py:43-47), `cot` (12 Asset/Leveraged long/short position fields,
cot_reports_spider.py:125-156) and `ind` (the 13-event indicator template of
config.py:58-65, economic_indicators_spider.py:201-209).

`FeedProducers` re-creates that decomposition over the synthetic market: it
derives the raw per-topic messages from a `SyntheticMarket` table (absolute
book prices reconstructed from the stored close/spread/relative distances)
and publishes them to the in-process bus, so the downstream
`StreamAssembler` (the spark_consumer.py equivalent) can rebuild the joined
108-feature rows from raw messages alone.

Publication cadence mirrors the reference: deep/volume/vix every 5-minute
bar (producer.py:111-150), cot once per week (weekly report,
producer.py:142), ind only on bars where an event fires.
"""
from typing import Optional

from ..config import (ASK_LEVELS, BID_LEVELS, EVENT_LIST_REPL, EVENT_VALUES,
                      empty_indicator_message)
from ..features import FEATURE_NAMES
from .bus import MessageBus

BARS_PER_DAY = 78
BARS_PER_WEEK = BARS_PER_DAY * 5


def sanitize_keys(d: dict) -> dict:
    """Key sanitizer with the reference's `change_keys` semantics
    (getMarketData.py:10-35): strips the Alpha-Vantage-style numbered-dot
    prefixes' separators — spaces, dots and dashes become underscores so
    keys are valid SQL column / dict identifiers."""
    out = {}
    for k, v in d.items():
        kk = str(k).replace(". ", "_").replace(".", "_")
        kk = kk.replace(" ", "_").replace("-", "_")
        out[kk] = sanitize_keys(v) if isinstance(v, dict) else v
    return out


def value_to_number(v):
    """String->number coercion of the reference `value_to_number`
    (getMarketData.py:38-58): numeric strings become int/float,
    K/M/B/% suffixed quantities are scaled, everything else passes
    through unchanged."""
    if not isinstance(v, str):
        return v
    s = v.strip().replace(",", "")
    mult = 1.0
    if s.endswith("%"):
        s = s[:-1]
    elif s[-1:] in ("K", "M", "B"):
        mult = {"K": 1e3, "M": 1e6, "B": 1e9}[s[-1]]
        s = s[:-1]
    try:
        f = float(s) * mult
        return int(f) if f.is_integer() and "." not in s and mult == 1.0 else f
    except ValueError:
        return v


class FeedProducers:
    """Publish the five raw topics for each bar of a SyntheticMarket."""

    def __init__(self, market, bus: Optional[MessageBus] = None,
                 freq: float = 300.0, t0: float = 0.0,
                 registry_path: Optional[str] = None):
        self.market = market
        self.bus = bus or MessageBus()
        self.freq = freq
        self.t0 = t0
        self._col = {name: i for i, name in enumerate(FEATURE_NAMES)}
        # Indicator dedup registry: the reference pipeline keeps a pickle
        # of already-published events and only sends new ones
        # (economic_indicators_spider.py:42-48,94-96; loaded at
        # producer.py:108-109). Same mechanism, optionally persisted.
        self.registry_path = registry_path
        self._ind_registry = {}
        if registry_path is not None:
            import os
            import pickle
            if os.path.exists(registry_path):
                with open(registry_path, "rb") as f:
                    self._ind_registry = pickle.load(f)

    def save_registry(self) -> None:
        if self.registry_path is not None:
            import pickle
            with open(self.registry_path, "wb") as f:
                pickle.dump(self._ind_registry, f)

    def _x(self, i: int, name: str) -> float:
        return float(self.market.X[i, self._col[name]])

    def publish_bar(self, i: int) -> float:
        """Publish every topic due at bar i; returns the bar timestamp."""
        ts = self.t0 + i * self.freq
        x = self._x

        # --- deep: flat book snapshot with ABSOLUTE level prices
        # (getMarketData.py:117-127 shape). The table stores the close,
        # the (negated) spread and per-level distances from best; the raw
        # snapshot carries prices, which the assembler re-derives from.
        close = x(i, "sd.4_close")
        spread_mag = -x(i, "sd.spread")
        best_bid = close - spread_mag / 2.0
        best_ask = close + spread_mag / 2.0
        deep = {"Timestamp": ts, "bid_0": best_bid, "ask_0": best_ask}
        for k in range(BID_LEVELS):
            deep[f"bid_{k}_size"] = x(i, f"sd.bid_{k}_size")
        for k in range(ASK_LEVELS):
            deep[f"ask_{k}_size"] = x(i, f"sd.ask_{k}_size")
        for k in range(1, BID_LEVELS):
            deep[f"bid_{k}"] = best_bid - x(i, f"sd.bid_{k}")
        for k in range(1, ASK_LEVELS):
            deep[f"ask_{k}"] = best_ask + x(i, f"sd.ask_{k}")
        self.bus.publish("deep", deep)

        # --- volume: sanitized AV intraday OHLCV keys (change_keys,
        # getMarketData.py:10-35)
        self.bus.publish("volume", {
            "Timestamp": ts,
            "1_open": x(i, "sd.1_open"), "2_high": x(i, "sd.2_high"),
            "3_low": x(i, "sd.3_low"), "4_close": close,
            "5_volume": x(i, "sd.5_volume"),
        })

        # --- vix (vix_spider.py:43-47 message shape)
        self.bus.publish("vix", {"VIX": x(i, "sd.VIX"), "Timestamp": ts})

        # --- cot: weekly report (cot_reports_spider.py:125-156 fields)
        if i % BARS_PER_WEEK == 0 or i == 0:
            cot = {"Timestamp": ts}
            for side in ("Asset", "Leveraged"):
                for direction in ("long", "short"):
                    for suffix in ("pos", "pos_change", "open_int"):
                        name = f"{side}_{direction}_{suffix}"
                        cot[name] = x(i, f"sd.{name}")
            self.bus.publish("cot", cot)

        # --- ind: template message only on bars where an event fired
        # (economic_indicators_spider.py:201-209; template config.py:58-65)
        fired = False
        msg = empty_indicator_message()
        msg["Timestamp"] = ts
        for event in EVENT_LIST_REPL:
            vals = tuple(x(i, f"sd.{event}_{v}") for v in EVENT_VALUES)
            if any(v != 0.0 for v in vals):
                # dedup: an event row already in the registry is not
                # re-sent (the registry accumulates every published item,
                # like the reference items.pickle)
                seen = self._ind_registry.setdefault(event, set())
                if vals in seen:
                    continue
                seen.add(vals)
                for value, v in zip(EVENT_VALUES, vals):
                    msg[event][value] = v
                fired = True
        if fired:
            self.bus.publish("ind", msg)
        return ts

    def run(self, max_bars: Optional[int] = None) -> int:
        n = self.market.n_rows if max_bars is None else min(
            max_bars, self.market.n_rows)
        for i in range(n):
            self.publish_bar(i)
        return n

    def run_range(self, start: int, stop: int) -> int:
        """Publish bars [start, stop) — out-of-order/late-arrival test
        harness (the reference's delayed-data scenarios,
        getMarketData.py:208-218)."""
        stop = min(stop, self.market.n_rows)
        for i in range(start, stop):
            self.publish_bar(i)
        return max(0, stop - start)
