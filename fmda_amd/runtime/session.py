"""Market-session orchestration over synthetic time.

Replaces the reference producer (producer.py): the market-calendar gate
(start_day_session, producer.py:168-254), the FOREX-hours fallback
(:239-243) and the self-correcting polling loop (intraday_data, :52-150)
become a session driver over the synthetic market table. Each poll tick
publishes the bar's feature row to the `deep` topic and its timestamp to
`predict_timestamp` (the signal spark_consumer.py:490-502 produced), pushing
the row into the predictor's ring buffer on the way.

Time is virtual by default (no sleeping); `pace` > 0 plays bars back in
real time for demos.
"""
import time
from dataclasses import dataclass
from typing import Optional


from ..data.generator import SyntheticMarket
from .bus import MessageBus
from .streaming import StreamingPredictor


@dataclass
class MarketCalendar:
    """Synthetic market calendar with the reference's full gating
    semantics (get_market_calendar, getMarketData.py:251; gated at
    producer.py:218-254):

    - per-day status: weekday gate + HOLIDAY table (the Tradier calendar's
      'closed' days);
    - intraday market hours: each open day trades bars [0, close); an
      EARLY-CLOSE entry shortens `close` (the reference reads per-day
      open/close times from the calendar entry and bounds the poll loop);
    - instrument-conditioned FOREX fallback (producer.py:239-243): a
      non-IEX instrument trades the continuous FOREX week — no intraday
      bounds and no holidays, dark only on Saturday (the Sunday-17:00 ->
      Friday-16:00 ET week window, compressed to the synthetic day grid).

    Days are absolute indices from the session start (day 0 = Monday);
    bars within a day index the 5-minute grid.
    """
    bars_per_day: int = 78          # 6.5 h of 5-min bars
    open_days: tuple = (0, 1, 2, 3, 4)  # Mon-Fri weekdays
    holidays: tuple = ()            # absolute day indices with status closed
    early_close: Optional[dict] = None  # day index -> bars traded that day
    forex_fallback: bool = False    # non-IEX instrument: FOREX week window

    def day_index(self, bar_index: int) -> int:
        return bar_index // self.bars_per_day

    def status(self, day: int) -> str:
        """'open' | 'early' | 'closed' for an absolute day index."""
        if self.forex_fallback:
            return "closed" if day % 7 == 5 else "open"   # Saturday only
        if (day % 7) not in self.open_days or day in self.holidays:
            return "closed"
        if self.early_close and day in self.early_close:
            return "early"
        return "open"

    def market_hours(self, day: int):
        """(open_bar, close_bar) within the day's bar grid; (0, 0) when
        closed. The FOREX week has no intraday bounds."""
        st = self.status(day)
        if st == "closed":
            return (0, 0)
        if self.forex_fallback:
            return (0, self.bars_per_day)
        if st == "early":
            return (0, int(self.early_close[day]))
        return (0, self.bars_per_day)

    def is_open(self, bar_index: int) -> bool:
        day = self.day_index(bar_index)
        lo, hi = self.market_hours(day)
        return lo <= (bar_index % self.bars_per_day) < hi


class MarketSession:
    """Polling session: one tick per 5-minute bar of the synthetic market."""

    def __init__(self, market: SyntheticMarket, bus: Optional[MessageBus] = None,
                 predictor: Optional[StreamingPredictor] = None,
                 calendar: Optional[MarketCalendar] = None,
                 freq: float = 300.0, pace: float = 0.0,
                 start_time: Optional[float] = None):
        self.market = market
        self.bus = bus or MessageBus()
        self.predictor = predictor
        self.calendar = calendar or MarketCalendar()
        self.freq = freq
        self.pace = pace
        self.t0 = time.time() if start_time is None else start_time
        self.cursor = 0
        self.published = 0

    def tick(self) -> bool:
        """Publish one bar; returns False when the table is exhausted."""
        if self.cursor >= self.market.n_rows:
            return False
        i = self.cursor
        self.cursor += 1
        if not self.calendar.is_open(i):
            return True  # market closed: skip the bar (producer.py:159-165)
        ts = self.t0 + i * self.freq
        row = self.market.X[i]
        self.bus.publish("deep", {"Timestamp": ts, "row": row})
        if self.predictor is not None:
            self.predictor.push_row(row, ts=ts)
        self.bus.publish("predict_timestamp", {"Timestamp": ts})
        self.published += 1
        return True

    def run(self, max_bars: Optional[int] = None) -> int:
        """Self-correcting polling loop (producer.py:111-150); virtual time
        unless pace > 0."""
        n = 0
        try:
            while self.tick():
                n += 1
                if max_bars is not None and n >= max_bars:
                    break
                if self.pace > 0:
                    next_t = time.time() + self.pace
                    delay = next_t - time.time()
                    if delay > 0:
                        time.sleep(delay)
        except KeyboardInterrupt:  # graceful stop (producer.py:155-157)
            pass
        return n
