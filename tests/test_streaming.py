"""Streaming inference path: bus, ring buffer, stale filter, end-to-end
session -> predictor -> prediction topic (reference predict.py semantics)."""
import time

import pytest
import torch

from fmda_amd.data import SyntheticMarket, ChunkLoader, load_norm_params
from fmda_amd.models import BiGRU
from fmda_amd.runtime import (FeatureRing, MarketCalendar, MarketSession,
                              MessageBus, StreamingPredictor)


def _predictor(window=5, bus=None, **kw):
    torch.manual_seed(0)
    m = BiGRU(8, 108, 4, spatial_dropout=False, dropout=0.2)
    x_min = torch.zeros(108)
    x_max = torch.ones(108)
    return StreamingPredictor(m, x_min, x_max, window, bus=bus,
                              use_graph=False, **kw)


def test_bus_publish_consume():
    bus = MessageBus()
    for i in range(5):
        bus.publish("vix", {"VIX": float(i)})
    got = list(bus.consume("vix", from_end=False, timeout=0.01))
    assert [m["VIX"] for m in got] == [0.0, 1.0, 2.0, 3.0, 4.0]
    # seek-to-end semantics (predict.py:30)
    assert list(bus.consume("vix", from_end=True, timeout=0.01)) == []


def test_ring_buffer_window():
    ring = FeatureRing(3, 4)
    for i in range(5):
        ring.push(torch.full((4,), float(i)))
    assert ring.full
    assert torch.equal(ring.buf[:, 0], torch.tensor([2.0, 3.0, 4.0]))


def test_stale_message_filter():
    p = _predictor()
    for i in range(6):
        p.push_row(torch.rand(108))
    now = time.time()
    assert p.handle_timestamp({"Timestamp": now - 300.0}, now=now) is None
    assert p.handle_timestamp({"Timestamp": now - 10.0}, now=now) is not None


def test_prediction_schema_and_threshold():
    p = _predictor()
    for i in range(6):
        p.push_row(torch.rand(108))
    pred = p.predict_window()
    assert set(pred) == {"probabilities", "prob_threshold", "pred_indices",
                        "pred_labels"}
    probs = torch.tensor(pred["probabilities"])
    assert probs.shape == (4,)
    expect_idx = (probs > 0.5).nonzero().flatten().tolist()
    assert pred["pred_indices"] == expect_idx
    assert pred["pred_labels"] == [["up1", "up2", "down1", "down2"][i]
                                   for i in expect_idx]


def test_session_end_to_end(tmp_path):
    """Session publishes bars; predictor consumes timestamps and publishes
    predictions with real norm_params scaling."""
    mk = SyntheticMarket(260, seed=5)
    path = str(tmp_path / "norm_params")
    ChunkLoader(mk.X, chunk_size=100, window=5, norm_params_path=path)
    names, x_min, x_max = load_norm_params(path)

    bus = MessageBus()
    torch.manual_seed(1)
    model = BiGRU(8, 108, 4, spatial_dropout=False)
    pred = StreamingPredictor(model, x_min, x_max, window=5, bus=bus,
                              use_graph=False, stale_after=1e12)
    sess = MarketSession(mk, bus=bus, predictor=pred,
                         calendar=MarketCalendar(forex_fallback=True))
    # interleave ticks with message handling (live operation)
    for _ in range(50):
        assert sess.tick()
        pred.run(max_messages=1, timeout=0.01)
    preds = list(bus.consume("prediction", from_end=False, timeout=0.01))
    # first window-1 bars have no full window yet
    assert len(preds) == 50 - 4
    assert all(torch.isfinite(torch.tensor(p["probabilities"])).all()
               for p in preds)


def test_calendar_gate():
    cal = MarketCalendar()
    assert cal.is_open(0)            # Monday
    assert not cal.is_open(5 * 78)   # Saturday
    # FOREX week window (producer.py:239-243): continuous Sun->Fri, dark
    # only on Saturday; no holiday or intraday gating
    fx = MarketCalendar(forex_fallback=True, holidays=(0,),
                        early_close={1: 10})
    assert not fx.is_open(5 * 78)    # Saturday stays dark
    assert fx.is_open(6 * 78)        # Sunday session
    assert fx.is_open(0)             # holiday ignored for FOREX
    assert fx.is_open(1 * 78 + 50)   # early close ignored for FOREX


def test_calendar_holiday_and_early_close():
    """Holiday-skip day and an early close (reference market calendar's
    per-day status + open/close hours, producer.py:218-243)."""
    cal = MarketCalendar(holidays=(2,), early_close={3: 39})
    assert cal.status(0) == "open"
    assert cal.status(2) == "closed" and not cal.is_open(2 * 78)
    assert cal.status(3) == "early"
    assert cal.market_hours(3) == (0, 39)
    assert cal.is_open(3 * 78 + 38)       # last bar before the early close
    assert not cal.is_open(3 * 78 + 39)   # first bar after
    assert cal.market_hours(2) == (0, 0)
    assert cal.market_hours(5) == (0, 0)  # weekend


def test_session_skips_holiday_and_early_close_bars():
    from fmda_amd.data.generator import SyntheticMarket
    from fmda_amd.runtime import MarketSession, MessageBus

    market = SyntheticMarket(78 * 7, seed=5)
    cal = MarketCalendar(holidays=(1,), early_close={2: 39})
    bus = MessageBus()
    s = MarketSession(market, bus=bus, calendar=cal)
    s.run()
    # Mon full + Tue holiday (0) + Wed half + Thu/Fri full, weekend closed
    assert s.published == 78 + 0 + 39 + 78 + 78


def test_bus_threaded_producer_consumer():
    """Topic reads block until publish (condition variable) and deliver in
    order across threads — the Kafka-consumer-loop semantics predict.py
    relies on."""
    import threading
    import time

    from fmda_amd.runtime.bus import MessageBus

    bus = MessageBus()
    got = []

    def consume():
        for msg in bus.consume("predict_timestamp", from_end=False,
                               timeout=5.0):
            got.append(msg["i"])
            if msg["i"] == 99:
                return

    t = threading.Thread(target=consume)
    t.start()
    for i in range(100):
        bus.publish("predict_timestamp", {"i": i})
        if i % 25 == 0:
            time.sleep(0.002)   # let the consumer interleave
    t.join(timeout=10)
    assert not t.is_alive()
    assert got == list(range(100))


def test_bus_seek_to_end_semantics():
    """from_end=True must skip everything already published (the reference
    consumer's seek_to_end, predict.py:30)."""
    from fmda_amd.runtime.bus import MessageBus

    bus = MessageBus()
    for i in range(5):
        bus.publish("vix", {"i": i})
    it = bus.consume("vix", from_end=True, timeout=0.05)
    bus.publish("vix", {"i": 99})
    first = next(it)
    assert first["i"] == 99


def test_market_calendar_gates_bars():
    """Closed days are skipped (producer.py:159-165 market-closed path);
    the FOREX fallback keeps a 24h session (producer.py:239-243)."""
    from fmda_amd.data.generator import SyntheticMarket
    from fmda_amd.runtime import MarketCalendar, MarketSession, MessageBus

    market = SyntheticMarket(78 * 7, seed=3)   # one full week of bars
    cal = MarketCalendar(bars_per_day=78, open_days=(0, 1))  # Mon+Tue only
    bus = MessageBus()
    s = MarketSession(market, bus=bus, calendar=cal)
    s.run()
    assert s.published == 78 * 2          # two open days of the seven
    assert bus.topic("deep").end_offset() == 78 * 2

    cal24 = MarketCalendar(forex_fallback=True)
    s2 = MarketSession(market, bus=MessageBus(), calendar=cal24)
    s2.run()
    assert s2.published == 78 * 6         # FOREX week: only Saturday dark


def test_delayed_data_retry_and_drop():
    """Delayed-data tolerance (predict.py:141-157): a timestamp message
    whose feature row hasn't arrived is retried once after the settle
    delay, and dropped if the row is still missing."""
    import threading

    from fmda_amd.models import BiGRU
    from fmda_amd.runtime import StreamingPredictor

    torch.manual_seed(0)
    model = BiGRU(8, 16, 4, spatial_dropout=False)
    now = 1_000_000.0

    # row never arrives -> retry once, then drop
    p = StreamingPredictor(model, torch.zeros(16), torch.ones(16), window=3,
                           use_graph=False, settle_delay=0.0)
    for i in range(3):
        p.push_row(torch.rand(16), ts=now + i)
    assert p.handle_timestamp({"Timestamp": now + 10}, now=now + 10) is None
    assert p.n_retries == 1 and p.n_dropped_missing == 1

    # row lands during the settle sleep (producer lag) -> prediction runs
    p2 = StreamingPredictor(model, torch.zeros(16), torch.ones(16), window=3,
                            use_graph=False, settle_delay=0.2)
    for i in range(3):
        p2.push_row(torch.rand(16), ts=now + i)
    t = threading.Timer(0.05, lambda: p2.push_row(torch.rand(16),
                                                  ts=now + 10))
    t.start()
    out = p2.handle_timestamp({"Timestamp": now + 10}, now=now + 10)
    t.join()
    assert out is not None and p2.n_retries == 1
    assert p2.n_dropped_missing == 0

    # on-time message needs no retry
    p3 = StreamingPredictor(model, torch.zeros(16), torch.ones(16), window=3,
                            use_graph=False)
    for i in range(3):
        p3.push_row(torch.rand(16), ts=now + i)
    assert p3.handle_timestamp({"Timestamp": now + 2}, now=now + 2) is not None
    assert p3.n_retries == 0
