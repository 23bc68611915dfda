"""Round-trip test of the raw-feed decomposition: per-topic producers ->
in-process bus -> stream assembler must rebuild the exact joined feature
table the synthetic market generated directly (the reference's
producer -> Kafka -> spark_consumer -> MariaDB chain collapsed in-process)."""
import torch

from fmda_amd.data.generator import SyntheticMarket
from fmda_amd.features import FEATURE_NAMES
from fmda_amd.runtime.assembler import StreamAssembler
from fmda_amd.runtime.bus import MessageBus
from fmda_amd.runtime.producers import FeedProducers


def _roundtrip(n_rows=450, seed=77):
    market = SyntheticMarket(n_rows, seed=seed)
    bus = MessageBus()
    asm = StreamAssembler(bus, emit_signal=True)
    prod = FeedProducers(market, bus)
    prod.run()
    return market, bus, asm.table()


def test_assembled_table_matches_generator():
    market, _, X = _roundtrip()
    assert X.shape == market.X.shape
    # column-by-column so a drift names the feature
    for j, name in enumerate(FEATURE_NAMES):
        torch.testing.assert_close(
            X[:, j], market.X[:, j], rtol=2e-4, atol=2e-4,
            msg=lambda m, name=name: f"{name}: {m}")


def test_one_hots_and_sparse_columns_exact():
    market, _, X = _roundtrip(n_rows=400, seed=3)
    col = {n: i for i, n in enumerate(FEATURE_NAMES)}
    exact = (["sd.session_start"] + [f"sd.day_{d}" for d in range(1, 5)]
             + [f"sd.week_{w}" for w in range(1, 5)] + ["sd.VIX"])
    # every indicator-event column: zero except at fired bars, exact values
    exact += [n for n in FEATURE_NAMES if "_Actual" in n or "_diff" in n]
    for name in exact:
        assert torch.equal(X[:, col[name]], market.X[:, col[name]]), name


def test_predict_timestamp_signal_per_bar():
    market, bus, _ = _roundtrip(n_rows=100, seed=5)
    t = bus.topic("predict_timestamp")
    assert t.end_offset() == market.n_rows


def test_cot_forward_fill_weekly():
    market, _, X = _roundtrip(n_rows=450, seed=9)  # > one 390-bar week
    col = {n: i for i, n in enumerate(FEATURE_NAMES)}
    j = col["sd.Asset_long_pos"]
    torch.testing.assert_close(X[:, j], market.X[:, j], rtol=1e-6, atol=1e-6)
    # the weekly value actually changes across the week boundary
    assert float(X[0, j]) != float(X[449, j])


def test_sanitize_keys_and_value_to_number():
    from fmda_amd.runtime.producers import sanitize_keys, value_to_number
    d = sanitize_keys({"1. open": 1, "Fed-Rate Decision": {"a b": 2}})
    assert d == {"1_open": 1, "Fed_Rate_Decision": {"a_b": 2}}
    assert value_to_number("1,234") == 1234
    assert value_to_number("2.5K") == 2500.0
    assert value_to_number("1.2M") == 1200000.0
    assert value_to_number("3.1%") == 3.1
    assert value_to_number("n/a") == "n/a"
    assert value_to_number(7) == 7


def test_indicator_dedup_registry(tmp_path):
    import pickle
    from fmda_amd.data.generator import SyntheticMarket
    from fmda_amd.runtime.bus import MessageBus
    from fmda_amd.runtime.producers import FeedProducers
    market = SyntheticMarket(300, seed=11)
    reg = str(tmp_path / "items.pickle")
    bus = MessageBus()
    prod = FeedProducers(market, bus, registry_path=reg)
    prod.run()
    prod.save_registry()
    n_first = bus.topic("ind").end_offset()
    assert n_first > 0
    # a second producer with the persisted registry publishes nothing new
    bus2 = MessageBus()
    prod2 = FeedProducers(market, bus2, registry_path=reg)
    prod2.run()
    assert bus2.topic("ind").end_offset() == 0
    with open(reg, "rb") as f:
        assert len(pickle.load(f)) > 0


def test_out_of_order_and_late_messages():
    """The reference handles late data with Spark watermarks
    (spark_consumer.py per-stream withWatermark); the in-process assembler
    must tolerate out-of-order bars and late joins keyed by the 5-minute
    floor."""
    from fmda_amd.data.generator import SyntheticMarket
    from fmda_amd.runtime.assembler import StreamAssembler
    from fmda_amd.runtime.bus import MessageBus
    from fmda_amd.runtime.producers import FeedProducers

    market = SyntheticMarket(60, seed=21)
    bus = MessageBus()
    asm = StreamAssembler(bus, emit_signal=False)
    prod = FeedProducers(market, bus)
    # publish bars out of order: evens first, then odds (vix/cot of a bar
    # arrive with their own bar's deep, so forward-fill sees jumps)
    for i in range(0, 60, 2):
        prod.publish_bar(i)
    for i in range(1, 60, 2):
        prod.publish_bar(i)
    X = asm.table()
    assert X.shape == market.X.shape
    col = {n: j for j, n in enumerate(FEATURE_NAMES)}
    # bar-keyed (non-forward-filled) columns must still match exactly
    for name in ("sd.4_close", "sd.5_volume", "sd.bid_0_size",
                 "sd.session_start"):
        assert torch.equal(X[:, col[name]], market.X[:, col[name]]), name
    # windowed features are computed on the sorted table -> exact too
    torch.testing.assert_close(X[:, col["p.price_MA20"]],
                               market.X[:, col["p.price_MA20"]],
                               rtol=1e-5, atol=1e-5)


def test_assembler_tolerates_missing_optional_topics():
    """Rows assemble even when vix/cot/ind never arrive (zeros like the
    reference table's IFNULL(...,0) fetch)."""
    from fmda_amd.data.generator import SyntheticMarket
    from fmda_amd.runtime.assembler import StreamAssembler
    from fmda_amd.runtime.bus import MessageBus
    from fmda_amd.runtime.producers import FeedProducers

    market = SyntheticMarket(40, seed=22)
    bus = MessageBus()
    asm = StreamAssembler(bus, emit_signal=False)
    prod = FeedProducers(market, bus)
    for i in range(40):
        ts = prod.t0 + i * prod.freq
        # deep + volume only
        full_bus = MessageBus()
        p2 = FeedProducers(market, full_bus)
        p2.publish_bar(i)
        deep = full_bus.topic("deep")._buf[0]
        vol = full_bus.topic("volume")._buf[0]
        bus.publish("deep", deep)
        bus.publish("volume", vol)
    X = asm.table()
    assert X.shape[0] == 40
    col = {n: j for j, n in enumerate(FEATURE_NAMES)}
    assert torch.all(X[:, col["sd.VIX"]] == 0)
    assert torch.all(X[:, col["sd.Asset_long_pos"]] == 0)
    assert torch.equal(X[:, col["sd.4_close"]],
                       market.X[:, col["sd.4_close"]])


def test_late_arrival_after_table_read():
    """A bar arriving AFTER table() has already been read must appear at
    its correct time position on the next read, with the windowed features
    (MAs, price_change) recomputed to include it — the in-process
    equivalent of the reference's delayed-data acceptance
    (getMarketData.py:208-218): late rows are incorporated, not dropped."""
    market = SyntheticMarket(60, seed=9)
    bus = MessageBus()
    asm = StreamAssembler(bus, emit_signal=False)
    prod = FeedProducers(market, bus)
    # publish bars 0..49 except bar 30 (delayed upstream)
    prod.run_range(0, 30)
    prod.run_range(31, 50)
    X1 = asm.table()
    assert X1.shape[0] == 49           # the missing bar is simply absent

    # the late bar lands after the first read
    prod.run_range(30, 31)
    X2 = asm.table()
    assert X2.shape[0] == 50
    col = {n: i for i, n in enumerate(FEATURE_NAMES)}
    # the late row sits at its correct position with its exact raw values
    torch.testing.assert_close(X2[30, col["sd.4_close"]],
                               market.X[30, col["sd.4_close"]])
    # downstream windowed features now include it (recomputed, not stale):
    # price_change at bar 31 pairs with the true bar-30 close again
    torch.testing.assert_close(X2[:50, col["pc.price_change"]],
                               market.X[:50, col["pc.price_change"]],
                               rtol=2e-4, atol=2e-4)
