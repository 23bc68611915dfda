#!/usr/bin/env python3
"""End-to-end demo: the full reference pipeline shape, natively.

    synthetic market (Kafka/Spark/MariaDB stand-in)
      -> chunked/windowed training (notebook semantics)
      -> checkpoint in the reference model_params.pt format
      -> streaming inference session (predict.py semantics)

Runs on CPU in under a minute; pass --device cuda on an MI355X to train
through the HIP engine and hipGraph-capture the predictor.
"""
import argparse
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from fmda_amd.config import DataConfig, ModelConfig, TrainConfig
from fmda_amd.data import SyntheticMarket
from fmda_amd.data.norm import load_norm_params
from fmda_amd.models.checkpoint import load_checkpoint
from fmda_amd.runtime import MarketSession, MessageBus, StreamingPredictor
from fmda_amd.train import train


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--rows", type=int, default=800)
    ap.add_argument("--outdir", default="/tmp/fmda_demo")
    args = ap.parse_args()
    os.makedirs(args.outdir, exist_ok=True)
    ck = os.path.join(args.outdir, "model_params.pt")
    np_path = os.path.join(args.outdir, "norm_params")

    print("== raw feed assembly ==")
    # the reference's producer -> Kafka -> Spark-join chain, in-process:
    # per-topic raw messages rebuilt into the joined 108-feature table
    from fmda_amd.data import SyntheticMarket as _SM
    from fmda_amd.runtime import FeedProducers, StreamAssembler
    feed_market = _SM(300, seed=13)
    feed_bus = MessageBus()
    assembler = StreamAssembler(feed_bus, emit_signal=False)
    FeedProducers(feed_market, feed_bus).run()
    X_assembled = assembler.table()
    dev = (X_assembled - feed_market.X).abs().max()
    print(f"assembled {tuple(X_assembled.shape)} from raw topics; "
          f"max deviation vs direct table: {float(dev):.2e}")

    print("== training ==")
    mcfg = ModelConfig(hidden_size=16, spatial_dropout=False, dropout=0.3)
    dcfg = DataConfig(n_rows=args.rows, chunk_size=100, window=20)
    tcfg = TrainConfig(batch_size=8, epochs=args.epochs, device=args.device)
    train(mcfg, dcfg, tcfg, checkpoint_path=ck, norm_params_path=np_path)

    print("== streaming inference ==")
    model = load_checkpoint(ck)
    _, x_min, x_max = load_norm_params(np_path)
    market = SyntheticMarket(200, seed=7)
    bus = MessageBus()
    predictor = StreamingPredictor(model, x_min, x_max, window=20,
                                   device=args.device)
    # wire the predict_timestamp topic into the predictor, publishing
    # results to the prediction topic (predict.py:124-197 message loop)
    preds = []

    def on_timestamp(msg):
        out = predictor.handle_timestamp(msg, now=msg["Timestamp"])
        if out is not None:
            bus.publish("prediction", out)
            preds.append(out)

    bus.topic("predict_timestamp").subscribe(on_timestamp)
    session = MarketSession(market, bus=bus, predictor=predictor)
    bars = session.run(max_bars=120)
    print(f"{bars} bars -> {len(preds)} predictions; last:",
          {k: preds[-1][k] for k in ("pred_labels", "timestamp")}
          if preds else None)


if __name__ == "__main__":
    main()
