"""Training driver — the reference notebook (`biGRU_model_training.ipynb`)
as a reproducible script.

Per-epoch flow matches the notebook's cell 29: chunk-level re-split,
per-chunk DataLoader over sliding windows, train_model / evaluate_model
with sklearn-formula metrics, class-imbalance loss weighting (cell 16:
weight = total/positives, pos_weight = negatives/positives), Adam lr=1e-3,
clip=50, and `torch.save(state_dict)` of the best model (cell 39).

Run: python -m fmda_amd.train [--epochs 25] [--device cuda]
"""
import argparse
import json
import os
import time
from typing import Optional

import torch
import torch.nn as nn
from torch.utils.data import DataLoader

from .config import DataConfig, ModelConfig, TrainConfig
from .data import BatchLoader, ChunkLoader, SyntheticMarket, TrainValTestSplit
from .models import BiGRU, save_checkpoint


def class_weights(Y: torch.Tensor):
    """Loss weights from label counts (notebook cell 16)."""
    total = Y.shape[0]
    pos = Y.sum(dim=0).clamp(min=1.0)
    weight = total / pos
    pos_weight = (total - pos) / pos
    return weight, pos_weight


def make_epoch_sets(market: SyntheticMarket, dcfg: DataConfig,
                    norm_params_path: Optional[str] = None):
    chunks = ChunkLoader(market.X[:, :dcfg.n_features], dcfg.chunk_size,
                         dcfg.window, norm_params_path=norm_params_path)
    split = TrainValTestSplit(chunks, dcfg.val_size, dcfg.test_size)
    return split.get_sets()


def chunk_batches(market, chunk_set, dcfg, batch_size):
    """Yield (x, y) batches over every chunk in the set (notebook epoch
    inner loop)."""
    for indices, norms in chunk_set:
        bl = BatchLoader(indices, norms, market.X[:, :dcfg.n_features],
                         market.Y, dcfg.window)
        dl = DataLoader(bl, batch_size=batch_size, shuffle=False)
        for x, y in dl:
            yield x, y


def train(mcfg: ModelConfig = None, dcfg: DataConfig = None,
          tcfg: TrainConfig = None, checkpoint_path: str = "model_params.pt",
          norm_params_path: str = "norm_params", log=print,
          resume: str = None):
    mcfg = mcfg or ModelConfig()
    dcfg = dcfg or DataConfig(n_features=mcfg.n_features)
    tcfg = tcfg or TrainConfig()

    device = torch.device(tcfg.device)
    torch.manual_seed(dcfg.seed)

    market = SyntheticMarket(dcfg.n_rows, seed=dcfg.seed)
    weight, pos_weight = class_weights(market.Y)
    # class-imbalance accounting (notebook cells 13-14 print the dataset
    # size and per-class positive counts before choosing loss weights)
    log(json.dumps({"n_rows": dcfg.n_rows,
                    "positives": [int(v) for v in market.Y.sum(dim=0)],
                    "loss_weight": [round(float(v), 3) for v in weight],
                    "pos_weight": [round(float(v), 3)
                                   for v in pos_weight]}))

    model = BiGRU(mcfg.hidden_size, mcfg.n_features, mcfg.output_size,
                  n_layers=mcfg.n_layers, clip=mcfg.clip,
                  dropout=mcfg.dropout, spatial_dropout=mcfg.spatial_dropout,
                  bidirectional=mcfg.bidirectional).to(device)
    model.add_loss_fn(nn.BCEWithLogitsLoss(weight=weight.to(device),
                                           pos_weight=pos_weight.to(device)))
    model.add_optimizer(torch.optim.Adam(model.parameters(), lr=tcfg.lr))
    model.add_device(device)
    if resume:
        # mid-training resume (reference only ever saves, predict.py:104
        # only loads; this is the SURVEY.md section 5 native extension)
        state = torch.load(resume, map_location=device, weights_only=True)
        sd = state["model"] if "model" in state else state
        # strict=False: the live model carries the attached loss module,
        # whose buffers are deliberately absent from checkpoints
        missing, unexpected = model.load_state_dict(sd, strict=False)
        bad = [k for k in list(missing) + list(unexpected)
               if not k.startswith("loss_fn.")]
        assert not bad, f"resume key mismatch: {bad}" 

    start_epoch = 1
    if resume:
        # full training-state sidecar (optimizer moments, epoch, RNG): a
        # TRUE mid-training resume — the reference only ever saves weights
        # (notebook cell 39) and reloads them in predict.py:104
        ts_path = resume + ".train_state.pt"
        if os.path.exists(ts_path):
            ts = torch.load(ts_path, map_location=device,
                            weights_only=False)
            # the sidecar's weights are the LAST epoch's (the main
            # checkpoint holds the BEST-val epoch — the wrong pairing for
            # the last-epoch optimizer moments)
            model.load_state_dict(ts["model"], strict=False)
            model.optimizer.load_state_dict(ts["optimizer"])
            start_epoch = ts["epoch"] + 1
            torch.set_rng_state(ts["rng"])

    history = []
    best_val_acc = -1.0
    for epoch in range(start_epoch, tcfg.epochs + 1):
        t0 = time.time()
        train_set, val_set, _ = make_epoch_sets(market, dcfg,
                                                norm_params_path)
        tr = model.train_model(
            chunk_batches(market, train_set, dcfg, tcfg.batch_size))
        va = model.evaluate_model(
            chunk_batches(market, val_set, dcfg, tcfg.batch_size))
        from .metrics import three_class_accuracy
        rec = {"epoch": epoch, "train_acc": float(tr[0]),
               "train_hamming": float(tr[1]), "train_loss": float(tr[2]),
               "val_acc": float(va[0]), "val_hamming": float(va[1]),
               # derived up/down/stall accuracy over the 4-label head
               # (SURVEY.md section 6 labeling nuance)
               "val_acc3": float(three_class_accuracy(va[4], va[3])),
               "sec": time.time() - t0}
        history.append(rec)
        log(json.dumps(rec))
        if rec["val_acc"] >= best_val_acc:
            best_val_acc = rec["val_acc"]
            save_checkpoint(model, checkpoint_path)
        # training-state sidecar for exact resume (always current epoch;
        # includes the CURRENT weights — the main checkpoint may hold an
        # earlier best-val epoch)
        sd = {k: v for k, v in model.state_dict().items()
              if not k.startswith("loss_fn.")}
        torch.save({"model": sd,
                    "optimizer": model.optimizer.state_dict(),
                    "epoch": epoch, "rng": torch.get_rng_state()},
                   checkpoint_path + ".train_state.pt")

    # Held-out test evaluation with per-class confusion matrices
    # (reference notebook cells 33-37).
    from .metrics import multilabel_confusion, three_class_accuracy
    _, _, test_set = make_epoch_sets(market, dcfg, norm_params_path)
    te = model.evaluate_model(
        chunk_batches(market, test_set, dcfg, tcfg.batch_size))
    import math

    import numpy as np
    has_rows = te[3].numel() > 0  # tiny configs can have an empty test split

    def _num(v):
        v = float(v)
        return None if math.isnan(v) else v   # keep the JSONL standard-valid

    from .metrics import fbeta_per_class
    # the notebook's held-out test metrics use beta=1 (cells 33-36), while
    # the epoch loops use beta=0.5 (biGRU_model.py:221,279)
    fb1 = ([_num(v) for v in fbeta_per_class(te[4], te[3], beta=1.0).tolist()]
           if has_rows else [None])
    test_rec = {"test_acc": _num(te[0]), "test_hamming": _num(te[1]),
                "test_fbeta": fb1,
                "test_fbeta_beta05": [_num(v) for v in np.atleast_1d(te[2])],
                "test_acc3": (float(three_class_accuracy(te[4], te[3]))
                              if has_rows else None),
                "confusion": (multilabel_confusion(te[4], te[3]).tolist()
                              if has_rows else None)}
    log(json.dumps(test_rec))
    return model, history


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=25)
    ap.add_argument("--hidden", type=int, default=32)
    ap.add_argument("--layers", type=int, default=1)
    ap.add_argument("--rows", type=int, default=3980)
    ap.add_argument("--window", type=int, default=30)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--checkpoint", default="model_params.pt")
    ap.add_argument("--resume", default=None,
                    help="checkpoint to resume from (model_params.pt format)")
    ap.add_argument("--log-file", default=None,
                    help="append per-epoch JSONL records to this file")
    ap.add_argument("--plots", default=None,
                    help="directory for learning-curve / confusion plots "
                         "(the notebook's cells 30-31 figures)")
    args = ap.parse_args()
    mcfg = ModelConfig(hidden_size=args.hidden, n_layers=args.layers,
                       spatial_dropout=False, dropout=0.5)
    dcfg = DataConfig(n_rows=args.rows, window=args.window)
    tcfg = TrainConfig(batch_size=args.batch, epochs=args.epochs,
                       device=args.device)
    if args.log_file:
        fh = open(args.log_file, "a")

        def log(line):
            print(line)
            fh.write(line + "\n")
            fh.flush()
    else:
        log = print
    records = []

    def log2(line):
        records.append(line)
        log(line)

    train(mcfg, dcfg, tcfg, checkpoint_path=args.checkpoint,
          resume=args.resume, log=log2)
    if args.plots:
        import json as _json
        import os as _os

        from .plots import confusion_heatmaps, learning_curves
        _os.makedirs(args.plots, exist_ok=True)
        recs = [_json.loads(r) for r in records]
        learning_curves(recs, _os.path.join(args.plots,
                                            "learning_curves.png"))
        test_rec = next((r for r in recs if "confusion" in r), None)
        if test_rec and test_rec["confusion"]:
            confusion_heatmaps(test_rec["confusion"],
                               _os.path.join(args.plots,
                                             "confusion_matrix.png"))


if __name__ == "__main__":
    main()
