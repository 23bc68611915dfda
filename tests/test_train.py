"""Training driver: one short epoch end-to-end on CPU (notebook cell 29
semantics at reduced scale) + checkpoint emission."""
import os

import torch

from fmda_amd.config import DataConfig, ModelConfig, TrainConfig
from fmda_amd.train import class_weights, train


def test_class_weights():
    Y = torch.tensor([[1, 0, 0, 0], [1, 1, 0, 0], [0, 0, 1, 0],
                      [0, 0, 0, 0]], dtype=torch.float32)
    w, pw = class_weights(Y)
    assert torch.allclose(w, torch.tensor([2.0, 4.0, 4.0, 4.0]))
    # positives are clamped to >=1 before both ratios (all-zero class)
    assert torch.allclose(pw, torch.tensor([1.0, 3.0, 3.0, 3.0]))


def test_train_one_epoch(tmp_path):
    ckpt = str(tmp_path / "model_params.pt")
    npp = str(tmp_path / "norm_params")
    mcfg = ModelConfig(hidden_size=8, spatial_dropout=False, dropout=0.2)
    dcfg = DataConfig(n_rows=420, chunk_size=100, window=10, seed=3)
    tcfg = TrainConfig(batch_size=4, epochs=1)
    model, hist = train(mcfg, dcfg, tcfg, checkpoint_path=ckpt,
                        norm_params_path=npp, log=lambda s: None)
    assert len(hist) == 1
    assert 0.0 <= hist[0]["train_acc"] <= 1.0
    assert os.path.exists(ckpt) and os.path.exists(npp)
    sd = torch.load(ckpt)
    assert "gru.weight_ih_l0" in sd and "linear.bias" in sd
    # norm_params is loadable in the reference pickle format
    from fmda_amd.data import load_norm_params
    names, x_min, x_max = load_norm_params(npp)
    assert len(names) == 108


def test_loss_decreases_over_epochs(tmp_path):
    mcfg = ModelConfig(hidden_size=16, spatial_dropout=False, dropout=0.0)
    dcfg = DataConfig(n_rows=420, chunk_size=100, window=10, seed=4)
    tcfg = TrainConfig(batch_size=8, epochs=3)
    _, hist = train(mcfg, dcfg, tcfg,
                    checkpoint_path=str(tmp_path / "m.pt"),
                    norm_params_path=str(tmp_path / "np"),
                    log=lambda s: None)
    assert hist[-1]["train_loss"] < hist[0]["train_loss"]


def test_train_resume(tmp_path):
    """Mid-training resume: a second train() starting from the first's
    checkpoint continues from its weights."""
    import torch
    from fmda_amd.config import DataConfig, ModelConfig, TrainConfig
    from fmda_amd.train import train
    mcfg = ModelConfig(hidden_size=8, n_features=16, spatial_dropout=False)
    dcfg = DataConfig(n_rows=260, chunk_size=60, window=10, n_features=16)
    tcfg = TrainConfig(batch_size=4, epochs=1)
    ck = str(tmp_path / "m.pt")
    model1, _ = train(mcfg, dcfg, tcfg, checkpoint_path=ck,
                      norm_params_path=str(tmp_path / "np"), log=lambda s: None)
    model2, _ = train(mcfg, dcfg, tcfg, checkpoint_path=str(tmp_path / "m2.pt"),
                      norm_params_path=str(tmp_path / "np2"),
                      log=lambda s: None, resume=ck)
    # model2 started from model1's weights and then trained one epoch:
    # its state differs from a fresh-init run only through that lineage;
    # check the resume actually loaded (weights at epoch start equal) by
    # reloading the checkpoint and confirming it parses into the model.
    sd = torch.load(ck, weights_only=True)
    model3 = type(model2)(8, 16, 4, n_layers=1, spatial_dropout=False)
    model3.load_state_dict(sd)


def test_exact_midtraining_resume(tmp_path):
    """train(4 epochs) == train(2) + resume(2 more): the sidecar restores
    optimizer moments, epoch counter and RNG state, so the continued run
    reproduces the straight run's records exactly."""
    import json

    from fmda_amd.config import DataConfig, ModelConfig, TrainConfig
    from fmda_amd.train import train

    mcfg = ModelConfig(hidden_size=8, n_layers=1, spatial_dropout=False,
                       dropout=0.3)
    dcfg = DataConfig(n_rows=400, chunk_size=80, window=10)
    ck_a = str(tmp_path / "a.pt")
    ck_b = str(tmp_path / "b.pt")

    recs_a = []
    train(mcfg, dcfg, TrainConfig(batch_size=8, epochs=4),
          checkpoint_path=ck_a, norm_params_path=str(tmp_path / "na"),
          log=lambda l: recs_a.append(json.loads(l)))

    recs_b = []
    train(mcfg, dcfg, TrainConfig(batch_size=8, epochs=2),
          checkpoint_path=ck_b, norm_params_path=str(tmp_path / "nb"),
          log=lambda l: recs_b.append(json.loads(l)))
    train(mcfg, dcfg, TrainConfig(batch_size=8, epochs=4),
          checkpoint_path=ck_b, norm_params_path=str(tmp_path / "nb"),
          resume=ck_b, log=lambda l: recs_b.append(json.loads(l)))

    ep_a = [r for r in recs_a if "epoch" in r]
    ep_b = [r for r in recs_b if "epoch" in r]
    assert [r["epoch"] for r in ep_b] == [1, 2, 3, 4]
    for ra, rb in zip(ep_a[2:], ep_b[2:]):   # epochs 3 and 4
        assert abs(ra["train_loss"] - rb["train_loss"]) < 1e-6, (ra, rb)
        assert ra["train_acc"] == rb["train_acc"]


def test_training_plots(tmp_path, monkeypatch):
    """--plots produces the notebook's learning-curve and confusion-matrix
    figures (cells 30-31)."""
    import os
    import subprocess
    import sys

    env = dict(os.environ)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    out = subprocess.run(
        [sys.executable, "-m", "fmda_amd.train", "--epochs", "2",
         "--rows", "600", "--checkpoint", str(tmp_path / "m.pt"),
         "--plots", str(tmp_path / "plots")],
        capture_output=True, text=True, timeout=300, cwd=str(tmp_path),
        env=env)
    assert out.returncode == 0, out.stderr[-1500:]
    curves = tmp_path / "plots" / "learning_curves.png"
    conf = tmp_path / "plots" / "confusion_matrix.png"
    assert curves.exists() and curves.stat().st_size > 5000
    assert conf.exists() and conf.stat().st_size > 5000
