import numpy as np
import torch
import torch.nn as nn

from fmda_amd.models import BiGRU, load_checkpoint, save_checkpoint
from fmda_amd.ops.gru_ref import bigru_forward, pooled_head

REF_CKPT = "/root/reference/model_params.pt"


def test_checkpoint_roundtrip_with_reference_artifact(tmp_path):
    """The real reference model_params.pt must load, run, save and reload
    byte-compatibly."""
    m = load_checkpoint(REF_CKPT)
    assert m.hidden_size == 8 and m.n_features == 108
    assert m.output_size == 4 and m.n_layers == 1 and m.bidirectional
    m.eval()
    x = torch.randn(2, 5, 108, generator=torch.Generator().manual_seed(0))
    with torch.no_grad():
        out1 = m(x)
    p = str(tmp_path / "model_params.pt")
    save_checkpoint(m, p)
    sd_ref = torch.load(REF_CKPT)
    sd_new = torch.load(p)
    assert set(sd_ref.keys()) == set(sd_new.keys())
    for k in sd_ref:
        assert sd_ref[k].shape == sd_new[k].shape
        assert torch.equal(sd_ref[k], sd_new[k])
    m2 = load_checkpoint(p)
    m2.eval()
    with torch.no_grad():
        out2 = m2(x)
    assert torch.allclose(out1, out2)


def test_golden_gru_matches_nn_gru():
    """ops.gru_ref (the HIP-kernel oracle) must match nn.GRU exactly."""
    torch.manual_seed(3)
    for n_layers, bidir, H, F, T, B in [(1, True, 8, 12, 7, 3),
                                        (2, True, 16, 10, 5, 2),
                                        (1, False, 8, 6, 4, 2)]:
        gru = nn.GRU(F, H, num_layers=n_layers, batch_first=True,
                     bidirectional=bidir)
        x = torch.randn(B, T, F)
        out_ref, h_ref = gru(x)
        out_g, h_g = bigru_forward(x, list(gru.parameters()), n_layers, bidir)
        assert torch.allclose(out_ref, out_g, atol=1e-5), (n_layers, bidir)
        assert torch.allclose(h_ref, h_g, atol=1e-5)


def test_pooled_head_matches_model_forward():
    torch.manual_seed(4)
    m = BiGRU(16, 10, 4, spatial_dropout=False, dropout=0.0)
    m.eval()
    x = torch.randn(3, 9, 10)
    with torch.no_grad():
        logits = m(x)
        out, h_n = m.gru(x)
        ref = pooled_head(out, h_n, m.n_layers, m.n_directions, m.hidden_size,
                          m.linear.weight, m.linear.bias)
    assert torch.allclose(logits, ref, atol=1e-6)


def test_train_and_evaluate_api():
    """train_model/evaluate_model semantics of biGRU_model.py:162-286."""
    torch.manual_seed(5)
    m = BiGRU(8, 6, 4, spatial_dropout=False, dropout=0.1)
    m.add_loss_fn(nn.BCEWithLogitsLoss())
    m.add_optimizer(torch.optim.Adam(m.parameters(), lr=1e-3))
    m.add_device(torch.device("cpu"))

    def batches():
        g = torch.Generator().manual_seed(0)
        for _ in range(6):
            x = torch.randn(4, 10, 6, generator=g)
            y = (torch.rand(4, 1, 4, generator=g) < 0.3).float()
            yield x, y

    acc, ham, loss, fbeta = m.train_model(batches())
    assert 0.0 <= acc <= 1.0 and 0.0 <= ham <= 1.0
    assert np.asarray(fbeta).shape == (4,)
    acc2, ham2, fbeta2, pred_tot, tgt_tot = m.evaluate_model(batches())
    assert pred_tot.shape == (24, 4) and tgt_tot.shape == (24, 4)


def test_spatial_dropout_path():
    torch.manual_seed(6)
    m = BiGRU(8, 6, 4, spatial_dropout=True, dropout=0.5)
    m.train()
    x = torch.randn(2, 10, 6)
    out = m(x)
    assert out.shape == (2, 4)
    m.eval()
    with torch.no_grad():
        o1, o2 = m(x), m(x)
    assert torch.allclose(o1, o2)  # dropout off in eval
