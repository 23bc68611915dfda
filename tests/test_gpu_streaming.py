"""GPU streaming-inference tests: hipGraph-captured batch-1 predict step."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _predictor(use_graph):
    from fmda_amd.models import BiGRU
    from fmda_amd.runtime import StreamingPredictor
    torch.manual_seed(0)
    m = BiGRU(128, 96, 4, n_layers=2, spatial_dropout=False).to(torch.bfloat16)
    return StreamingPredictor(m, torch.zeros(96), torch.ones(96), window=32,
                              device="cuda:0", dtype=torch.bfloat16,
                              use_graph=use_graph)


def test_hipgraph_predict_matches_eager():
    pg = _predictor(use_graph=True)
    pe = _predictor(use_graph=False)
    g = torch.Generator().manual_seed(7)
    rows = [torch.rand(96, generator=g) for _ in range(40)]
    for r in rows:
        pg.push_row(r.clone())
        pe.push_row(r.clone())
    for _ in range(3):
        a = torch.tensor(pg.predict_window()["probabilities"])
        b = torch.tensor(pe.predict_window()["probabilities"])
        assert torch.allclose(a, b, atol=1e-3), (a, b)
    # replay with changed input actually changes the output
    pg.push_row(torch.rand(96, generator=g) * 0.5)
    pe.push_row(pg.ring.buf[-1].clone())
    a = torch.tensor(pg.predict_window()["probabilities"])
    b = torch.tensor(pe.predict_window()["probabilities"])
    assert torch.allclose(a, b, atol=1e-3)


def test_graph_capture_happens_once():
    p = _predictor(use_graph=True)
    for _ in range(32):
        p.push_row(torch.rand(96))
    p.predict_window()
    g1 = p._graph
    p.predict_window()
    assert p._graph is g1 and g1 is not None
