"""Byte-compatibility against the REAL reference artifacts.

The reference deployment's checkpoint (`model_params.pt`, torch zip format,
H=8 F=108 C=4 bidirectional — SURVEY.md 2.1 #13) and normalization table
(`norm_params`, plain pickle of 108 feature ranges, #14) must load directly
into this framework. Skipped when the reference mount is absent (GPU boxes).
"""
import os

import pytest
import torch

REF = "/root/reference"
pytestmark = pytest.mark.skipif(
    not os.path.exists(os.path.join(REF, "model_params.pt")),
    reason="reference artifacts not mounted")


def test_reference_checkpoint_loads_into_bigru():
    from fmda_amd.models.checkpoint import load_checkpoint
    model = load_checkpoint(os.path.join(REF, "model_params.pt"))
    assert model.hidden_size == 8 and model.n_features == 108
    assert model.output_size == 4 and model.n_directions == 2
    model.eval()
    torch.manual_seed(0)
    x = torch.rand(1, 5, 108)  # predict.py window=5
    logits = model(x)
    assert logits.shape == (1, 4) and torch.isfinite(logits).all()


def test_reference_checkpoint_native_reader_matches_torch():
    from fmda_amd.ops import _fmda_hip
    p = os.path.join(REF, "model_params.pt")
    sd_torch = torch.load(p, weights_only=True)
    sd_native = dict(_fmda_hip.load_state_dict_native(p))
    assert set(sd_native.keys()) == set(sd_torch.keys())
    for k in sd_torch:
        assert torch.equal(sd_torch[k], sd_native[k]), k


def test_reference_norm_params_loads():
    from fmda_amd.data.norm import load_norm_params
    from fmda_amd.features import FEATURE_NAMES
    names, x_min, x_max = load_norm_params(os.path.join(REF, "norm_params"))
    assert len(names) == 108
    assert names == list(FEATURE_NAMES)  # canonical registry order matches
    assert x_min.shape == (108,)
    # note: the reference artifact itself contains one feature whose
    # recorded MIN exceeds its MAX (a quirk of its last-chunk capture);
    # loaded verbatim, not "fixed".
    assert (x_max >= x_min).float().mean() > 0.9


def test_streaming_predictor_runs_reference_model():
    """predict.py semantics end to end on the real artifacts: window 5,
    min-max normalize, forward, sigmoid threshold (predict.py:71-197)."""
    from fmda_amd.data.norm import load_norm_params
    from fmda_amd.models.checkpoint import load_checkpoint
    from fmda_amd.runtime import StreamingPredictor
    model = load_checkpoint(os.path.join(REF, "model_params.pt"))
    _, x_min, x_max = load_norm_params(os.path.join(REF, "norm_params"))
    pred = StreamingPredictor(model, x_min, x_max, window=5, device="cpu")
    g = torch.Generator().manual_seed(1)
    for _ in range(5):
        pred.push_row(x_min + (x_max - x_min) * torch.rand(108, generator=g))
    pred_dict = pred.predict_window()
    assert len(pred_dict["probabilities"]) == 4
    assert all(0.0 <= p <= 1.0 for p in pred_dict["probabilities"])


def test_reference_ranges_match_real_artifact():
    """feature_ranges.REFERENCE_RANGES (the generator's value bounds) must
    equal the real norm_params artifact's recorded MIN/MAX."""
    import torch

    from fmda_amd.data.norm import load_norm_params
    from fmda_amd.features import REFERENCE_RANGES
    names, x_min, x_max = load_norm_params(os.path.join(REF, "norm_params"))
    ours_lo = torch.tensor([REFERENCE_RANGES[n][0] for n in names])
    ours_hi = torch.tensor([REFERENCE_RANGES[n][1] for n in names])
    torch.testing.assert_close(ours_lo, x_min, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(ours_hi, x_max, rtol=1e-5, atol=1e-5)
