import pickle

import torch

from fmda_amd.features import (FEATURE_NAMES, N_FEATURES, REFERENCE_RANGES,
                               TARGET_NAMES)


def test_registry_is_108_columns():
    assert N_FEATURES == 108
    assert len(FEATURE_NAMES) == 108
    assert TARGET_NAMES == ["up1", "up2", "down1", "down2"]


def test_registry_matches_reference_norm_params_order():
    """Column order must equal the reference norm_params artifact
    (predict.py:110-122 relies on dict insertion order)."""
    with open("/root/reference/norm_params", "rb") as f:
        ref = pickle.load(f)
    assert list(ref.keys()) == FEATURE_NAMES
    for name in FEATURE_NAMES:
        lo, hi = REFERENCE_RANGES[name]
        rmin, rmax = float(ref[name]["MIN"]), float(ref[name]["MAX"])
        assert abs(lo - rmin) <= 1e-6 * max(1.0, abs(rmin))
        assert abs(hi - rmax) <= 1e-6 * max(1.0, abs(rmax))


def test_known_indices():
    assert FEATURE_NAMES[0] == "sd.bid_0_size"
    assert FEATURE_NAMES[-1] == "pc.price_change"
    assert FEATURE_NAMES.index("sd.VIX") == 41
