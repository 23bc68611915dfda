import numpy as np
import torch
from sklearn.metrics import accuracy_score, fbeta_score, hamming_loss

from fmda_amd.metrics import fbeta_per_class, hamming, subset_accuracy


def _rand_labels(n=64, c=4, seed=0):
    g = torch.Generator().manual_seed(seed)
    t = (torch.rand(n, c, generator=g) < 0.4).long()
    p = (torch.rand(n, c, generator=g) < 0.4).long()
    return t, p


def test_matches_sklearn():
    for seed in range(5):
        t, p = _rand_labels(seed=seed)
        assert abs(float(subset_accuracy(t, p)) - accuracy_score(t, p)) < 1e-9
        assert abs(float(hamming(t, p)) - hamming_loss(t, p)) < 1e-9
        ours = fbeta_per_class(t, p, beta=0.5).numpy()
        ref = fbeta_score(t, p, beta=0.5, average=None)
        assert np.allclose(ours, ref, atol=1e-7)


def test_zero_division_matches_sklearn():
    t = torch.zeros(8, 4).long()
    p = torch.zeros(8, 4).long()
    ref = fbeta_score(t, p, beta=0.5, average=None, zero_division=0)
    ours = fbeta_per_class(t, p, beta=0.5).numpy()
    assert np.allclose(ours, ref)
