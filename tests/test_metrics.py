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


def test_three_class_accuracy():
    from fmda_amd.metrics import three_class_accuracy
    import torch
    # rows: up-match, down-match, stall-match, up-vs-down, conflict->stall
    t = torch.tensor([[1, 0, 0, 0], [0, 0, 1, 0], [0, 0, 0, 0],
                      [0, 1, 0, 0], [1, 0, 1, 0]])
    p = torch.tensor([[0, 1, 0, 0],   # up2 vs up1: still up -> match
                      [0, 0, 0, 1],   # down -> match
                      [1, 0, 1, 0],   # conflict collapses to stall -> match
                      [0, 0, 1, 0],   # up vs down -> miss
                      [0, 0, 0, 0]])  # stall vs conflict-stall -> match
    assert abs(float(three_class_accuracy(t, p)) - 0.8) < 1e-6


def test_three_class_all_match_is_one():
    from fmda_amd.metrics import three_class_accuracy
    import torch
    t = (torch.rand(64, 4) < 0.3).long()
    assert float(three_class_accuracy(t, t)) == 1.0


def test_multilabel_confusion_matches_sklearn():
    from fmda_amd.metrics import multilabel_confusion
    from sklearn.metrics import multilabel_confusion_matrix
    import torch
    g = torch.Generator().manual_seed(7)
    t = (torch.rand(50, 4, generator=g) < 0.3).long()
    p = (torch.rand(50, 4, generator=g) < 0.3).long()
    ours = multilabel_confusion(t, p).numpy()
    ref = multilabel_confusion_matrix(t.numpy(), p.numpy())
    assert (ours == ref).all()


def test_metrics_match_sklearn_property():
    """Property test vs sklearn on random multilabel matrices, including
    degenerate columns (all-positive / all-negative)."""
    import torch
    from hypothesis import given, settings, strategies as st
    from sklearn.metrics import (accuracy_score, fbeta_score, hamming_loss,
                                 multilabel_confusion_matrix)

    from fmda_amd.metrics import (fbeta_per_class, hamming,
                                  multilabel_confusion, subset_accuracy)

    @settings(max_examples=30, deadline=None)
    # c >= 2: sklearn reinterprets an (n, 1) indicator matrix as BINARY
    # labels (one 2-class problem), not one-column multilabel — different
    # semantics than the per-column formulas (the reference always has 4)
    @given(st.integers(1, 40), st.integers(2, 6), st.integers(0, 2 ** 31 - 1),
           st.sampled_from([0.5, 1.0, 2.0]))
    def check(n, c, seed, beta):
        g = torch.Generator().manual_seed(seed)
        t = (torch.rand(n, c, generator=g) < 0.4).long()
        p = (torch.rand(n, c, generator=g) < 0.4).long()
        assert abs(float(subset_accuracy(t, p))
                   - accuracy_score(t.numpy(), p.numpy())) < 1e-6
        assert abs(float(hamming(t, p))
                   - hamming_loss(t.numpy(), p.numpy())) < 1e-6
        ours = fbeta_per_class(t, p, beta=beta).numpy()
        ref = fbeta_score(t.numpy(), p.numpy(), beta=beta, average=None,
                          zero_division=0)
        assert abs(ours - ref).max() < 1e-6
        assert (multilabel_confusion(t, p).numpy()
                == multilabel_confusion_matrix(t.numpy(), p.numpy())).all()

    check()
