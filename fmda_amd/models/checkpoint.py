"""Checkpoint I/O in the reference `model_params.pt` format.

The reference saves `torch.save(model.state_dict(), 'model_params.pt')`
(training notebook cell 39) and reloads it with
`model.load_state_dict(torch.load(...))` (predict.py:104). Because BiGRU's
parameter container uses the same module/parameter names as the reference
(`gru.weight_ih_l0`, ..., `linear.bias`), plain torch serialization is
byte-format compatible in both directions.
"""
from typing import Optional

import torch

from .bigru import BiGRU


def save_checkpoint(model: BiGRU, path: str) -> None:
    """Write the state_dict in the reference `model_params.pt` format.
    Uses the native C++ zip/pickle writer (ops/csrc/checkpoint.cpp) when
    the extension is importable; the output is torch.load-compatible
    either way."""
    sd = {k: v for k, v in model.state_dict().items()
          if not k.startswith("loss_fn.")}
    # (add_loss_fn attaches the loss module, whose weight/pos_weight
    # buffers would otherwise leak into the artifact; the reference
    # model_params.pt holds only gru.* / linear.* entries.)
    try:
        from ..ops import _fmda_hip
        _fmda_hip.save_state_dict_native(path, list(sd.keys()),
                                         [t.cpu() for t in sd.values()])
    except ImportError:
        torch.save(sd, path)


def load_checkpoint(path: str, model: Optional[BiGRU] = None,
                    map_location="cpu") -> BiGRU:
    """Load a `model_params.pt` state_dict. When `model` is None, the
    architecture (hidden size, features, classes, layers, directions) is
    inferred from the checkpoint shapes."""
    sd = torch.load(path, map_location=map_location)
    if model is None:
        w_ih = sd["gru.weight_ih_l0"]
        hidden = w_ih.shape[0] // 3
        n_features = w_ih.shape[1]
        output_size = sd["linear.weight"].shape[0]
        bidirectional = "gru.weight_ih_l0_reverse" in sd
        n_layers = 1
        while f"gru.weight_ih_l{n_layers}" in sd:
            n_layers += 1
        model = BiGRU(hidden, n_features, output_size, n_layers=n_layers,
                      spatial_dropout=False, bidirectional=bidirectional)
    model.load_state_dict(sd)
    return model
