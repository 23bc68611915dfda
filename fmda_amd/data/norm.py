"""norm_params pickle compatibility.

The reference saves normalization parameters as a plain pickle of an ordered
dict: feature name -> {'MIN': 0-d torch.Tensor, 'MAX': 0-d torch.Tensor}
(reference sql_pytorch_dataloader.py:146-153), reloaded by predict.py:110-122.
This module reads and writes that exact byte format so checkpoints and
scaling tables interoperate with the reference deployment.
"""
import pickle
from typing import Dict, List, Tuple

import torch


def save_norm_params(path: str, names: List[str], x_min: torch.Tensor,
                     x_max: torch.Tensor) -> None:
    params: Dict[str, Dict[str, torch.Tensor]] = {}
    for i, name in enumerate(names):
        params[name] = {"MIN": x_min[i].clone(), "MAX": x_max[i].clone()}
    with open(path, "wb") as f:
        pickle.dump(params, f)


def load_norm_params(path: str) -> Tuple[List[str], torch.Tensor, torch.Tensor]:
    """Returns (names, x_min (F,), x_max (F,)) like predict.py:110-122."""
    with open(path, "rb") as f:
        params = pickle.load(f)
    names = list(params.keys())
    x_min = torch.tensor([float(params[n]["MIN"]) for n in names])
    x_max = torch.tensor([float(params[n]["MAX"]) for n in names])
    return names, x_min, x_max
