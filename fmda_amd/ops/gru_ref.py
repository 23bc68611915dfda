"""Golden pure-PyTorch GRU + pooled-head reference ops (fp32).

This is the oracle every HIP kernel is validated against, and the CPU
execution path of the model. Gate math follows the PyTorch packed (r, z, n)
convention so weights are interchangeable with `nn.GRU` and with the
reference checkpoint `model_params.pt` (keys/shapes documented in SURVEY.md
2.1 #13; reference biGRU_model.py:54-56).

    r_t = sigmoid(W_ir x_t + b_ir + W_hr h_{t-1} + b_hr)
    z_t = sigmoid(W_iz x_t + b_iz + W_hz h_{t-1} + b_hz)
    n_t = tanh(W_in x_t + b_in + r_t * (W_hn h_{t-1} + b_hn))
    h_t = (1 - z_t) * n_t + z_t * h_{t-1}
"""
from typing import List, Optional, Tuple

import torch


def gru_cell(x_t: torch.Tensor, h: torch.Tensor, w_ih: torch.Tensor,
             w_hh: torch.Tensor, b_ih: torch.Tensor,
             b_hh: torch.Tensor) -> torch.Tensor:
    gi = x_t @ w_ih.t() + b_ih
    gh = h @ w_hh.t() + b_hh
    i_r, i_z, i_n = gi.chunk(3, dim=-1)
    h_r, h_z, h_n = gh.chunk(3, dim=-1)
    r = torch.sigmoid(i_r + h_r)
    z = torch.sigmoid(i_z + h_z)
    n = torch.tanh(i_n + r * h_n)
    return (1.0 - z) * n + z * h


def gru_layer_direction(x: torch.Tensor, w_ih: torch.Tensor,
                        w_hh: torch.Tensor, b_ih: torch.Tensor,
                        b_hh: torch.Tensor, reverse: bool,
                        h0: Optional[torch.Tensor] = None) -> Tuple[torch.Tensor, torch.Tensor]:
    """One direction of one layer. x: (B, T, F) -> out (B, T, H), h_T (B, H)."""
    B, T, _ = x.shape
    H = w_hh.shape[1]
    h = torch.zeros(B, H, dtype=x.dtype, device=x.device) if h0 is None else h0
    outs = []
    steps = range(T - 1, -1, -1) if reverse else range(T)
    for t in steps:
        h = gru_cell(x[:, t], h, w_ih, w_hh, b_ih, b_hh)
        outs.append(h)
    if reverse:
        outs.reverse()
    return torch.stack(outs, dim=1), h


def bigru_forward(x: torch.Tensor, flat_weights: List[torch.Tensor],
                  n_layers: int, bidirectional: bool,
                  dropout_p: float = 0.0, training: bool = False
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stacked (bi)GRU matching nn.GRU(batch_first=True) semantics.

    flat_weights: per layer, per direction: [w_ih, w_hh, b_ih, b_hh] in
    nn.GRU._flat_weights order. Returns (out (B,T,D*H), h_n (L*D, B, H)).
    Inter-layer dropout is applied between stacked layers when n_layers > 1
    (biGRU_model.py:54-56).
    """
    D = 2 if bidirectional else 1
    h_n = []
    inp = x
    for layer in range(n_layers):
        outs = []
        for direction in range(D):
            base = (layer * D + direction) * 4
            w_ih, w_hh, b_ih, b_hh = flat_weights[base:base + 4]
            out, h_T = gru_layer_direction(inp, w_ih, w_hh, b_ih, b_hh,
                                           reverse=(direction == 1))
            outs.append(out)
            h_n.append(h_T)
        inp = torch.cat(outs, dim=-1) if D == 2 else outs[0]
        if training and dropout_p > 0 and layer < n_layers - 1:
            inp = torch.nn.functional.dropout(inp, p=dropout_p, training=True)
    return inp, torch.stack(h_n, dim=0)


def pooled_head(gru_out: torch.Tensor, h_n: torch.Tensor, n_layers: int,
                n_directions: int, hidden_size: int,
                linear_weight: torch.Tensor,
                linear_bias: torch.Tensor) -> torch.Tensor:
    """The reference's 3-way pooling head (biGRU_model.py:108-137):
    concat[sum-of-directions last hidden, max-pool over T, avg-pool over T]
    of the direction-summed gru_out, then Linear(3H -> C)."""
    B, T, _ = gru_out.shape
    hidden = h_n.view(n_layers, n_directions, B, hidden_size)
    last_hidden = hidden[-1].sum(dim=0)                      # (B, H)
    if n_directions == 2:
        summed = gru_out[:, :, :hidden_size] + gru_out[:, :, hidden_size:]
    else:
        summed = gru_out
    max_pool = summed.max(dim=1).values                      # (B, H)
    avg_pool = summed.sum(dim=1) / float(T)                  # (B, H)
    concat = torch.cat([last_hidden, max_pool, avg_pool], dim=1)
    return concat @ linear_weight.t() + linear_bias
