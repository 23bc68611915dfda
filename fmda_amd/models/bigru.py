"""BiGRU model with the reference-compatible public API.

The class signature, parameter names/shapes, checkpoint format and the
train/evaluate loop semantics match the reference `BiGRU`
(biGRU_model.py:8-286): state_dict keys are `gru.weight_ih_l{k}[...]` /
`linear.weight` / `linear.bias`, so the reference `model_params.pt` loads
directly.

Execution paths:
- CPU: PyTorch's own GRU kernels (ATen) — also the golden reference.
- CUDA (ROCm/MI355X): the fmda_amd HIP engine — time-batched input
  projections on MFMA (rocBLAS GEMM) + hand-written persistent CDNA4
  recurrence kernels; no MIOpen RNN, no cuDNN. The HIP extension is
  REQUIRED on GPU: the model raises rather than falling back silently.
"""
from typing import Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..metrics import batch_metrics


class BiGRU(nn.Module):
    """BiDirectional GRU model (API of reference biGRU_model.py:8-61).

    Parameters
    ----------
    hidden_size: number of features in the hidden state.
    n_features: number of input features.
    output_size: number of classes.
    n_layers: number of stacked recurrent layers.
    clip: max norm of the gradients.
    dropout: probability of an element to be zeroed.
    spatial_dropout: whether to use spatial (per-feature-channel) dropout.
    bidirectional: whether to use the bidirectional GRU.
    """

    def __init__(self, hidden_size, n_features, output_size, n_layers=1,
                 clip=50, dropout=0.2, spatial_dropout=True,
                 bidirectional=True):
        super().__init__()
        self.hidden_size = hidden_size
        self.n_features = n_features
        self.output_size = output_size
        self.n_layers = n_layers
        self.clip = clip
        self.dropout_p = dropout
        self.spatial_dropout = spatial_dropout
        self.bidirectional = bidirectional
        self.n_directions = 2 if bidirectional else 1

        self.dropout = nn.Dropout(self.dropout_p)
        if self.spatial_dropout:
            self.spatial_dropout1d = nn.Dropout2d(self.dropout_p)

        # nn.GRU is the parameter container (state_dict-compatible with the
        # reference) and the CPU execution path. Its forward is NEVER called
        # on a GPU tensor: the CUDA path runs the fmda_amd HIP engine with
        # these same weights.
        self.gru = nn.GRU(self.n_features, self.hidden_size,
                          num_layers=self.n_layers,
                          dropout=(0 if n_layers == 1 else self.dropout_p),
                          batch_first=True, bidirectional=self.bidirectional)

        # Linear head input is hidden_size * 3: concat of summed-last-hidden,
        # max pooling and avg pooling (biGRU_model.py:58-60).
        self.linear = nn.Linear(self.hidden_size * 3, self.output_size)

    # ------------------------------------------------------------------ #

    def forward(self, input_seq: torch.Tensor,
                hidden: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Forward pass; returns (B, output_size) logits
        (semantics of biGRU_model.py:63-138)."""
        concat_out = self.forward_features(input_seq, hidden)
        if concat_out.dtype != self.linear.weight.dtype:
            # bf16 compute path with fp32 master weights: cast the head.
            return F.linear(concat_out, self.linear.weight.to(concat_out.dtype),
                            self.linear.bias.to(concat_out.dtype))
        return self.linear(concat_out)

    def forward_features(self, input_seq: torch.Tensor,
                         hidden: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Everything before the linear head: returns the pooled concat
        (B, 3H) feature vector — the input of the reference's linear head
        (biGRU_model.py:133-137). Used by the fused head+loss kernel."""
        batch_size = input_seq.size(0)
        input_length = input_seq.size(1)

        if self.spatial_dropout:
            if (self.training and input_seq.is_cuda
                    and input_seq.dtype == torch.bfloat16
                    and self.dropout_p > 0):
                # channel-mask HIP kernel: Dropout2d semantics without the
                # permute(0,2,1) round trips (biGRU_model.py:87-94)
                from ..ops.interface import fused_spatial_dropout
                input_seq = fused_spatial_dropout(input_seq, self.dropout_p)
            else:
                # Dropout2d over (B, F, T): zeroes whole feature channels.
                input_seq = input_seq.permute(0, 2, 1)
                input_seq = self.spatial_dropout1d(input_seq)
                input_seq = input_seq.permute(0, 2, 1)
        elif (self.training and input_seq.is_cuda
              and input_seq.dtype == torch.bfloat16 and self.dropout_p > 0):
            from ..ops.interface import fused_dropout
            input_seq = fused_dropout(input_seq, self.dropout_p)
        else:
            input_seq = self.dropout(input_seq)

        if input_seq.is_cuda:
            gru_out, h_n = self._gru_hip(input_seq, hidden)
        else:
            gru_out, h_n = self.gru(input_seq, hidden)

        # 3-way pooling head (biGRU_model.py:108-137).
        hidden_v = h_n.view(self.n_layers, self.n_directions, batch_size,
                            self.hidden_size)
        last_hidden = hidden_v[-1].sum(dim=0)

        if input_seq.is_cuda and self.hidden_size % 2 == 0:
            # fused direction-sum + max/avg pooling HIP kernel (pairs
            # columns; odd H takes the eager path below)
            from ..ops.interface import dirsum_pool
            max_pool, avg_pool = dirsum_pool(gru_out, self.n_directions)
        else:
            if self.bidirectional:
                gru_out = (gru_out[:, :, :self.hidden_size]
                           + gru_out[:, :, self.hidden_size:])
            max_pool = gru_out.max(dim=1).values
            avg_pool = gru_out.sum(dim=1) / float(input_length)

        return torch.cat([last_hidden, max_pool, avg_pool], dim=1)

    def _gru_hip(self, x: torch.Tensor, hidden: Optional[torch.Tensor]):
        """CUDA path: MFMA input projections + persistent HIP recurrence."""
        from ..ops.interface import bigru_stack
        return bigru_stack(x, self.gru, self.n_layers, self.bidirectional,
                           self.dropout_p, self.training, hidden)

    # ---------------------- reference training API --------------------- #

    def add_loss_fn(self, loss_fn):
        """Add loss function to the model (biGRU_model.py:141-145)."""
        self.loss_fn = loss_fn

    def add_optimizer(self, optimizer):
        """Add optimizer to the model (biGRU_model.py:148-152)."""
        self.optimizer = optimizer

    def add_device(self, device=torch.device('cpu')):
        """Specify the device (biGRU_model.py:155-159)."""
        self.device = device

    def train_model(self, train_iterator):
        """One training epoch; returns (mean accuracy, mean Hamming loss,
        mean loss, mean per-class fbeta[beta=0.5]) like
        biGRU_model.py:162-224."""
        self.train()
        losses, accuracies, hamm_losses, fbetas_list = [], [], [], []

        for input_seq, target in train_iterator:
            target = target.squeeze(1)
            input_seq = input_seq.to(self.device)
            target = target.to(self.device)

            self.optimizer.zero_grad()
            pred = self.forward(input_seq)
            loss = self.loss_fn(pred, target)
            loss.backward()
            losses.append(loss.detach().cpu().numpy())

            nn.utils.clip_grad_norm_(self.parameters(), self.clip)
            self.optimizer.step()

            pred = torch.sigmoid(pred) > 0.5
            acc, ham, fbeta = batch_metrics(target.detach(), pred.detach(),
                                            beta=0.5)
            accuracies.append(float(acc))
            hamm_losses.append(float(ham))
            fbetas_list.append(fbeta.cpu().numpy())

        return (np.mean(accuracies), np.mean(hamm_losses), np.mean(losses),
                np.mean(fbetas_list, axis=0))

    def evaluate_model(self, eval_iterator):
        """One evaluation epoch; returns (mean accuracy, mean Hamming loss,
        mean per-class fbeta, pred_total, target_total) like
        biGRU_model.py:227-286."""
        self.eval()
        accuracies, hamm_losses, fbetas_list = [], [], []
        pred_total = torch.LongTensor()
        target_total = torch.LongTensor()

        with torch.no_grad():
            for input_seq, target in eval_iterator:
                target = target.squeeze(1)
                input_seq = input_seq.to(self.device)
                target = target.to(self.device)

                pred = self.forward(input_seq)
                pred = torch.sigmoid(pred) > 0.5

                acc, ham, fbeta = batch_metrics(target, pred, beta=0.5)
                accuracies.append(float(acc))
                hamm_losses.append(float(ham))
                fbetas_list.append(fbeta.cpu().numpy())

                pred_total = torch.cat(
                    [pred_total, pred.cpu().type(torch.LongTensor)], dim=0)
                target_total = torch.cat(
                    [target_total, target.cpu().type(torch.LongTensor)], dim=0)

        return (np.mean(accuracies), np.mean(hamm_losses),
                np.mean(fbetas_list, axis=0), pred_total, target_total)
