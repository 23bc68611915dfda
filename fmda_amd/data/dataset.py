"""Chunked dataset with sliding windows and train/val/test splitting.

Mirrors the semantics of the reference SQL-backed loaders
(sql_pytorch_dataloader.py) on top of the in-memory synthetic market table:

- `ChunkLoader` == MySQLChunkLoader (:21): chunk index ranges with window-1
  overlap (:72-78), per-chunk MIN/MAX (:96-115), MIN==MAX epsilon fix
  (:107-113), shared order-book-size min/max per side (:119-144), and saving
  the last chunk's normalization parameters in the `norm_params` pickle
  format (:146-153).
- `BatchLoader` == MySQLBatchLoader (:162): chunk rows normalized by the
  chunk's min/max (:239), served as stride-1 sliding windows whose label is
  the last row of the window (:241-245).
- `TrainValTestSplit` (:251): chunk-granularity split with the same
  int-truncation + 1 sizing (:299-320).
- `window_indices` (:8): the same width-n sliding tuple generator.

The reference's 1-based SQL IDs are kept: chunk index ranges are 1-based row
ids into the market table, converted to 0-based at fetch time.
"""
from itertools import islice
from typing import Iterator, List, Tuple

import torch
from torch.utils.data import Dataset

from ..features import ASK_SIZE_IDX, BID_SIZE_IDX, FEATURE_NAMES
from .norm import save_norm_params


def window_indices(seq, n: int = 2) -> Iterator[tuple]:
    """Sliding window (width n, stride 1) over an iterable
    (reference sql_pytorch_dataloader.py:8-18)."""
    it = iter(seq)
    result = tuple(islice(it, n))
    if len(result) == n:
        yield result
    for elem in it:
        result = result[1:] + (elem,)
        yield result


class ChunkLoader(Dataset):
    """Chunk index/normalization catalogue over the synthetic market table.

    Parameters
    ----------
    X: (N, F) float32 market table (1-based ids are row+1).
    chunk_size, window: as in the reference.
    norm_params_path: where to save the last chunk's normalization pickle
        (None to skip saving).
    """

    def __init__(self, X: torch.Tensor, chunk_size: int, window: int,
                 norm_params_path: str = None,
                 feature_names: List[str] = None):
        db_length = X.shape[0]
        self.X = X
        self.window = window
        self.num_chunks = db_length // chunk_size
        self.chunk_indices: List[range] = []

        # Same ranges as reference :72-78 (1-based ids, window-1 overlap).
        for chunk in range(self.num_chunks + 1):
            if chunk == 0:
                self.chunk_indices.append(range(window, chunk_size))
            elif chunk < (db_length // chunk_size):
                self.chunk_indices.append(
                    range(chunk_size * chunk - window + 1, chunk_size * (chunk + 1)))
            else:
                self.chunk_indices.append(
                    range(chunk_size * chunk - window + 1, db_length + 1))

        names = feature_names if feature_names is not None else FEATURE_NAMES[:X.shape[1]]

        # Per-chunk MIN/MAX (reference :96-115).
        self.norm_params: List[Tuple[torch.Tensor, torch.Tensor]] = []
        for chunk in range(self.num_chunks + 1):
            rows = self._rows(chunk)
            x_min = rows.min(dim=0).values.clone().unsqueeze(0)
            x_max = rows.max(dim=0).values.clone().unsqueeze(0)
            # MIN == MAX epsilon fix (reference :107-113)
            eq = x_min[0] == x_max[0]
            nonzero = eq & (x_max[0] != 0)
            zero = eq & (x_max[0] == 0)
            x_max[0][nonzero] += x_max[0][nonzero] * 0.001
            x_max[0][zero] += 0.001
            self.norm_params.append((x_min, x_max))

        # Order-book size levels share one min/max per side (reference :119-144).
        if "sd.bid_0_size" in names:
            bid_idx = [i for i in BID_SIZE_IDX if i < X.shape[1]]
            ask_idx = [i for i in ASK_SIZE_IDX if i < X.shape[1]]
            for x_min, x_max in self.norm_params:
                if ask_idx:
                    x_min[0][ask_idx] = x_min[0][ask_idx].min()
                    x_max[0][ask_idx] = x_max[0][ask_idx].max()
                if bid_idx:
                    x_min[0][bid_idx] = x_min[0][bid_idx].min()
                    x_max[0][bid_idx] = x_max[0][bid_idx].max()

        if norm_params_path is not None:
            save_norm_params(norm_params_path, names,
                             self.norm_params[-1][0][0], self.norm_params[-1][1][0])

    def _rows(self, chunk: int) -> torch.Tensor:
        ids = torch.tensor(list(self.chunk_indices[chunk]), dtype=torch.long)
        return self.X[ids - 1]

    def __getitem__(self, idx):
        return tuple(self.chunk_indices[idx]), self.norm_params[idx]

    def __len__(self):
        return self.num_chunks + 1


class BatchLoader(Dataset):
    """Normalized sliding windows over one chunk
    (reference MySQLBatchLoader, sql_pytorch_dataloader.py:162-248).

    Unlike the reference (whose __getitem__ draws from a shared generator and
    whose __len__ over-reports by window-1), windows are materialized as
    index tuples so that len() and random access are exact; iteration order
    with a sequential DataLoader is identical.
    """

    def __init__(self, indices, norm_params, X: torch.Tensor, Y: torch.Tensor,
                 window: int):
        ids = torch.tensor(list(indices), dtype=torch.long)
        x = X[ids - 1]
        y = Y[ids - 1]
        # NULL -> 0 like IFNULL(field, 0) (reference :219)
        x = torch.nan_to_num(x, nan=0.0)
        x_min, x_max = norm_params[0][0], norm_params[1][0]
        self.x = (x - x_min) / (x_max - x_min)
        self.y = torch.nan_to_num(y, nan=0.0)
        self.windows = list(window_indices(range(len(ids)), window))

    def __getitem__(self, idx):
        w = self.windows[idx]
        return self.x[list(w)], self.y[[w[-1]]]

    def __len__(self):
        return len(self.windows)


class TrainValTestSplit:
    """Chunk-granularity split (reference sql_pytorch_dataloader.py:251-320)."""

    def __init__(self, dataset: ChunkLoader, val_size: float = 0.1,
                 test_size: float = 0.1):
        assert (val_size + test_size) < 1, \
            'Validation size and test size sum is greater or equal 1'
        assert val_size >= 0 and test_size >= 0, 'Negative size is not accepted'
        self.dataset = dataset
        self.train_size = 1 - val_size - test_size
        self.val_size = val_size
        self.test_size = test_size
        self.dataset_len = len(dataset)

    def _slice(self, start: int, end: int):
        inds = [self.dataset[i][0] for i in range(start, end)]
        norms = [self.dataset[i][1] for i in range(start, end)]
        return zip(inds, norms)

    def get_train(self):
        self.train_end_idx = int(self.train_size * self.dataset_len)
        return self._slice(0, self.train_end_idx)

    def get_val(self):
        self.val_start_idx = self.train_end_idx
        self.val_end_idx = self.val_start_idx + int(self.val_size * self.dataset_len) + 1
        return self._slice(self.val_start_idx, min(self.val_end_idx, self.dataset_len))

    def get_test(self):
        self.test_start_idx = self.val_end_idx
        self.test_end_idx = self.test_start_idx + int(self.test_size * self.dataset_len) + 1
        return self._slice(self.test_start_idx, min(self.test_end_idx, self.dataset_len))

    def get_sets(self):
        return self.get_train(), self.get_val(), self.get_test()
