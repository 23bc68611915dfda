from .generator import SyntheticMarket, synthetic_batch  # noqa: F401
from .dataset import (ChunkLoader, BatchLoader, TrainValTestSplit,  # noqa: F401
                      window_indices)
from .norm import save_norm_params, load_norm_params  # noqa: F401
