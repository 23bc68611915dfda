"""fmda_amd — an MI355X-native re-build of `financial-market-data-analysis`.

A bidirectional-GRU training and real-time streaming-inference engine for
5-minute market-microstructure data, designed from scratch for AMD MI355X
(gfx950, CDNA4):

- the biGRU recurrence runs as hand-written persistent HIP kernels
  (MFMA input projections, LDS-resident recurrent weights, fused gates and
  pooling epilogue) instead of cuDNN/MIOpen RNNs;
- data-parallel training uses one process per GPU with RCCL over xGMI,
  gradient all-reduce overlapped with backward;
- the streaming predict path is hipGraph-captured;
- the Kafka/Spark/MariaDB pipeline of the reference is replaced by a
  seedable synthetic order-book/indicator generator with the exact
  108-feature layout and `norm_params` scaling of the reference
  (reference files: producer.py, spark_consumer.py, create_database.py).

The reference's public model API (`BiGRU` with forward / add_loss_fn /
add_optimizer / add_device / train_model / evaluate_model — reference
biGRU_model.py:8-286) and its checkpoint format (`model_params.pt`
state_dict with keys gru.weight_ih_l0 ... linear.bias) are preserved.
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
from .features import FEATURE_NAMES, N_FEATURES, TARGET_NAMES  # noqa: F401
from .models.bigru import BiGRU  # noqa: F401
