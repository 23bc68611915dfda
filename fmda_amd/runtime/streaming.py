"""Real-time streaming inference engine (the reference predict.py path).

Semantics of predict.py:124-197, MI355X-native:
- consume timestamp messages from the `predict_timestamp` topic (in-process
  bus instead of Kafka);
- drop stale messages older than `stale_after` (predict.py:135);
- assemble the window of the latest `window` feature rows from a ring buffer
  (replacing the SQL `SELECT ... WHERE sd.ID IN (...)` fetch,
  predict.py:162-166);
- min-max normalize with the norm_params scaling table (predict.py:175);
- forward -> sigmoid -> threshold 0.5 -> publish labels to the `prediction`
  topic (predict.py:177-197).

On GPU the whole tick is GPU-resident (torch.cuda.CUDAGraph == hipGraph on
ROCm). Fast path (bf16, unpadded hidden sizes): the window ring LIVES on
the GPU — push_row uploads one raw 384-byte fp32 row and a fused ingest
kernel shifts + normalizes + casts in place; the captured graph is then
just [gi GEMM -> persistent b1 recurrence] x L -> fused pool/concat ->
fused head+sigmoid (~6 kernels, no eager glue). Other configs fall back to
a captured model.forward with a host-side ring.
"""
import threading
import time
from typing import Dict, List, Optional

import torch

from ..features import TARGET_NAMES
from ..models.bigru import BiGRU
from .bus import MessageBus


class FeatureRing:
    """Ring buffer of the latest feature rows (replaces the warehouse's
    last-N-rows query)."""

    def __init__(self, window: int, n_features: int):
        self.window = window
        self.buf = torch.zeros(window, n_features)
        self.count = 0

    def push(self, row: torch.Tensor) -> None:
        self.buf = torch.roll(self.buf, -1, dims=0)
        self.buf[-1] = row
        self.count += 1

    @property
    def full(self) -> bool:
        return self.count >= self.window


class StreamingPredictor:
    def __init__(self, model: BiGRU, x_min: torch.Tensor, x_max: torch.Tensor,
                 window: int, bus: Optional[MessageBus] = None,
                 prob_threshold: float = 0.5, stale_after: float = 240.0,
                 device: str = "cpu", dtype: torch.dtype = torch.float32,
                 use_graph: Optional[bool] = None,
                 settle_delay: float = 0.0):
        self.model = model.eval().to(device)
        self.window = window
        self.n_features = model.n_features
        self.bus = bus or MessageBus()
        self.prob_threshold = prob_threshold
        self.stale_after = stale_after
        # delayed-data tolerance (predict.py:141-157): wait `settle_delay`
        # for the bar's feature row to land, retry ONCE, then give up
        self.settle_delay = settle_delay
        self.last_row_ts: Optional[float] = None
        self.n_retries = 0
        self.n_dropped_missing = 0
        self.device = torch.device(device)
        self.dtype = dtype
        self.ring = FeatureRing(window, self.n_features)
        # serve.py runs handlers on a threadpool: ingestion and prediction
        # must not interleave on the ring / staging buffers / graph.
        # handle_timestamp's settle-delay sleep stays OUTSIDE the lock so a
        # concurrent push_row can satisfy the retry.
        self._lock = threading.Lock()
        self.x_min = x_min.clone()
        # MIN==MAX guard mirroring the chunk loader's epsilon fix
        # (sql_pytorch_dataloader.py:107-113): a degenerate range must not
        # produce inf/NaN in the normalize (CPU or the ingest kernel)
        self.x_rng = (x_max - x_min).clamp(min=1e-6)
        self.y_fields: List[str] = list(TARGET_NAMES)
        self._graph = None
        self._use_graph = (self.device.type == "cuda"
                           if use_graph is None else use_graph)
        self._static_in = torch.zeros(1, window, self.n_features,
                                      device=self.device, dtype=self.dtype)
        self._static_out = None
        self.n_predictions = 0

        # GPU-resident fast path: ring on device, fused ingest kernel,
        # dedicated captured graph. Requires bf16 + an unpadded hidden size
        # (Hp == H, so the direction-concat layout needs no unpad slicing).
        self._gpu_fast = False
        if self.device.type == "cuda" and dtype == torch.bfloat16:
            from ..ops.interface import _pad_h
            if (_pad_h(model.hidden_size) == model.hidden_size
                    and (window - 1) * self.n_features <= 48 * 1024):
                self._gpu_fast = True
                self._ring_gpu = torch.zeros(window, self.n_features,
                                             device=self.device,
                                             dtype=torch.bfloat16)
                self._row_staging = torch.zeros(self.n_features,
                                                device=self.device)
                self._xmin_gpu = self.x_min.float().to(self.device)
                self._xrng_gpu = self.x_rng.float().to(self.device)

    # ---------------- feature ingestion ----------------

    def push_row(self, row: torch.Tensor,
                 ts: Optional[float] = None) -> None:
        """Ingest one raw (unnormalized) 108-feature row."""
        with self._lock:
            if self._gpu_fast:
                from ..ops import load_extension
                ext = load_extension()
                self._row_staging.copy_(row.float())
                ext.ingest_row(self._ring_gpu, self._row_staging,
                               self._xmin_gpu, self._xrng_gpu)
                self.ring.count += 1
            else:
                self.ring.push(row)
            if ts is not None:
                # published AFTER the row is in the ring: a reader seeing
                # last_row_ts >= ts may predict on the window immediately
                self.last_row_ts = ts

    # ---------------- the batch-1 step ----------------

    def _fast_forward(self):
        """Direct-kernel forward over the GPU ring: gi GEMMs + persistent
        b1 recurrence per layer, fused pool/concat, fused head+sigmoid."""
        from ..ops import load_extension
        from ..ops.interface import _packed_inference_weights, _pad_h
        ext = load_extension()
        m = self.model
        H = m.hidden_size
        Hp = _pad_h(H)
        D = m.n_directions
        T = self.window
        inp = self._ring_gpu          # (T, F) normalized bf16
        out = hl = None
        for layer in range(m.n_layers):
            params = []
            for d, sfx_d in enumerate(["", "_reverse"][:D]):
                sfx = f"l{layer}{sfx_d}"
                params.append((getattr(m.gru, f"weight_ih_{sfx}"),
                               getattr(m.gru, f"weight_hh_{sfx}"),
                               getattr(m.gru, f"bias_ih_{sfx}"),
                               getattr(m.gru, f"bias_hh_{sfx}")))
            w_ih, b_ih, w_hh, b_hh = _packed_inference_weights(
                m.gru, layer, params, Hp, H, self.dtype)
            gi = torch.addmm(b_ih, inp, w_ih.t()).view(1, T, -1)
            out, hl = ext.gru_fwd(gi, w_hh, b_hh)
            inp = out.view(T, -1)
        feat = ext.pool_concat_infer(out, hl, H)
        if not hasattr(self, "_head_wb"):
            self._head_wb = (m.linear.weight.to(self.dtype).contiguous(),
                             m.linear.bias.to(self.dtype).contiguous())
        return ext.head_sigmoid(feat, *self._head_wb)

    def _capture_graph(self):
        fwd = (self._fast_forward if self._gpu_fast else
               lambda: torch.sigmoid(self.model(self._static_in)))
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(3):
                out = fwd()
        torch.cuda.current_stream().wait_stream(s)
        self._pinned_out = None
        if self._gpu_fast:
            # capture the D2H probs copy INTO the graph (pinned dst): the
            # replay then delivers host-readable results with no separate
            # copy enqueue — one replay + one stream sync per tick
            pin = torch.zeros(self.model.output_size, dtype=torch.float32,
                              pin_memory=True)
            try:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g), torch.no_grad():
                    self._static_out = fwd()
                    pin.copy_(self._static_out.reshape(-1),
                              non_blocking=True)
                self._graph = g
                self._pinned_out = pin
                return
            except RuntimeError:
                pass  # D2H capture unsupported: plain graph below
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g), torch.no_grad():
            self._static_out = fwd()
        self._graph = g

    def _forward_probs(self, x_norm: Optional[torch.Tensor]) -> torch.Tensor:
        if self._use_graph:
            if self._graph is None:
                self._capture_graph()
            if not self._gpu_fast:
                self._static_in.copy_(x_norm.to(self.device, self.dtype))
            self._graph.replay()
            return self._static_out.float()
        with torch.no_grad():
            if self._gpu_fast:
                return self._fast_forward().float()
            x = x_norm.to(self.device, self.dtype)
            return torch.sigmoid(self.model(x)).float()

    def predict_window(self) -> Dict:
        """Run inference on the current window; returns the prediction dict
        (shape of predict.py:193-194)."""
        with self._lock:
            return self._predict_window_locked()

    def _predict_window_locked(self) -> Dict:
        if self._gpu_fast and self._use_graph:
            if self._graph is None:
                self._capture_graph()
            self._graph.replay()
            # (an event-query spin measured identical p50 — the tick is
            # GPU-pipeline-bound, not sync-latency-bound)
            torch.cuda.current_stream().synchronize()
            if self._pinned_out is not None:
                pl = self._pinned_out.tolist()
            else:
                pl = self._static_out.reshape(-1).cpu().tolist()
        elif self._gpu_fast:
            with torch.no_grad():
                pl = self._fast_forward().reshape(-1).cpu().tolist()
        else:
            x = self.ring.buf.unsqueeze(0)  # (1, window, F)
            x_norm = (x - self.x_min) / self.x_rng
            pl = self._forward_probs(x_norm).reshape(-1).cpu().tolist()
        idx = [i for i, v in enumerate(pl) if v > self.prob_threshold]
        labels = [self.y_fields[i] for i in idx]
        self.n_predictions += 1
        return {"probabilities": pl,
                "prob_threshold": self.prob_threshold,
                "pred_indices": idx, "pred_labels": labels}

    # ---------------- message loop ----------------

    def handle_timestamp(self, msg: Dict, now: Optional[float] = None) -> Optional[Dict]:
        """Process one predict_timestamp message; returns the published
        prediction dict, or None if stale/insufficient data."""
        ts = float(msg["Timestamp"])
        now = time.time() if now is None else now
        if ts <= now - self.stale_after:    # stale filter (predict.py:135)
            return None
        # delayed-data tolerance (predict.py:141-157): if the bar's
        # feature row hasn't landed yet (producer lag), wait for the
        # settle delay and retry ONCE; drop the message if still missing
        if self.last_row_ts is not None and self.last_row_ts < ts:
            self.n_retries += 1
            if self.settle_delay > 0:
                time.sleep(self.settle_delay)
            if self.last_row_ts < ts:
                self.n_dropped_missing += 1
                return None
        if not self.ring.full:
            return None
        pred = self.predict_window()
        pred["timestamp"] = ts
        self.bus.publish("prediction", pred)
        return pred

    def run(self, max_messages: Optional[int] = None,
            timeout: Optional[float] = 1.0) -> int:
        """Consume predict_timestamp messages until the topic drains (or
        max_messages). The consumer offset persists across calls."""
        topic = self.bus.topic("predict_timestamp")
        if not hasattr(self, "_offset"):
            self._offset = 0
        n = 0
        while True:
            msg = topic.read(self._offset, timeout=timeout)
            if msg is None:
                break
            self._offset += 1
            self.handle_timestamp(msg)
            n += 1
            if max_messages is not None and n >= max_messages:
                break
        return n
