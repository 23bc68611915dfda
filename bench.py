#!/usr/bin/env python3
"""fmda_amd benchmark driver.

Measures the headline metric of BASELINE.json: train seq/sec for the biGRU
on synthetic order-book-shaped data (random-init weights), on N GPUs of one
node (weak scaling: per-GPU batch fixed).

    python bench.py --gpus N --steps K --warmup W [--config repo]

Launched by torchrun for N > 1 (reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*).
Rank 0 prints exactly one JSON line.

The timed step is the full reference training step (biGRU_model.py:162-210):
dropout -> biGRU forward -> 3-way pooled head -> BCEWithLogitsLoss(weight,
pos_weight) -> backward -> gradient all-reduce (N > 1) -> clip_grad_norm(50)
-> Adam step. Nothing is skipped or cached inside the timed region.
"""
import argparse
import json
import os
import time

import torch
import torch.nn as nn

from fmda_amd.config import BENCH_CONFIGS
from fmda_amd.data.generator import synthetic_batch
from fmda_amd.metrics import subset_accuracy
from fmda_amd.models import BiGRU


def three_class_accuracy(target: torch.Tensor, pred: torch.Tensor) -> float:
    """Derived up/down/stall accuracy (SURVEY.md section 6 labeling note)."""
    from fmda_amd.metrics import three_class_accuracy as _acc3
    return float(_acc3(target, pred))


def bench_predict(cfg, args, use_cuda):
    """BASELINE.json configs[4]: batch=1 streaming inference p50 latency,
    hipGraph-captured step on GPU."""
    import numpy as np
    from fmda_amd.runtime import StreamingPredictor
    device = "cuda:0" if use_cuda else "cpu"
    dtype = torch.bfloat16 if use_cuda else torch.float32
    torch.manual_seed(0)
    model = BiGRU(cfg.hidden_size, cfg.n_features, 4, n_layers=cfg.n_layers,
                  spatial_dropout=False, dropout=0.2)
    if dtype == torch.bfloat16:
        model = model.to(dtype)
    pred = StreamingPredictor(model, torch.zeros(cfg.n_features),
                              torch.ones(cfg.n_features), cfg.seq_len,
                              device=device, dtype=dtype)
    g = torch.Generator().manual_seed(1)
    for _ in range(cfg.seq_len):
        pred.push_row(torch.rand(cfg.n_features, generator=g))
    for _ in range(max(args.warmup, 3)):
        pred.predict_window()
    if use_cuda:
        torch.cuda.synchronize()
    lats = []
    for i in range(max(args.steps, 50)):
        pred.push_row(torch.rand(cfg.n_features, generator=g))
        t0 = time.perf_counter()
        pred.predict_window()
        if use_cuda:
            torch.cuda.synchronize()
        lats.append((time.perf_counter() - t0) * 1000.0)
    p50 = float(np.percentile(lats, 50))
    result = {
        "metric": "p50 batch=1 streaming predict latency",
        "value": p50, "unit": "ms", "n_gpus": 1 if use_cuda else 0,
        "steps": len(lats), "warmup": max(args.warmup, 3),
        "ms_per_step": p50, "higher_is_better": False, "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
        "data": "synthetic",
        "config": {"model": f"biGRU L{cfg.n_layers} H{cfg.hidden_size}",
                   "global_batch": 1, "seq_len": cfg.seq_len,
                   "parallelism": "single",
                   "p90_ms": float(np.percentile(lats, 90)),
                   "hipgraph": bool(use_cuda)},
    }
    print(json.dumps(result))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--config", default=None,
                    help="repo|cpu|stress|predict (default: repo on GPU, cpu otherwise)")
    ap.add_argument("--batch", type=int, default=None,
                    help="override per-GPU batch")
    ap.add_argument("--ddp-mode", default=None, choices=["bucketed", "fused"],
                    help="gradient sync mode (default: bucketed; fused for "
                         "the stress config, whose 64 MB payload would "
                         "otherwise split into ~64 latency-bound collectives)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    # --gpus documents the intended N; the actual world size comes from the
    # torchrun env. Catch a mismatched launch early instead of reporting a
    # wrong aggregate.
    if world > 1 and args.gpus not in (1, world):
        raise SystemExit(f"--gpus {args.gpus} != WORLD_SIZE {world}: launch "
                         "with torchrun --nproc-per-node matching --gpus")
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()

    cfg_name = args.config or ("repo" if use_cuda else "cpu")
    cfg = BENCH_CONFIGS[cfg_name]

    if cfg_name == "predict":
        bench_predict(cfg, args, use_cuda)
        return
    batch = args.batch or cfg.batch_per_gpu
    dtype = torch.bfloat16 if (cfg.dtype == "bf16" and use_cuda) else torch.float32

    dist = None
    if world > 1:
        import torch.distributed as tdist
        from fmda_amd.parallel import GradAllReduce, init_distributed
        init_distributed()
        dist = tdist
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
    else:
        device = torch.device("cpu")
        if args.config is None:
            batch = min(batch, 8)

    torch.manual_seed(1234)  # same init on every rank
    model = BiGRU(cfg.hidden_size, cfg.n_features, 4, n_layers=cfg.n_layers,
                  clip=50, dropout=0.5, spatial_dropout=False,
                  bidirectional=True).to(device)

    engine = None
    if world > 1:
        from fmda_amd.parallel import GradAllReduce
        ddp_mode = args.ddp_mode or ("fused" if cfg_name == "stress"
                                     else "bucketed")
        engine = GradAllReduce(model, mode=ddp_mode)

    # Class weighting like the training notebook (cell 16): weight =
    # total/positives, pos_weight = negatives/positives, from the synthetic
    # class rates (~reference rates).
    rates = torch.tensor([0.24, 0.16, 0.25, 0.16], device=device)
    weight = 1.0 / rates
    pos_weight = (1.0 - rates) / rates
    loss_fn = nn.BCEWithLogitsLoss(weight=weight, pos_weight=pos_weight)
    # Fused multi-tensor grad-clip + Adam (HIP kernels on GPU, equivalent
    # eager pair on CPU) — reference semantics: clip_grad_norm_(50) + Adam.
    from fmda_amd.optim import FusedClipAdam
    opt = FusedClipAdam(model.parameters(), lr=1e-3, clip=50.0)

    # Pre-generate a small rotating pool of synthetic batches per rank.
    pool = []
    for i in range(2):
        x, y = synthetic_batch(batch, cfg.seq_len, cfg.n_features,
                               seed=1000 + 7 * rank + i)
        pool.append((x.to(device=device, dtype=dtype), y.to(device)))

    acc_sum = torch.zeros((), device=device)
    acc3_sum = 0.0
    n_acc = 0

    # Optional roctx ranges (FMDA_ROCTX=1): torch's nvtx API emits roctx
    # markers on ROCm, letting `rocprofv3 --marker-trace` segment the step
    # into forward / backward / optimizer phases.
    if os.environ.get("FMDA_ROCTX") == "1" and use_cuda:
        rng_push, rng_pop = torch.cuda.nvtx.range_push, torch.cuda.nvtx.range_pop
    else:
        rng_push = rng_pop = None

    def step(i, measure_quality=False):
        nonlocal acc3_sum, n_acc
        x, y = pool[i % len(pool)]
        # set_to_none=True: autograd ASSIGNS fresh grad tensors (no fill +
        # no accumulate-add kernels, ~40 launches/step saved); the fused
        # optimizer patches its chunk table's grad pointers from a pinned
        # staging copy, so the rebuild costs ~10 KB H2D, not a Python loop.
        opt.zero_grad(set_to_none=True)
        if rng_push:
            rng_push("forward")
        if use_cuda:
            # fused head GEMM + BCEWithLogitsLoss(weight, pos_weight) kernel
            from fmda_amd.ops.interface import fused_head_loss
            feats = model.forward_features(x)
            loss, logits = fused_head_loss(feats, model.linear.weight,
                                           model.linear.bias, y, weight,
                                           pos_weight)
        else:
            logits = model(x)
            loss = loss_fn(logits.float(), y)
        if rng_push:
            rng_pop()
            rng_push("backward")
        loss.backward()
        if engine is not None:
            engine.finalize()
        if rng_push:
            rng_pop()
            rng_push("optimizer")
        opt.step()  # fused clip(50) + Adam
        if rng_push:
            rng_pop()
        pred = torch.sigmoid(logits.detach().float()) > 0.5
        acc_sum.add_(subset_accuracy(y, pred))  # stays on device, no sync
        if measure_quality:
            acc3_sum += three_class_accuracy(y, pred)
        n_acc += 1

    model.train()
    for i in range(args.warmup):
        step(i)

    if dist is not None:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    if dist is not None:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t1 = time.perf_counter()

    dt = torch.tensor([t1 - t0], dtype=torch.float64,
                      device=device if dist is not None and use_cuda else "cpu")
    if dist is not None:
        dist.all_reduce(dt, op=dist.ReduceOp.MAX)
    dt_max = float(dt.item())

    # quality snapshot (outside the timed region), averaged over several
    # steps — a single step's 3-class accuracy is too noisy for a headline
    n_quality = 4
    for i in range(n_quality):
        step(i, measure_quality=True)
    acc = float(acc_sum.item()) / max(n_acc, 1)

    if rank == 0:
        n_gpus = world if (use_cuda or world > 1) else 1
        seq_per_sec = n_gpus * batch * args.steps / dt_max
        result = {
            "metric": "train seq/sec + 3-class acc, biGRU on synthetic order-book",
            "value": seq_per_sec,
            "unit": "seq/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": dt_max / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"biGRU L{cfg.n_layers} H{cfg.hidden_size} bidirectional",
                "global_batch": n_gpus * batch,
                "seq_len": cfg.seq_len,
                "n_features": cfg.n_features,
                "parallelism": f"dp{n_gpus}",
                "bench_config": cfg_name,
                "train_subset_acc": acc,
                "train_3class_acc": acc3_sum / n_quality,
            },
        }
        print(json.dumps(result))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
