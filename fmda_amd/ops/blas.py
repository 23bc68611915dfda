"""GEMM policy for the training step on MI355X.

Two facts measured on MI355X (profiles/r01_repo_train_stats.md,
gpurun_out/gemm_ab_*.json):

1. hipBLASLt's heuristic picks a 61-workgroup stream-K algo for the
   (B*T, 768) x (768, K) projection GEMMs inside the training step (~1.25 ms
   each, ~5x off the HBM bound), while tuned algo selection reaches
   220-280 us. We therefore ship a TunableOp result file for the gfx950 hot
   shapes and enable TunableOp (tuning off) at engine init.

2. The fat-K gradient reductions dW = dG^T @ X with K = B*T ~ 5e5 are
   dispatched with too few workgroups (49 WGs, ~0.9-1.8 ms). A chunked
   batched GEMM (split-K by hand: C chunks of rows, bmm, fp32 sum) runs at
   150-250 us. `chunked_outer` implements that formulation.
"""
import os


import torch

_TUNABLE_READY = False


def enable_tunableop() -> bool:
    """Load the shipped TunableOp results for gfx950 and enable algo lookup
    (tuning itself stays off unless FMDA_TUNE=1). Safe to call many times."""
    global _TUNABLE_READY
    if _TUNABLE_READY or not torch.cuda.is_available():
        return _TUNABLE_READY
    try:
        tun = torch.cuda.tunable
        csv = os.path.join(os.path.dirname(__file__), "tunableop_gfx950.csv")
        tune = os.environ.get("FMDA_TUNE", "0") == "1"
        # filename must be set before enabling so results load from the repo
        tun.set_filename(csv, insert_device_ordinal=False)
        tun.enable(True)
        tun.tuning_enable(tune)
        _TUNABLE_READY = True
    except Exception:
        _TUNABLE_READY = False
    return _TUNABLE_READY


def chunked_colsum(dg: torch.Tensor, chunks: int = 48) -> torch.Tensor:
    """Column sum of a tall (M, N) matrix in two stages. torch's single
    reduce over M ~ 5e5 rows dispatches 2-3 workgroups on MI355X (185 us);
    the two-stage form parallelizes the first pass (~25 us)."""
    M, N = dg.shape
    c = chunks
    while c > 1 and M % c != 0:
        c //= 2
    if c <= 1:
        return dg.sum(dim=0, dtype=torch.float32)
    # sum(dtype=fp32) accumulates in fp32 WITHOUT materializing an fp32
    # copy of the (M, N) input (a .float() here costs 1.5 GB of traffic).
    return dg.view(c, M // c, N).sum(dim=1, dtype=torch.float32).sum(dim=0)


def chunked_outer(dg: torch.Tensor, x: torch.Tensor,
                  chunks: int = 64, out_fp32: bool = False) -> torch.Tensor:
    """(M, N)^T @ (M, K) -> (N, K) via hand split-K: C row-chunks, batched
    GEMM, fp32 chunk sum. Beats the library's unsplit fat-K GEMM by ~7x on
    MI355X for the training-step dW shapes. out_fp32=True returns the fp32
    chunk sum directly — callers that want fp32 master-grads then skip a
    round trip through the compute dtype (two elementwise kernels each)."""
    M, N = dg.shape
    K = x.shape[1]
    c = chunks
    while c > 1 and M % c != 0:
        c //= 2
    if c <= 1:
        r = torch.matmul(dg.t(), x)
        return r.float() if out_fp32 else r
    parts = torch.bmm(dg.view(c, M // c, N).transpose(1, 2),
                      x.view(c, M // c, K))
    s = parts.float().sum(dim=0)
    return s if out_fp32 else s.to(dg.dtype)
