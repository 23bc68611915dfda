#!/usr/bin/env python3
"""A/B the training-step GEMM shapes across blas backends on MI355X.

The round-1 profile (profiles/r01_repo_train_stats.md) showed hipBLASLt
dispatching the gi projection GEMMs with 61 workgroups (256-CU GPU) at
~1254 us. This times each hot shape under the current backend so we can
choose: keep hipBLASLt, switch to rocBLAS (preferred_blas_library
'cublas'), or write a custom MFMA projection kernel.

Run twice:  python scripts/gemm_ab.py            (default = hipBLASLt)
            python scripts/gemm_ab.py --lib cublas   (rocBLAS)
TunableOp:  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
            python scripts/gemm_ab.py --tag tunable
"""
import argparse
import json
import time

import torch

B, T = 4096, 120
M = B * T
SHAPES = [
    # (name, M, N, K, bias)  out = A(M,K) @ W(N,K)^T [+ bias]
    ("gi_l1", M, 768, 96, True),
    ("gi_l2", M, 768, 256, True),
    ("dx_l2", M, 256, 1536, False),
    ("dW_ih_l1.T", 768, 96, M, False),   # dgi^T @ x
    ("dW_ih_l2.T", 1536, 256, M, False),
    ("dW_hh_cross", 768, 256, M, False),  # dgh^T @ out
]


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--lib", default=None, help="cublas|cublaslt|ck")
    ap.add_argument("--tag", default=None)
    args = ap.parse_args()
    if args.lib:
        torch.backends.cuda.preferred_blas_library(args.lib)
    tag = args.tag or (args.lib or "default")

    results = {}
    for name, m, n, k, bias in SHAPES:
        A = torch.randn(m, k, device="cuda").bfloat16()
        W = torch.randn(n, k, device="cuda").bfloat16()
        if bias:
            b = torch.randn(n, device="cuda").bfloat16()
            fn = lambda: torch.addmm(b, A, W.t())
        else:
            fn = lambda: torch.matmul(A, W.t())
        us = timeit(fn)
        gb = (m * k + n * k + m * n) * 2 / 1e9
        tf = 2 * m * n * k / 1e12
        results[name] = round(us, 1)
        print(f"[{tag}] {name}: {us:8.1f} us  "
              f"({gb / (us / 1e6):7.0f} GB/s eff, {tf / (us / 1e6):7.1f} TFLOP/s)",
              flush=True)

    # chunked split-K alternative for the fat-K reductions
    for name, n_out, k_small in [("dW_ih_l1.bmm", 768, 96),
                                 ("dW_hh.bmm", 768, 256)]:
        A = torch.randn(M, n_out, device="cuda").bfloat16()
        Xs = torch.randn(M, k_small, device="cuda").bfloat16()
        C = 64
        Ac = A.view(C, M // C, n_out)
        Xc = Xs.view(C, M // C, k_small)
        fn = lambda: torch.bmm(Ac.transpose(1, 2), Xc).sum(dim=0)
        us = timeit(fn)
        results[name] = round(us, 1)
        print(f"[{tag}] {name}: {us:8.1f} us", flush=True)

    with open(f"gpurun_out/gemm_ab_{tag}.json", "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
