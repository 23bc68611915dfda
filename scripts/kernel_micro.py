#!/usr/bin/env python3
"""Microbenchmark the persistent GRU kernels at the repo bench shape.

Times raw ext.gru_fwd / ext.gru_bwd launches (CUDA events, no autograd) so
kernel changes can be iterated without the full bench. Run under rocprofv3
--pmc for counters (own run, no trace domains).
"""
import argparse
import time

import torch

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from fmda_amd.ops import load_extension


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--B", type=int, default=4096)
    ap.add_argument("--T", type=int, default=120)
    ap.add_argument("--H", type=int, default=128)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--what", default="both", help="fwd|bwd|both")
    args = ap.parse_args()
    ext = load_extension()
    B, T, Hp = args.B, args.T, args.H
    n_dir = 2
    torch.manual_seed(0)
    gi = (torch.randn(B, T, n_dir * 3 * Hp, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(n_dir, 3 * Hp, Hp, device="cuda") * 0.2).bfloat16()
    bhh = (torch.randn(n_dir, 3 * Hp, device="cuda") * 0.1).float()
    out, hlast = ext.gru_fwd(gi, w, bhh)
    dout = torch.randn_like(out)
    dhT = torch.zeros_like(hlast)

    def run_fwd():
        return ext.gru_fwd(gi, w, bhh)

    def run_bwd():
        return ext.gru_bwd(gi, w, bhh, out, dout, dhT)

    cases = [("fwd", run_fwd), ("bwd", run_bwd)]

    if args.what == "outer":
        # dW reduction GEMM policy sweep (chunked_outer split-K chunks)
        from fmda_amd.ops.blas import chunked_outer, enable_tunableop
        enable_tunableop()
        M = B * T
        dgh2 = (torch.randn(M, n_dir * 3 * Hp, device="cuda") * 0.1).bfloat16()
        out2 = out.reshape(M, -1)
        cases = []
        for c in (16, 32, 64, 128, 256):
            cases.append((f"dW_hh outer c={c}",
                          lambda c=c: chunked_outer(dgh2, out2, chunks=c)))
        x2d = (torch.randn(M, 2 * Hp, device="cuda") * 0.1).bfloat16()
        for c in (32, 64, 128):
            cases.append((f"dW_ih outer c={c}",
                          lambda c=c: chunked_outer(dgh2, x2d, chunks=c)))

    for name, fn in cases:
        if args.what not in ("both", "outer", name):
            continue
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            fn()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / args.iters * 1e6
        per_step = us / T * 1000.0
        print(f"{name}: {us:9.1f} us  ({per_step:6.1f} ns/step, "
              f"B={B} T={T} Hp={Hp})", flush=True)


if __name__ == "__main__":
    main()
