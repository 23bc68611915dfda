#!/usr/bin/env python3
"""Numerics check: gru_bwd_fused (in-kernel dW_hh) vs gru_bwd + host GEMM."""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from fmda_amd.ops import load_extension
from fmda_amd.ops.blas import chunked_outer

ext = load_extension()
torch.manual_seed(0)
for B in (4096, 300, 33):
    T, Hp, n_dir = 24, 128, 2
    gi = (torch.randn(B, T, n_dir*3*Hp, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(n_dir, 3*Hp, Hp, device="cuda") * 0.2).bfloat16()
    bhh = (torch.randn(n_dir, 3*Hp, device="cuda") * 0.1).float()
    out, hlast = ext.gru_fwd(gi, w, bhh)
    dout = torch.randn_like(out)
    dhT = torch.randn_like(hlast)
    dgi0, dgh, dh00, dbhh0, dbih0 = ext.gru_bwd(gi, w, bhh, out, dout, dhT)
    dgi1, dwhh, dh01, dbhh1, dbih1 = ext.gru_bwd_fused(gi, w, bhh, out, dout, dhT)
    M = B*T
    cross = chunked_outer(dgh.reshape(M, -1), out.reshape(M, -1))
    ref = torch.stack([cross[d*3*Hp:(d+1)*3*Hp, d*Hp:(d+1)*Hp].float()
                       for d in range(n_dir)])
    def rel(a, b):
        return (a.float()-b.float()).norm().item()/max(b.float().norm().item(),1e-30)
    print(f"B={B}: dgi_bitwise={torch.equal(dgi0, dgi1)} "
          f"dh0={rel(dh01, dh00):.2e} dbhh={rel(dbhh1, dbhh0):.2e} "
          f"dbih={rel(dbih1, dbih0):.2e} dwhh_rel={rel(dwhh, ref):.2e} "
          f"dwhh_max={float((dwhh-ref).abs().max()):.3e} "
          f"ref_max={float(ref.abs().max()):.3e}", flush=True)
